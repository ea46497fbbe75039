"""Error taxonomy (reference crates/arkflow-core/src/lib.rs Error enum)."""


class ArkError(Exception):
    """Base error."""


class ConfigError(ArkError):
    """Invalid configuration."""


class ConnectionError_(ArkError):
    """Component failed to connect."""


class DisconnectionError(ArkError):
    """Transient disconnection — the stream loop reconnects with backoff
    (reference stream/mod.rs:289-306)."""


class EOFError_(ArkError):
    """Input exhausted — stream winds down cleanly (reference Error::EOF)."""


class ProcessError(ArkError):
    """Processor failed on a batch — routed to error_output."""


class ReadError(ArkError):
    """Input read failed (non-fatal)."""


class GpuExtensionMissing(ArkError):
    """A GPU is visible but the native HIP extension is not importable.

    Ops raise this instead of silently falling back to eager PyTorch so a GPU
    test can never pass on a non-native path.
    """
