"""arkflow_amd — MI355X-native stream-processing engine.

A from-scratch rebuild of the capabilities of arkflow-rs/arkflow (reference
at /root/reference) designed for AMD Instinct MI355X (gfx950, CDNA4):
GPU-resident columnar batches in HBM3E, hand-written HIP kernels for the SQL
and ML-inference processors, device ring-buffer windows, and RCCL over xGMI
for multi-GPU sharding. Host orchestration (YAML config, stream supervision,
HTTP control plane, WAL) is Python asyncio + C++/HIP extensions.
"""
__version__ = "0.1.0"

from .batch import (  # noqa: F401
    Column,
    DEFAULT_BINARY_VALUE_FIELD,
    DEFAULT_RECORD_BATCH,
    MessageBatch,
    concat_batches,
    split_batch,
)
from .config import EngineConfig, StreamConfig  # noqa: F401
from .engine import Engine  # noqa: F401
from .registry import (  # noqa: F401
    build_component,
    build_config_schema,
    component_metadata,
    list_components,
    register,
    registry,
)
from .spi import (  # noqa: F401
    Ack,
    Buffer,
    Codec,
    Input,
    NoopAck,
    Output,
    Processor,
    Temporary,
    VecAck,
)

_initialized = False


def init() -> None:
    """Populate the global builder registries (reference main.rs:23-29
    plugin init()s). Importing the package calls this automatically."""
    global _initialized
    if _initialized:
        return
    _initialized = True
    from .inputs import generate, memory  # noqa: F401
    from .outputs import basic  # noqa: F401
    # wider component families register on import; keep additive
    for mod in (
        "arkflow_amd.processors.sql",
        "arkflow_amd.processors.json_proc",
        "arkflow_amd.processors.batch_proc",
        "arkflow_amd.processors.inference",
        "arkflow_amd.processors.python_proc",
        "arkflow_amd.processors.protobuf_proc",
        "arkflow_amd.processors.expr_proc",
        "arkflow_amd.processors.repartition",
        "arkflow_amd.buffers.memory_buffer",
        "arkflow_amd.buffers.windows",
        "arkflow_amd.codecs.json_codec",
        "arkflow_amd.codecs.protobuf_codec",
        "arkflow_amd.inputs.file",
        "arkflow_amd.inputs.http",
        "arkflow_amd.inputs.multiple",
        "arkflow_amd.inputs.brokers",
        "arkflow_amd.inputs.sql_io",
        "arkflow_amd.inputs.websocket",
        "arkflow_amd.inputs.modbus",
        "arkflow_amd.codecs.debezium",
        "arkflow_amd.codecs.schema_registry",
        "arkflow_amd.outputs.file",
        "arkflow_amd.outputs.http",
        "arkflow_amd.outputs.influxdb",
        "arkflow_amd.temporary.memory_table",
        "arkflow_amd.wal.store",
        "arkflow_amd.wal.segment_store",
    ):
        try:
            __import__(mod)
        except ImportError:
            pass  # optional families appear as they are built


init()
