"""Engine configuration: YAML/JSON/TOML → EngineConfig.

Mirrors reference crates/arkflow-core/src/config.rs (EngineConfig, stream list,
logging, health_check/server block) and stream/mod.rs:1452+ (StreamConfig).
Component blobs stay opaque dicts dispatched by ``type:`` through the
registries, exactly like the reference's flattened serde_json::Value.
"""
from __future__ import annotations

import json
import os
import re
from dataclasses import dataclass, field
from typing import List, Optional

import yaml

from .errors import ConfigError

_STREAM_ID_RE = re.compile(r"^[A-Za-z0-9._~-]+$")  # URL-safe (stream/mod.rs:1476)


@dataclass
class PipelineConfig:
    thread_num: int = 0  # 0 → os.cpu_count() (reference pipeline/mod.rs:97)
    processors: List[dict] = field(default_factory=list)

    def resolved_thread_num(self) -> int:
        return self.thread_num if self.thread_num > 0 else (os.cpu_count() or 4)


@dataclass
class DurabilityConfig:
    """WAL config subset (reference wal/mod.rs:78-233 WalConfig)."""
    enabled: bool = False
    path: str = "./wal"
    sync_policy: str = "group_commit"  # per_entry | group_commit | periodic
    group_window_ms: int = 5
    periodic_interval_ms: int = 200
    backend: str = "local"  # local | segment
    extra: dict = field(default_factory=dict)


@dataclass
class StreamConfig:
    id: str
    input: dict = field(default_factory=dict)
    pipeline: PipelineConfig = field(default_factory=PipelineConfig)
    output: dict = field(default_factory=dict)
    error_output: Optional[dict] = None
    buffer: Optional[dict] = None
    temporary: List[dict] = field(default_factory=list)
    durability: Optional[DurabilityConfig] = None
    device: Optional[str] = None  # "cuda:0" pins GPU placement; None = auto
    dedicated_thread: bool = False  # own event loop in a thread


@dataclass
class LoggingConfig:
    level: str = "info"
    format: str = "plain"  # plain | json
    file: Optional[str] = None


@dataclass
class ServerConfig:
    """reference HealthCheckConfig (config.rs:56-93): node HTTP API."""
    enabled: bool = False
    address: str = "127.0.0.1:8111"
    api_prefix: str = "/api/v1"
    token: Optional[str] = None
    cors: bool = False
    hub_url: Optional[str] = None
    node_id: Optional[str] = None
    node_token: Optional[str] = None
    node_labels: dict = field(default_factory=dict)
    lease_ttl_secs: float = 15.0
    config_store: Optional[str] = None  # persist config versions for rollback


def _interpolate_env(text: str) -> str:
    """``${VAR}`` / ``${VAR:-default}`` substitution in config files — lets
    one YAML serve every rank of a torchrun launch (per-rank output paths,
    ports). Unset vars without a default are left untouched."""
    import os
    import re

    def sub(m):
        name, sep, default = m.group(1).partition(":-")
        val = os.environ.get(name)
        if val is not None:
            return val
        return default if sep else m.group(0)

    return re.sub(r"\$\{([A-Za-z_][A-Za-z0-9_]*(?::-[^}]*)?)\}", sub, text)


@dataclass
class EngineConfig:
    streams: List[StreamConfig] = field(default_factory=list)
    logging: LoggingConfig = field(default_factory=LoggingConfig)
    server: ServerConfig = field(default_factory=ServerConfig)

    # ------------------------------------------------------------------ load
    @staticmethod
    def from_file(path: str) -> "EngineConfig":
        with open(path, "r") as f:
            text = f.read()
        text = _interpolate_env(text)
        if path.endswith((".yaml", ".yml")):
            raw = yaml.safe_load(text)
        elif path.endswith(".json"):
            raw = json.loads(text)
        elif path.endswith(".toml"):
            import tomli
            raw = tomli.loads(text)
        else:
            raw = yaml.safe_load(text)
        return EngineConfig.from_dict(raw)

    @staticmethod
    def from_dict(raw: dict) -> "EngineConfig":
        if not isinstance(raw, dict):
            raise ConfigError("top-level config must be a mapping")
        streams_raw = raw.get("streams")
        if not isinstance(streams_raw, list) or not streams_raw:
            raise ConfigError("config requires a non-empty 'streams' list")
        streams = []
        seen = set()
        for i, s in enumerate(streams_raw):
            sc = _parse_stream(s, i)
            if sc.id in seen:
                raise ConfigError(f"duplicate stream id {sc.id!r}")
            seen.add(sc.id)
            streams.append(sc)
        log_raw = raw.get("logging") or {}
        if not isinstance(log_raw, dict):
            raise ConfigError("'logging' must be a mapping")
        logging_cfg = LoggingConfig(
            level=log_raw.get("level", "info"),
            format=log_raw.get("format", "plain"),
            file=log_raw.get("file"),
        )
        srv_raw = raw.get("health_check") or raw.get("server") or {}
        if not isinstance(srv_raw, dict):
            raise ConfigError("'server' must be a mapping")
        server_cfg = ServerConfig(
            enabled=bool(srv_raw.get("enabled", bool(srv_raw))),
            address=srv_raw.get("address", "127.0.0.1:8111"),
            api_prefix=srv_raw.get("api_prefix", "/api/v1"),
            token=srv_raw.get("token"),
            cors=bool(srv_raw.get("cors", False)),
            hub_url=srv_raw.get("hub_url"),
            node_id=srv_raw.get("node_id"),
            node_token=srv_raw.get("node_token"),
            node_labels=srv_raw.get("node_labels") or {},
            lease_ttl_secs=float(srv_raw.get("lease_ttl_secs", 15.0)),
            config_store=srv_raw.get("config_store"),
        )
        return EngineConfig(streams, logging_cfg, server_cfg)

    def validate(self) -> List[str]:
        """Structural validation against the registries; returns error list
        (reference configuration.rs:176 validate_config)."""
        from .registry import registry as _registry

        def check_types(where: str, kind: str, t: str, spec: dict) -> None:
            """Primitive-type check against the component's JSON Schema
            (additionalProperties stays open; only declared keys checked)."""
            md = _registry(kind).metadata.get(t)
            schema = (md.config_schema or {}) if md else {}
            checks = {"string": str, "integer": int, "number": (int, float),
                      "boolean": bool, "array": list, "object": dict}
            for key, sub in (schema.get("properties") or {}).items():
                if key == "type" or key not in spec:
                    continue
                expect = checks.get(sub.get("type"))
                val = spec[key]
                if sub.get("type") == "string" and isinstance(val, dict) \
                        and "expr" in val:
                    continue  # Expr<T>: per-row SQL expression accepted
                    # wherever a string constant is (reference expr/mod.rs)
                if expect and not isinstance(val, expect) \
                        or (expect is int and isinstance(val, bool)):
                    errors.append(
                        f"{where}: {kind} {t!r} field {key!r} expects "
                        f"{sub.get('type')}, got {type(val).__name__}")

        errors: List[str] = []
        for s in self.streams:
            where = f"stream {s.id!r}"
            for kind, spec in (
                ("input", s.input), ("output", s.output),
                ("output", s.error_output), ("buffer", s.buffer),
            ):
                if spec is None:
                    continue
                t = spec.get("type") if isinstance(spec, dict) else None
                if t is not None and not isinstance(t, str):
                    errors.append(f"{where}: {kind} 'type' must be a string")
                    continue
                if not t:
                    errors.append(f"{where}: {kind} missing 'type'")
                elif t not in _registry(kind).builders:
                    errors.append(f"{where}: unknown {kind} type {t!r}")
                else:
                    check_types(where, kind, t, spec)
                codec_spec = spec.get("codec") if isinstance(spec, dict) \
                    else None
                if codec_spec is not None:
                    ct = codec_spec.get("type") \
                        if isinstance(codec_spec, dict) else None
                    if ct is not None and not isinstance(ct, str):
                        errors.append(
                            f"{where}: codec 'type' must be a string")
                        ct = None
                        continue
                    if not ct:
                        errors.append(f"{where}: codec missing 'type'")
                    elif ct not in _registry("codec").builders:
                        errors.append(f"{where}: unknown codec type {ct!r}")
            for p in s.pipeline.processors:
                t = p.get("type") if isinstance(p, dict) else None
                if t is not None and not isinstance(t, str):
                    errors.append(f"{where}: processor 'type' must be a "
                                  "string")
                    continue
                if not t:
                    errors.append(f"{where}: processor missing 'type'")
                elif t not in _registry("processor").builders:
                    errors.append(f"{where}: unknown processor type {t!r}")
                else:
                    check_types(where, "processor", t, p)
            for t_ in s.temporary:
                tt = t_.get("type") if isinstance(t_, dict) else None
                if tt and tt not in _registry("temporary").builders:
                    errors.append(f"{where}: unknown temporary type {tt!r}")
        return errors


def _parse_stream(raw: dict, index: int) -> StreamConfig:
    if not isinstance(raw, dict):
        raise ConfigError(f"streams[{index}] must be a mapping")
    sid = str(raw.get("id") or f"stream-{index}")
    if not _STREAM_ID_RE.match(sid):
        raise ConfigError(f"stream id {sid!r} must be URL-safe")
    if "input" not in raw:
        raise ConfigError(f"stream {sid!r}: missing input")
    if "output" not in raw:
        raise ConfigError(f"stream {sid!r}: missing output")
    p_raw = raw.get("pipeline") or {}
    if not isinstance(p_raw, dict):
        raise ConfigError(f"stream {sid!r}: 'pipeline' must be a mapping")
    try:
        pipeline = PipelineConfig(
            thread_num=int(p_raw.get("thread_num") or 0),
            processors=list(p_raw.get("processors") or []),
        )
    except (TypeError, ValueError) as e:
        raise ConfigError(f"stream {sid!r}: bad pipeline config: {e}")
    dur = None
    d_raw = raw.get("durability")
    if d_raw is not None and not isinstance(d_raw, dict):
        raise ConfigError(f"stream {sid!r}: 'durability' must be a mapping")
    if d_raw:
        known = {"enabled", "path", "sync_policy", "group_window_ms",
                 "periodic_interval_ms", "backend"}
        try:
            dur = DurabilityConfig(
                enabled=bool(d_raw.get("enabled", True)),
                path=str(d_raw.get("path", "./wal")),
                sync_policy=str(d_raw.get("sync_policy", "group_commit")),
                group_window_ms=int(d_raw.get("group_window_ms") or 5),
                periodic_interval_ms=int(
                    d_raw.get("periodic_interval_ms") or 200),
                backend=str(d_raw.get("backend", "local")),
                extra={k: v for k, v in d_raw.items() if k not in known},
            )
        except (TypeError, ValueError) as e:
            raise ConfigError(f"stream {sid!r}: bad durability config: {e}")
        if dur.sync_policy not in ("per_entry", "group_commit", "periodic"):
            raise ConfigError(
                f"stream {sid!r}: invalid sync_policy {dur.sync_policy!r}"
            )
    temps = raw.get("temporary") or []
    if isinstance(temps, dict):
        temps = [temps]
    return StreamConfig(
        id=sid,
        input=raw["input"],
        pipeline=pipeline,
        output=raw["output"],
        error_output=raw.get("error_output"),
        buffer=raw.get("buffer"),
        temporary=list(temps),
        durability=dur,
        device=raw.get("device"),
        dedicated_thread=bool(raw.get("dedicated_thread", False)),
    )
