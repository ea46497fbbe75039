"""ControlPlane: async lifecycle operations + config versioning.

Mirrors reference crates/arkflow-core/src/{control_plane.rs,configuration.rs}:
lifecycle ops with OperationStore bookkeeping and timeout
(control_plane.rs:235-369), config get/validate/apply/rollback/versions
(:390-443), and secret redaction (configuration.rs:209-259).
"""
from __future__ import annotations

import asyncio
import json
import os
import time
from typing import List, Optional

from .config import EngineConfig
from .errors import ArkError, ConfigError
from .runtime import OperationState

LIFECYCLE_TIMEOUT_SECS = 30.0
SECRET_KEYS = ("token", "password", "secret", "key", "credential")


def redact_secrets(obj):
    """configuration.rs:209-259."""
    if isinstance(obj, dict):
        out = {}
        for k, v in obj.items():
            if any(s in k.lower() for s in SECRET_KEYS) and isinstance(v, str):
                out[k] = "***"
            else:
                out[k] = redact_secrets(v)
        return out
    if isinstance(obj, list):
        return [redact_secrets(v) for v in obj]
    return obj


class ConfigVersionStore:
    """On-disk config version store (configuration.rs:44-112)."""

    def __init__(self, path: Optional[str] = None):
        self.path = path
        self.versions: List[dict] = []
        if path and os.path.isfile(path):
            try:
                with open(path) as f:
                    self.versions = json.load(f)
            except (json.JSONDecodeError, OSError):
                self.versions = []

    def append(self, raw_config: dict, note: str = "") -> int:
        version = (self.versions[-1]["version"] + 1) if self.versions else 1
        self.versions.append({
            "version": version, "config": raw_config,
            "note": note, "ts": time.time(),
        })
        self._flush()
        return version

    def get(self, version: int) -> Optional[dict]:
        for v in self.versions:
            if v["version"] == version:
                return v
        return None

    def list(self) -> List[dict]:
        return [
            {"version": v["version"], "note": v["note"], "ts": v["ts"]}
            for v in self.versions
        ]

    def _flush(self) -> None:
        if self.path:
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            with open(self.path, "w") as f:
                json.dump(self.versions, f)


class ControlPlane:
    def __init__(self, engine, version_store_path: Optional[str] = None):
        self.engine = engine
        self.versions = ConfigVersionStore(version_store_path)

    @property
    def runtime(self):
        return self.engine.runtime

    # ------------------------------------------------------------- lifecycle
    async def lifecycle(self, stream_id: str, op: str,
                        timeout: float = LIFECYCLE_TIMEOUT_SECS) -> dict:
        """start/stop/restart with operation bookkeeping
        (control_plane.rs:235-369)."""
        if op not in ("start", "stop", "restart"):
            raise ArkError(f"unknown lifecycle op {op!r}")
        record = self.runtime.operations.create(stream_id, op)
        record.state = OperationState.RUNNING
        fn = {"start": self.runtime.start, "stop": self.runtime.stop,
              "restart": self.runtime.restart}[op]
        try:
            await asyncio.wait_for(fn(stream_id), timeout)
            self.runtime.operations.finish(record.id, OperationState.SUCCEEDED)
        except asyncio.TimeoutError:
            self.runtime.operations.finish(record.id, OperationState.TIMED_OUT,
                                           "timeout")
        except Exception as e:  # noqa: BLE001
            self.runtime.operations.finish(record.id, OperationState.FAILED,
                                           str(e))
        return self.runtime.operations.get(record.id).to_dict()

    # ----------------------------------------------------------------- state
    def system_status(self) -> dict:
        states = [e.state.value for e in self.runtime.entries.values()]
        return {
            "ready": self.engine.ready,
            "running": self.engine.running,
            "streams": len(states),
            "states": {s: states.count(s) for s in set(states)},
        }

    def stream_snapshots(self) -> List[dict]:
        return self.runtime.list_streams()

    # ---------------------------------------------------------------- config
    def validate_config(self, raw: dict) -> dict:
        """Parse candidate + structural validation report
        (configuration.rs:176)."""
        try:
            cfg = EngineConfig.from_dict(raw)
        except ConfigError as e:
            return {"valid": False, "errors": [str(e)]}
        errors = cfg.validate()
        return {"valid": not errors, "errors": errors}

    async def apply_configuration(self, raw: dict, note: str = "") -> dict:
        """Validate, version, diff-and-restart changed streams
        (control_plane.rs:390-443)."""
        report = self.validate_config(raw)
        if not report["valid"]:
            return {"applied": False, **report}
        new_cfg = EngineConfig.from_dict(raw)
        version = self.versions.append(redact_secrets(raw), note)
        new_by_id = {s.id: s for s in new_cfg.streams}
        old_ids = set(self.runtime.entries)
        # removed streams
        for sid in old_ids - set(new_by_id):
            await self.runtime.stop(sid)
            del self.runtime.entries[sid]
        # added / changed streams
        for sid, sc in new_by_id.items():
            if sid not in self.runtime.entries:
                self.runtime.register(sc)
                await self.runtime.start(sid)
            elif self.runtime.entries[sid].config != sc:
                await self.runtime.replace_config(sid, sc)
        self.engine.config = new_cfg
        return {"applied": True, "version": version, "errors": []}

    def diff_config(self, raw: dict) -> dict:
        """Structured diff of a candidate config against the running one
        (reference node API config diff route, server/src/lib.rs router)."""
        from dataclasses import asdict
        current = {s.id: asdict(s) for s in self.engine.config.streams}
        candidate = {s.get("id", f"?{i}"): s
                     for i, s in enumerate(raw.get("streams", []))}
        added = sorted(set(candidate) - set(current))
        removed = sorted(set(current) - set(candidate))
        changed = []
        for sid in sorted(set(current) & set(candidate)):
            cur = redact_secrets(current[sid])
            cand = redact_secrets(candidate[sid])
            fields = sorted(k for k in set(cur) | set(cand)
                            if cur.get(k) != cand.get(k)
                            and (cur.get(k) or cand.get(k)))
            if fields:
                changed.append({"stream_id": sid, "fields": fields})
        return {"added": added, "removed": removed, "changed": changed,
                "unchanged": sorted(
                    sid for sid in set(current) & set(candidate)
                    if not any(c["stream_id"] == sid for c in changed))}

    async def rollback(self, version: int) -> dict:
        entry = self.versions.get(version)
        if entry is None:
            return {"applied": False, "errors": [f"unknown version {version}"]}
        return await self.apply_configuration(
            entry["config"], note=f"rollback to {version}")
