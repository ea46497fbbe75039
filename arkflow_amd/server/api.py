"""Node HTTP API: system/status/streams/lifecycle/operations/events/
configuration/components/schema/metrics routes.

Mirrors reference crates/arkflow-server/src/lib.rs (router :183-239,
hand-rendered Prometheus text :2048-2065, SSE events :1113, health/readiness/
liveness). FastAPI app served by uvicorn; `create_app(engine)` is also
directly testable via httpx ASGI transport (the reference tests its axum
Router with tower oneshot the same way).
"""
from __future__ import annotations

import asyncio
import json
from typing import Optional

from fastapi import Depends, FastAPI, HTTPException, Request
from fastapi.responses import PlainTextResponse, StreamingResponse

from ..registry import (
    build_config_schema,
    component_metadata,
    list_components,
)
from ..config import ServerConfig


def create_app(engine, server_config: Optional[ServerConfig] = None) -> FastAPI:
    cfg = server_config or engine.config.server
    cp = engine.control_plane
    prefix = cfg.api_prefix.rstrip("/")
    app = FastAPI(title="arkflow_amd node API", version="1")
    if cfg.cors:
        from fastapi.middleware.cors import CORSMiddleware
        app.add_middleware(CORSMiddleware, allow_origins=["*"],
                           allow_methods=["*"], allow_headers=["*"])

    def auth(request: Request):
        if cfg.token:
            import hmac
            header = request.headers.get("authorization", "")
            if not hmac.compare_digest(header, f"Bearer {cfg.token}"):
                raise HTTPException(401, "unauthorized")

    # ---- health (unauthenticated, reference health endpoints) --------------
    @app.get("/health")
    async def health():
        return {"status": "ok"}

    @app.get("/readiness")
    async def readiness():
        if not engine.ready:
            raise HTTPException(503, "not ready")
        return {"ready": True}

    @app.get("/liveness")
    async def liveness():
        return {"alive": True}

    # ---- system -------------------------------------------------------------
    @app.get(f"{prefix}/system/status", dependencies=[Depends(auth)])
    async def system_status():
        return cp.system_status()

    # ---- streams ------------------------------------------------------------
    @app.get(f"{prefix}/streams", dependencies=[Depends(auth)])
    async def streams():
        return cp.stream_snapshots()

    @app.get(f"{prefix}/streams/{{stream_id}}", dependencies=[Depends(auth)])
    async def stream(stream_id: str):
        try:
            return engine.runtime.get(stream_id).snapshot()
        except Exception as e:  # noqa: BLE001
            raise HTTPException(404, str(e)) from e

    @app.post(f"{prefix}/streams/{{stream_id}}/{{op}}",
              dependencies=[Depends(auth)])
    async def lifecycle(stream_id: str, op: str):
        if op not in ("start", "stop", "restart"):
            raise HTTPException(400, f"unknown op {op}")
        if stream_id not in engine.runtime.entries:
            raise HTTPException(404, f"unknown stream {stream_id}")
        return await cp.lifecycle(stream_id, op)

    # ---- operations ---------------------------------------------------------
    @app.get(f"{prefix}/streams/{{stream_id}}/metrics",
             dependencies=[Depends(auth)])
    async def stream_metrics(stream_id: str):
        from ..errors import ArkError
        try:
            e = engine.runtime.get(stream_id)
        except ArkError:
            raise HTTPException(404, f"unknown stream {stream_id!r}")
        return e.metrics.snapshot()

    @app.get(f"{prefix}/operations", dependencies=[Depends(auth)])
    async def operations(limit: int = 100):
        return [o.to_dict() for o in engine.runtime.operations.list(limit)]

    @app.get(f"{prefix}/operations/{{op_id}}", dependencies=[Depends(auth)])
    async def operation(op_id: str):
        op = engine.runtime.operations.get(op_id)
        if op is None:
            raise HTTPException(404, "unknown operation")
        return op.to_dict()

    # ---- events -------------------------------------------------------------
    @app.get(f"{prefix}/events", dependencies=[Depends(auth)])
    async def events(after_seq: int = 0, limit: int = 100):
        return [e.to_dict() for e in engine.runtime.events.list(after_seq,
                                                                limit)]

    @app.get(f"{prefix}/events/stream", dependencies=[Depends(auth)])
    async def events_stream(request: Request):
        """SSE with Last-Event-ID resume (reference lib.rs:1113)."""
        last_id = int(request.headers.get("last-event-id", 0) or 0)

        async def gen():
            for ev in engine.runtime.events.list(after_seq=last_id,
                                                 limit=10_000):
                yield f"id: {ev.seq}\ndata: {json.dumps(ev.to_dict())}\n\n"
            q = engine.runtime.events.subscribe()
            try:
                from ..aio import queue_get
                while True:
                    if await request.is_disconnected():
                        return
                    got, ev = await queue_get(q, 5.0)
                    if got:
                        yield (f"id: {ev.seq}\n"
                               f"data: {json.dumps(ev.to_dict())}\n\n")
                    else:
                        yield ": keepalive\n\n"
            finally:
                engine.runtime.events.unsubscribe(q)

        return StreamingResponse(gen(), media_type="text/event-stream")

    # ---- configuration ------------------------------------------------------
    @app.get(f"{prefix}/configuration", dependencies=[Depends(auth)])
    async def get_configuration():
        from ..control_plane import redact_secrets
        from dataclasses import asdict
        return redact_secrets({
            "streams": [asdict(s) for s in engine.config.streams],
        })

    @app.post(f"{prefix}/configuration/validate", dependencies=[Depends(auth)])
    async def validate_configuration(body: dict):
        return cp.validate_config(body)

    @app.post(f"{prefix}/configuration/diff", dependencies=[Depends(auth)])
    async def diff_configuration(body: dict):
        return cp.diff_config(body)

    @app.post(f"{prefix}/configuration/apply", dependencies=[Depends(auth)])
    async def apply_configuration(body: dict):
        return await cp.apply_configuration(body)

    @app.post(f"{prefix}/configuration/rollback/{{version}}",
              dependencies=[Depends(auth)])
    async def rollback(version: int):
        return await cp.rollback(version)

    @app.get(f"{prefix}/configuration/versions", dependencies=[Depends(auth)])
    async def versions():
        return cp.versions.list()

    @app.get(f"{prefix}/configuration/versions/{{version}}",
             dependencies=[Depends(auth)])
    async def version_detail(version: int):
        from ..control_plane import redact_secrets
        v = cp.versions.get(version)
        if v is None:
            raise HTTPException(404, f"no config version {version}")
        return redact_secrets(v)

    # ---- components ----------------------------------------------------------
    @app.get(f"{prefix}/components", dependencies=[Depends(auth)])
    async def components(kind: Optional[str] = None):
        return [md.__dict__ for md in list_components(kind)]

    @app.get(f"{prefix}/components/{{kind}}/{{name}}",
             dependencies=[Depends(auth)])
    async def component(kind: str, name: str):
        try:
            return component_metadata(kind, name).__dict__
        except Exception as e:  # noqa: BLE001
            raise HTTPException(404, str(e)) from e

    @app.get(f"{prefix}/schema", dependencies=[Depends(auth)])
    async def schema():
        return build_config_schema()

    # ---- metrics (hand-rendered Prometheus text, reference lib.rs:2048) ------
    @app.get("/metrics", response_class=PlainTextResponse)
    async def metrics():
        return render_prometheus(engine)

    from .console import mount_console
    mount_console(app)
    return app


def render_prometheus(engine) -> str:
    lines = [
        "# HELP arkflow_stream_info per-stream state",
        "# TYPE arkflow_stream_info gauge",
    ]
    counters = (
        "input_batches", "input_messages", "processing_errors",
        "output_batches", "output_messages", "input_errors",
        "input_reconnects", "output_errors", "restarts",
    )
    for name in counters:
        lines.append(f"# TYPE arkflow_{name}_total counter")
    for entry in engine.runtime.entries.values():
        sid = entry.stream_id
        snap = entry.metrics.snapshot()
        state = entry.state.value
        lines.append(
            f'arkflow_stream_info{{stream="{sid}",state="{state}"}} 1')
        for name in counters:
            lines.append(
                f'arkflow_{name}_total{{stream="{sid}"}} {snap[name]}')
        lines.append(
            f'arkflow_stream_uptime_seconds{{stream="{sid}"}} '
            f'{snap["uptime_secs"]:.3f}')
        lines.append(
            f'arkflow_wal_lag{{stream="{sid}"}} {snap.get("wal_lag", 0)}')
        for stage, ms in snap["stage_ms"].items():
            lines.append(
                f'arkflow_stage_ms_total{{stream="{sid}",stage="{stage}"}} '
                f'{ms:.3f}')
    return "\n".join(lines) + "\n"


async def serve(engine, cancel: Optional[asyncio.Event] = None) -> None:
    """Bind and serve the node API (reference serve(), lib.rs:242)."""
    import uvicorn
    cfg = engine.config.server
    host, _, port = cfg.address.partition(":")
    app = create_app(engine)
    config = uvicorn.Config(app, host=host or "127.0.0.1",
                            port=int(port or 8111), log_level="warning")
    server = uvicorn.Server(config)
    if cancel is None:
        await server.serve()
        return
    task = asyncio.ensure_future(server.serve())
    await cancel.wait()
    server.should_exit = True
    await task
