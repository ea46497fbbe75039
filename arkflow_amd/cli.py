"""CLI: run the engine, validate configs, discover components.

Mirrors reference crates/arkflow-core/src/cli/mod.rs (:33-135): `arkflow
--config file [--validate]`, subcommands `components list`, `components show
<kind> <name>`, `schema`; logging init (plain/JSON, file/console,
cli/mod.rs:270-338). Entry point: `python -m arkflow_amd ...`.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import logging
import sys
from typing import List, Optional


def init_logging(level: str = "info", fmt: str = "plain",
                 file: Optional[str] = None) -> None:
    lvl = getattr(logging, level.upper(), logging.INFO)
    handlers: List[logging.Handler] = []
    handler = logging.FileHandler(file) if file else logging.StreamHandler()
    if fmt == "json":
        class JsonFormatter(logging.Formatter):
            def format(self, record):
                return json.dumps({
                    "ts": self.formatTime(record),
                    "level": record.levelname.lower(),
                    "target": record.name,
                    "message": record.getMessage(),
                })
        handler.setFormatter(JsonFormatter())
    else:
        handler.setFormatter(logging.Formatter(
            "%(asctime)s %(levelname)-5s %(name)s: %(message)s"))
    handlers.append(handler)
    logging.basicConfig(level=lvl, handlers=handlers, force=True)


def main(argv: Optional[List[str]] = None) -> int:
    import arkflow_amd as af
    from .registry import component_metadata as _cmd, list_components as _lc, build_config_schema as _bcs

    parser = argparse.ArgumentParser(
        prog="arkflow_amd",
        description="MI355X-native stream processing engine")
    sub = parser.add_subparsers(dest="command")

    from . import __version__
    parser.add_argument("--version", action="version",
                        version=f"arkflow_amd {__version__}")
    run_p = sub.add_parser("run", help="run the engine (default)")
    for p in (parser, run_p):
        p.add_argument("--config", "-c", help="YAML/JSON/TOML config file")
        p.add_argument("--validate", action="store_true",
                       help="validate config and exit")

    comp_p = sub.add_parser("components", help="component discovery")
    comp_sub = comp_p.add_subparsers(dest="comp_command")
    comp_sub.add_parser("list")
    show_p = comp_sub.add_parser("show")
    show_p.add_argument("kind")
    show_p.add_argument("name")

    sub.add_parser("schema", help="print the engine config JSON schema")

    hub_p = sub.add_parser("hub", help="serve the fleet hub")
    hub_p.add_argument("--address", default="127.0.0.1:9000")
    hub_p.add_argument("--store", default=":memory:",
                       help="SQLite path for durable hub state")
    hub_p.add_argument("--lease-ttl", type=float, default=15.0)
    hub_p.add_argument("--operator-token", action="append", default=[],
                       help="token:role (role in admin|operator|viewer)")
    hub_p.add_argument("--registration-token", default=None,
                       help="shared secret agents must present to register")

    args = parser.parse_args(argv)

    if args.command == "components":
        if args.comp_command == "show":
            md = _cmd(args.kind, args.name)
            print(json.dumps(md.__dict__, indent=2))
        else:
            for md in _lc():
                print(f"{md.kind:10} {md.name:20} {md.description}")
        return 0
    if args.command == "schema":
        print(json.dumps(_bcs(), indent=2))
        return 0
    if args.command == "hub":
        return _run_hub(args)

    if not args.config:
        parser.error("--config is required to run the engine")
    config = af.EngineConfig.from_file(args.config)
    init_logging(config.logging.level, config.logging.format,
                 config.logging.file)
    errors = config.validate()
    if args.validate:
        if errors:
            for e in errors:
                print(f"error: {e}", file=sys.stderr)
            return 1
        print("configuration OK")
        return 0
    if errors:
        for e in errors:
            print(f"error: {e}", file=sys.stderr)
        return 1

    engine = af.Engine(config)

    async def run_all():
        cancel = asyncio.Event()
        tasks = []
        # rank 0 owns HTTP/metrics in a torchrun launch; peers would
        # collide on the bind address
        if config.server.enabled and engine.rank == 0:
            from .server.api import serve
            tasks.append(asyncio.ensure_future(serve(engine, cancel)))
            if config.server.hub_url:
                # node serves its own API AND joins the fleet hub
                # (reference arkflow-server runs both)
                from .server.agent import agent_run
                tasks.append(asyncio.ensure_future(
                    agent_run(engine, cancel)))
        try:
            await engine.run_with_cancellation(cancel,
                                               install_signal_handlers=True)
        finally:
            cancel.set()
            for t in tasks:
                try:
                    await asyncio.wait_for(t, 10)
                except (asyncio.TimeoutError, Exception):  # noqa: BLE001
                    t.cancel()

    asyncio.run(run_all())
    return 0


if __name__ == "__main__":
    sys.exit(main())


def _run_hub(args) -> int:
    """Serve the fleet hub (reference arkflow-server hub mode)."""
    import uvicorn

    from .server.hub import Hub, create_hub_app, hub_background
    from .server.storage import HubStore

    tokens = {}
    for spec in args.operator_token:
        token, _, role = spec.partition(":")
        tokens[token] = role or "operator"
    hub = Hub(HubStore(args.store), lease_ttl=args.lease_ttl,
              operator_tokens=tokens,
              registration_token=args.registration_token)
    app = create_hub_app(hub)
    host, _, port = args.address.partition(":")

    async def serve():
        cancel = asyncio.Event()
        bg = asyncio.ensure_future(hub_background(hub, cancel))
        config = uvicorn.Config(app, host=host or "127.0.0.1",
                                port=int(port or 9000), log_level="info")
        server = uvicorn.Server(config)
        try:
            await server.serve()
        finally:
            cancel.set()
            await bg

    asyncio.run(serve())
    return 0
