"""Node HTTP API tests via ASGI transport (the reference tests its axum
Router with tower::oneshot the same way — no network)."""
import asyncio
import json

import httpx
import pytest

import arkflow_amd as af
from arkflow_amd.config import EngineConfig
from arkflow_amd.server.api import create_app


def _engine():
    cfg = EngineConfig.from_dict({
        "streams": [{
            "id": "s1",
            "input": {"type": "generate", "batch_size": 5, "interval": "10ms",
                      "fields": {"v": {"dtype": "float32"}}},
            "pipeline": {"thread_num": 1, "processors": [
                {"type": "sql", "query": "SELECT * FROM flow WHERE v >= 0"},
            ]},
            "output": {"type": "drop"},
        }],
        "server": {"enabled": True, "address": "127.0.0.1:0"},
    })
    return af.Engine(cfg)


def test_api_surface(run):
    async def main():
        eng = _engine()
        for sc in eng.config.streams:
            eng.runtime.register(sc)
        await eng.runtime.start("s1")
        eng.ready = True
        eng.running = True
        app = create_app(eng)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://node") as c:
            assert (await c.get("/health")).json() == {"status": "ok"}
            assert (await c.get("/readiness")).status_code == 200
            assert (await c.get("/liveness")).status_code == 200

            r = await c.get("/api/v1/system/status")
            assert r.json()["streams"] == 1

            streams = (await c.get("/api/v1/streams")).json()
            assert streams[0]["id"] == "s1"
            assert streams[0]["state"] == "running"

            one = (await c.get("/api/v1/streams/s1")).json()
            assert one["convergence"] == "converged"
            assert (await c.get("/api/v1/streams/nope")).status_code == 404

            # lifecycle: stop → snapshot stopped → restart
            op = (await c.post("/api/v1/streams/s1/stop")).json()
            assert op["state"] in ("succeeded", "running")
            one = (await c.get("/api/v1/streams/s1")).json()
            assert one["state"] == "stopped"
            ops = (await c.get("/api/v1/operations")).json()
            assert len(ops) >= 1
            opd = (await c.get(f"/api/v1/operations/{ops[-1]['id']}")).json()
            assert opd["stream_id"] == "s1"

            evs = (await c.get("/api/v1/events")).json()
            kinds = {e["kind"] for e in evs}
            assert {"registered", "running", "stopping"} <= kinds

            comps = (await c.get("/api/v1/components")).json()
            names = {(m["kind"], m["name"]) for m in comps}
            assert ("processor", "sql") in names
            assert ("input", "generate") in names
            md = (await c.get("/api/v1/components/processor/sql")).json()
            assert "query" in str(md["example"])

            schema = (await c.get("/api/v1/schema")).json()
            assert "streams" in schema["properties"]

            # config validate / apply
            bad = {"streams": [{"id": "x", "input": {"type": "nope"},
                                "output": {"type": "drop"}}]}
            v = (await c.post("/api/v1/configuration/validate",
                              json=bad)).json()
            assert not v["valid"]

            good = {"streams": [{
                "id": "s2",
                "input": {"type": "generate", "batch_size": 1,
                          "interval": "50ms",
                          "fields": {"v": {"dtype": "float32"}}},
                "output": {"type": "drop"},
            }]}
            ap = (await c.post("/api/v1/configuration/apply",
                               json=good)).json()
            assert ap["applied"] and ap["version"] == 1
            streams = (await c.get("/api/v1/streams")).json()
            ids = {s["id"] for s in streams}
            assert ids == {"s2"}  # s1 removed, s2 added
            vs = (await c.get("/api/v1/configuration/versions")).json()
            assert len(vs) == 1

            # prometheus text
            m = (await c.get("/metrics")).text
            assert "arkflow_input_messages_total" in m
            assert 'stream="s2"' in m
        await eng.runtime.stop_all()

    run(main(), timeout=60)


def test_api_token_auth(run):
    async def main():
        eng = _engine()
        eng.config.server.token = "sekret"
        app = create_app(eng)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://node") as c:
            assert (await c.get("/api/v1/streams")).status_code == 401
            ok = await c.get("/api/v1/streams",
                             headers={"Authorization": "Bearer sekret"})
            assert ok.status_code == 200
            # health stays open
            assert (await c.get("/health")).status_code == 200

    run(main())


def test_cli_components_and_schema(capsys):
    from arkflow_amd.cli import main
    assert main(["components", "list"]) == 0
    out = capsys.readouterr().out
    assert "generate" in out and "sql" in out
    assert main(["components", "show", "processor", "sql"]) == 0
    assert main(["schema"]) == 0


def test_cli_validate(tmp_path, capsys):
    from arkflow_amd.cli import main
    p = tmp_path / "c.yaml"
    p.write_text("""
streams:
  - id: s1
    input: {type: generate, batch_size: 1, count: 1}
    output: {type: drop}
""")
    assert main(["--config", str(p), "--validate"]) == 0
    bad = tmp_path / "bad.yaml"
    bad.write_text("""
streams:
  - id: s1
    input: {type: nonexistent}
    output: {type: drop}
""")
    assert main(["--config", str(bad), "--validate"]) == 1


def test_console_served(run):
    async def main():
        eng = _engine()
        app = create_app(eng)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://node") as c:
            r = await c.get("/")
            assert r.status_code == 200
            assert "arkflow_amd console" in r.text

    run(main())


def test_config_versions_persist(tmp_path, run):
    async def main():
        cfg = EngineConfig.from_dict({
            "streams": [{"id": "s1",
                         "input": {"type": "generate", "batch_size": 1,
                                   "interval": "50ms",
                                   "fields": {"v": {"dtype": "float32"}}},
                         "output": {"type": "drop"}}],
            "server": {"config_store": str(tmp_path / "versions.json")},
        })
        eng = af.Engine(cfg)
        for sc in cfg.streams:
            eng.runtime.register(sc)
        r = await eng.control_plane.apply_configuration({
            "streams": [{"id": "s2",
                         "input": {"type": "generate", "batch_size": 1,
                                   "interval": "50ms",
                                   "fields": {"v": {"dtype": "float32"}}},
                         "output": {"type": "drop"}}]}, note="v1")
        assert r["applied"]
        await eng.runtime.stop_all()
        # a fresh engine sees the persisted version store
        eng2 = af.Engine(cfg)
        assert len(eng2.control_plane.versions.list()) == 1

    run(main(), timeout=30)


def test_config_diff(run):
    async def main():
        eng = _engine()
        d = eng.control_plane.diff_config({"streams": [
            {"id": "s1", "input": {"type": "generate", "batch_size": 9,
                                   "interval": "50ms",
                                   "fields": {"v": {"dtype": "float32"}}},
             "output": {"type": "drop"}},
            {"id": "brand_new", "input": {"type": "generate"},
             "output": {"type": "drop"}},
        ]})
        assert d["added"] == ["brand_new"]
        assert d["removed"] == []
        assert d["changed"] and d["changed"][0]["stream_id"] == "s1"
        assert "input" in d["changed"][0]["fields"]

    run(main())


def test_stream_metrics_and_version_detail(run):
    async def main():
        eng = _engine()
        for sc in eng.config.streams:
            eng.runtime.register(sc)
        app = create_app(eng)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://n") as c:
            r = await c.get("/api/v1/streams/s1/metrics")
            assert r.status_code == 200
            assert "input_messages" in r.json()
            assert (await c.get("/api/v1/streams/nope/metrics")
                    ).status_code == 404
            await eng.control_plane.apply_configuration(
                {"streams": [{"id": "s2",
                              "input": {"type": "generate", "batch_size": 1,
                                        "interval": "50ms",
                                        "fields": {"v": {"dtype": "float32"}}},
                              "output": {"type": "drop"}}]}, note="v1")
            r = await c.get("/api/v1/configuration/versions/1")
            assert r.status_code == 200
            assert (await c.get("/api/v1/configuration/versions/99")
                    ).status_code == 404
        await eng.runtime.stop_all()

    run(main(), timeout=30)


def test_console_pages_render(run):
    async def main():
        eng = _engine()
        app = create_app(eng)
        async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=app),
                base_url="http://n") as c:
            r = await c.get("/")
            assert r.status_code == 200
            assert "arkflow" in r.text.lower()
        from arkflow_amd.server.hub import Hub, create_hub_app
        happ = create_hub_app(Hub())
        async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=happ),
                base_url="http://h") as c:
            r = await c.get("/")
            assert r.status_code == 200 and "console" in r.text.lower()

    run(main())


def test_sse_last_event_id_resume(run):
    """SSE replays only events after the Last-Event-ID header (reference
    lib.rs:1113 resume semantics). Real socket: ASGITransport buffers
    streaming responses, so this boots uvicorn in-process."""
    import socket

    async def main():
        import uvicorn
        eng = _engine()
        for sc in eng.config.streams:
            eng.runtime.register(sc)
        for i in range(6):
            eng.runtime.events.push("s1", f"k{i}")
        evs = eng.runtime.events.list()
        cut = [e for e in evs if e.kind == "k2"][0].seq  # resume after k2
        with socket.socket() as s:
            s.bind(("127.0.0.1", 0))
            port = s.getsockname()[1]
        app = create_app(eng)
        server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=port, log_level="error"))
        stask = asyncio.ensure_future(server.serve())
        try:
            async with httpx.AsyncClient(
                    base_url=f"http://127.0.0.1:{port}") as c:
                for _ in range(100):
                    try:
                        if (await c.get("/health")).status_code == 200:
                            break
                    except httpx.TransportError:
                        await asyncio.sleep(0.1)
                got = []
                async with c.stream(
                        "GET", "/api/v1/events/stream",
                        headers={"Last-Event-ID": str(cut)}) as r:
                    async for line in r.aiter_lines():
                        if line.startswith("data: "):
                            got.append(json.loads(line[6:]))
                        if len(got) >= 3:
                            break
                kinds = [g["kind"] for g in got]
                assert kinds == ["k3", "k4", "k5"]
                assert all(g["seq"] > cut for g in got)
        finally:
            server.should_exit = True
            await asyncio.wait_for(stask, 10)
        await eng.runtime.stop_all()

    run(main(), timeout=60)


def test_console_spa_api_contract(run):
    """Every endpoint the console SPA calls must exist and answer — the
    build-free analog of the reference console's vitest suites."""
    async def main():
        eng = _engine()
        for sc in eng.config.streams:
            eng.runtime.register(sc)
        await eng.runtime.start("s1")
        eng.ready = True
        app = create_app(eng)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://n") as c:
            page = (await c.get("/")).text
            # the SPA covers the reference console's feature tabs
            for feature in ("overview", "runtime", "configuration",
                            "components", "events", "rollouts", "settings"):
                assert feature in page
            # endpoints the SPA fetches
            for path in ("/api/v1/system/status", "/api/v1/streams",
                         "/api/v1/streams/s1", "/api/v1/streams/s1/metrics",
                         "/api/v1/operations", "/api/v1/components",
                         "/api/v1/configuration",
                         "/api/v1/configuration/versions",
                         "/api/v1/events?limit=5"):
                r = await c.get(path)
                assert r.status_code == 200, path
            # lifecycle op buttons
            r = await c.post("/api/v1/streams/s1/restart")
            assert r.status_code == 200
            # config validate round-trip with the served config
            cfg = (await c.get("/api/v1/configuration")).json()
            r = await c.post("/api/v1/configuration/validate", json=cfg)
            assert r.status_code == 200
        await eng.runtime.stop_all()

    run(main(), timeout=60)


def test_console_hub_views(run):
    """Hub-side console views: /nodes, /intents, /rollouts + actions."""
    async def main():
        from arkflow_amd.server.hub import Hub, create_hub_app
        hub = Hub()
        await hub.store.upsert_node("n1", "tok", 5.0)
        app = create_hub_app(hub)
        transport = httpx.ASGITransport(app=app)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://h") as c:
            assert "rollouts" in (await c.get("/")).text
            assert (await c.get("/nodes")).status_code == 200
            assert (await c.get("/intents")).status_code == 200
            r = await c.post("/rollouts",
                             json={"config": {}, "nodes": ["n1"]})
            rid = r.json()["rollout_id"]
            assert (await c.get("/rollouts")).status_code == 200
            assert (await c.post(f"/rollouts/{rid}/pause")).status_code == 200
            assert (await c.post(f"/rollouts/{rid}/resume")).status_code \
                == 200

    run(main())
