"""Agent: node-side loop speaking the Hub protocol.

Mirrors reference crates/arkflow-server/src/agent.rs (:57-145): register with
backoff → heartbeat / report / poll-commands ticks → execute via the local
ControlPlane → post results, with an idempotent command replay cache.
`transport` injection lets tests run hub+agent in one process over ASGI
(no network), the way the reference uses mock HTTP routers.
"""
from __future__ import annotations

import asyncio
import logging
import socket
from typing import Optional, Set

import httpx

log = logging.getLogger("arkflow_amd.agent")


class Agent:
    def __init__(self, engine, hub_url: str, node_id: Optional[str] = None,
                 heartbeat_interval: float = 5.0,
                 report_interval: float = 5.0,
                 poll_interval: float = 1.0,
                 registration_token: Optional[str] = None,
                 labels: Optional[dict] = None,
                 transport: Optional[httpx.AsyncBaseTransport] = None):
        self.engine = engine
        self.hub_url = hub_url.rstrip("/")
        self.node_id = node_id or socket.gethostname()
        self.registration_token = registration_token
        self.labels = labels or {}
        self.heartbeat_interval = heartbeat_interval
        self.report_interval = report_interval
        self.poll_interval = poll_interval
        self.token: Optional[str] = None
        self._transport = transport
        self._executed: Set[str] = set()  # idempotent replay cache

    def _client(self) -> httpx.AsyncClient:
        headers = {}
        if self.token:
            headers["x-node-token"] = self.token
        return httpx.AsyncClient(transport=self._transport,
                                 base_url=self.hub_url, headers=headers,
                                 timeout=10.0)

    async def register(self) -> None:
        backoff = 0.2
        while True:
            try:
                async with self._client() as c:
                    headers = {}
                    if self.registration_token:
                        headers["x-registration-token"] = \
                            self.registration_token
                    r = await c.post("/agent/register",
                                     json={"node_id": self.node_id,
                                           "labels": self.labels},
                                     headers=headers)
                    r.raise_for_status()
                    self.token = r.json()["node_token"]
                    return
            except Exception:  # noqa: BLE001
                log.warning("register failed; retrying in %.1fs", backoff)
                await asyncio.sleep(backoff)
                backoff = min(backoff * 2, 10.0)

    async def _heartbeat(self) -> None:
        async with self._client() as c:
            await c.post(f"/agent/{self.node_id}/heartbeat")

    async def _report(self) -> None:
        snapshot = {
            "status": self.engine.control_plane.system_status(),
            "streams": self.engine.control_plane.stream_snapshots(),
        }
        async with self._client() as c:
            await c.post(f"/agent/{self.node_id}/report", json=snapshot)

    async def _poll_and_execute(self) -> int:
        async with self._client() as c:
            r = await c.get(f"/agent/{self.node_id}/commands")
            r.raise_for_status()
            commands = r.json()
            executed = 0
            for cmd in commands:
                aid = cmd["attempt_id"]
                if aid in self._executed:
                    continue  # replay: already executed, result may have raced
                self._executed.add(aid)
                ok, detail = await self._execute(cmd)
                await c.post(
                    f"/agent/{self.node_id}/commands/{aid}/result",
                    json={"ok": ok, "detail": detail})
                executed += 1
            return executed

    async def _execute(self, cmd: dict) -> tuple:
        try:
            if cmd.get("kind") == "lifecycle":
                result = await self.engine.control_plane.lifecycle(
                    cmd["stream_id"], cmd["op"])
                return result["state"] == "succeeded", result.get("error") or ""
            if cmd.get("kind") == "apply_config":
                result = await self.engine.control_plane.apply_configuration(
                    cmd["config"])
                return bool(result.get("applied")), "; ".join(
                    result.get("errors") or [])
            return False, f"unknown command kind {cmd.get('kind')!r}"
        except Exception as e:  # noqa: BLE001
            return False, str(e)

    async def run(self, cancel: asyncio.Event) -> None:
        await self.register()
        last_hb = last_report = 0.0
        loop = asyncio.get_running_loop()
        while not cancel.is_set():
            now = loop.time()
            try:
                if now - last_hb >= self.heartbeat_interval:
                    await self._heartbeat()
                    last_hb = now
                if now - last_report >= self.report_interval:
                    await self._report()
                    last_report = now
                await self._poll_and_execute()
            except Exception:  # noqa: BLE001
                log.exception("agent tick failed")
            from ..aio import event_wait
            await event_wait(cancel, self.poll_interval)


async def agent_run(engine, cancel: asyncio.Event) -> None:
    cfg = engine.config.server
    agent = Agent(engine, cfg.hub_url, cfg.node_id,
                  registration_token=cfg.node_token,
                  labels=getattr(cfg, "node_labels", None))
    await agent.run(cancel)
