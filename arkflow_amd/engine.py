"""Engine: registers streams with the RuntimeManager and runs them to
completion or cancellation (reference crates/arkflow-core/src/engine/mod.rs)."""
from __future__ import annotations

import asyncio
import logging
import signal
from typing import Optional

from .config import EngineConfig
from .runtime import RuntimeManager

log = logging.getLogger("arkflow_amd.engine")


class Engine:
    def __init__(self, config: EngineConfig):
        # Join the torchrun process group FIRST (idempotent, env-gated):
        # maps LOCAL_RANK→cuda:N as the default device before any stream is
        # built, so `torchrun --nproc-per-node N python -m arkflow_amd
        # --config …` runs one engine shard per GPU with RCCL repartition
        # live (ranks used to start independent and non-sharding)
        from .parallel import dist as afdist
        afdist.init_from_env()
        self.rank = afdist.rank()
        self.world_size = afdist.world_size()
        if self.rank != 0:
            log.info("rank %d/%d: HTTP/metrics served by rank 0 only",
                     self.rank, self.world_size)
        self.config = config
        self.runtime = RuntimeManager()
        self.ready = False
        self.running = False
        from .control_plane import ControlPlane
        self.control_plane = ControlPlane(
            self, version_store_path=config.server.config_store)

    async def run_with_cancellation(self, cancel: Optional[asyncio.Event] = None,
                                    install_signal_handlers: bool = False) -> None:
        """engine/mod.rs:45-89: register all → start_all → await
        SIGINT/SIGTERM/cancel or natural EOF → stop_all → wait_all."""
        cancel = cancel or asyncio.Event()
        for sc in self.config.streams:
            self.runtime.register(sc)
        await self.runtime.start_all()
        self.ready = True
        self.running = True

        if install_signal_handlers:
            loop = asyncio.get_running_loop()
            for sig in (signal.SIGINT, signal.SIGTERM):
                try:
                    loop.add_signal_handler(sig, cancel.set)
                except NotImplementedError:
                    pass

        cancel_task = asyncio.ensure_future(cancel.wait())
        streams_task = asyncio.ensure_future(self.runtime.wait_all())
        done, _ = await asyncio.wait(
            {cancel_task, streams_task}, return_when=asyncio.FIRST_COMPLETED
        )
        if streams_task in done:
            cancel_task.cancel()
        else:
            streams_task.cancel()
            try:
                await streams_task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
        await self.runtime.stop_all()
        await self.runtime.wait_all()
        self.running = False

    async def run(self) -> None:
        await self.run_with_cancellation(install_signal_handlers=True)
