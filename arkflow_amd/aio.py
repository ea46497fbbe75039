"""Small asyncio utilities.

`event_wait` exists because CPython 3.10's `asyncio.wait_for` converts an
external cancellation that lands on the same tick as its timeout into
TimeoutError. Loops that catch TimeoutError around
`wait_for(event.wait(), t)` therefore swallow cancellation and keep running
(this produced an intermittent engine-shutdown stall; NOTES.md #12).
`asyncio.wait`'s timeout never masks cancellation.
"""
from __future__ import annotations

import asyncio


async def event_wait(ev: asyncio.Event, timeout: float) -> bool:
    """Wait for `ev` up to `timeout` seconds; True if it was set.
    Cancellation always propagates (unlike wait_for on py3.10)."""
    if ev.is_set():
        return True
    waiter = asyncio.ensure_future(ev.wait())
    try:
        await asyncio.wait({waiter}, timeout=timeout)
        return waiter.done() and not waiter.cancelled()
    finally:
        waiter.cancel()


async def queue_get(q: "asyncio.Queue", timeout: float):
    """Get from `q` with a timeout; returns (True, item) or (False, None).
    Cancellation always propagates (same rationale as event_wait)."""
    getter = asyncio.ensure_future(q.get())
    try:
        await asyncio.wait({getter}, timeout=timeout)
        if getter.done() and not getter.cancelled():
            return True, getter.result()
        return False, None
    finally:
        getter.cancel()
