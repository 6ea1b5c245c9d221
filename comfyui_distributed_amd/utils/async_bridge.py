"""Sync-node <-> server-event-loop bridge.

Reference counterpart: utils/async_helpers.py:13-54
(``run_async_in_server_loop``): node bodies execute on a worker thread while
HTTP endpoints live on the asyncio loop; this schedules a coroutine on the
loop and blocks the calling thread until it finishes.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import threading


class ServerLoop:
    """A singleton-ish holder for the server's running event loop."""

    _loop: asyncio.AbstractEventLoop | None = None
    _thread: threading.Thread | None = None
    _lock = threading.Lock()

    @classmethod
    def set(cls, loop: asyncio.AbstractEventLoop) -> None:
        cls._loop = loop

    @classmethod
    def get(cls) -> asyncio.AbstractEventLoop:
        with cls._lock:
            if cls._loop is None or cls._loop.is_closed():
                # standalone mode (tests, worker CLI): own a background loop
                loop = asyncio.new_event_loop()
                t = threading.Thread(target=loop.run_forever, daemon=True,
                                     name="distgpu-loop")
                t.start()
                cls._loop = loop
                cls._thread = t
            return cls._loop


def run_async_in_server_loop(coro, timeout: float | None = None):
    """Run ``coro`` on the server loop from a sync thread; block for the
    result. Raises the coroutine's exception in the caller."""
    loop = ServerLoop.get()
    try:
        running = asyncio.get_running_loop()
    except RuntimeError:
        running = None
    if running is loop:
        raise RuntimeError("run_async_in_server_loop called from the loop thread")
    fut = asyncio.run_coroutine_threadsafe(coro, loop)
    try:
        return fut.result(timeout=timeout)
    except concurrent.futures.TimeoutError:
        fut.cancel()
        raise TimeoutError(f"async bridge timed out after {timeout}s")
