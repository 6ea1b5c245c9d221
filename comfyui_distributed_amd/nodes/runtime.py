"""Node runtime context (dependency injection for node bodies).

Node classes need server facilities: the distributed job state, HTTP access
to the master / workers, interrupt checks. The reference reaches into the
PromptServer singleton for these (nodes/collector.py imports
server.PromptServer.instance); here they are an explicit, swappable
context — tests install a fake, the real server installs itself.
"""

from __future__ import annotations

import threading

from ..server.job_state import JobState


class NodeRuntime:
    def __init__(self, job_state: JobState | None = None):
        self.job_state = job_state or JobState()
        self._interrupted = False

    # -- HTTP (overridden by tests with fakes) ------------------------------

    async def post_json(self, url: str, payload: dict, timeout: float = 60.0) -> dict:
        import aiohttp

        async with aiohttp.ClientSession() as session:
            async with session.post(url, json=payload,
                                    timeout=aiohttp.ClientTimeout(total=timeout)) as r:
                r.raise_for_status()
                try:
                    return await r.json()
                except Exception:  # noqa: BLE001
                    return {}

    async def probe_worker(self, worker_id: str) -> dict | None:
        """GET <worker>/prompt -> queue info dict, or None when offline.
        The real server resolves worker_id -> URL via config; the default
        runtime knows no workers."""
        return None

    # -- interrupts ----------------------------------------------------------

    def interrupt(self):
        self._interrupted = True

    def clear_interrupt(self):
        self._interrupted = False

    def throw_if_interrupted(self):
        if self._interrupted:
            raise InterruptedError("processing interrupted by user")


_runtime_lock = threading.Lock()
_runtime: NodeRuntime | None = None


def get_runtime() -> NodeRuntime:
    global _runtime
    with _runtime_lock:
        if _runtime is None:
            _runtime = NodeRuntime()
        return _runtime


def set_runtime(rt: NodeRuntime | None) -> None:
    global _runtime
    with _runtime_lock:
        _runtime = rt
