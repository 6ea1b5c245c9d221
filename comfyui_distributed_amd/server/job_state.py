"""Server-side distributed job state.

Reference counterpart: the two stores on the PromptServer singleton —
collector jobs ``distributed_pending_jobs: dict[job_id, asyncio.Queue]``
(api/queue_orchestration.py:42-52) and tile jobs
``distributed_pending_tile_jobs: dict[job_id, TileJobState|ImageJobState]``
(upscale/job_store.py:15-31). Here both live on an explicit ``JobState``
object owned by the server app (no singletons), same lock discipline.
"""

from __future__ import annotations

import asyncio
import time
from dataclasses import dataclass, field

from ..utils import constants


@dataclass
class TileJobState:
    """Static-mode tile job (reference upscale/job_models.py:6-30)."""

    job_id: str
    mode: str = "static"
    batched_static: bool = True
    total_tasks: int = 0
    pending_tasks: asyncio.Queue = field(default_factory=asyncio.Queue)
    results: asyncio.Queue = field(default_factory=asyncio.Queue)
    completed_tasks: dict = field(default_factory=dict)  # (tile, batch) -> payload
    worker_status: dict = field(default_factory=dict)  # worker_id -> last hb ts
    assigned_to_workers: dict = field(default_factory=dict)  # task -> worker_id
    finished_workers: set = field(default_factory=set)
    batch_size: int = 1


@dataclass
class ImageJobState:
    """Dynamic-mode whole-image job (reference upscale/job_models.py:32-49)."""

    job_id: str
    mode: str = "dynamic"
    total_images: int = 0
    pending_images: asyncio.Queue = field(default_factory=asyncio.Queue)
    results: asyncio.Queue = field(default_factory=asyncio.Queue)
    completed_images: dict = field(default_factory=dict)
    worker_status: dict = field(default_factory=dict)
    assigned_to_workers: dict = field(default_factory=dict)
    finished_workers: set = field(default_factory=set)


class JobState:
    #: abandoned jobs (a prompt that died before its collector/USDU node
    #: ran) are swept after this many seconds so queues can't leak forever
    STALE_JOB_TTL = 3600.0

    def __init__(self):
        # collector jobs
        self.pending_jobs: dict[str, asyncio.Queue] = {}
        self.jobs_lock = asyncio.Lock()
        # tile/image jobs
        self.tile_jobs: dict[str, TileJobState | ImageJobState] = {}
        self.tile_jobs_lock = asyncio.Lock()
        self._created: dict[str, float] = {}

    def _sweep_locked(self, store: dict) -> None:
        cutoff = time.monotonic() - self.STALE_JOB_TTL
        for jid in [j for j in store if self._created.get(j, cutoff) < cutoff]:
            store.pop(jid, None)
            self._created.pop(jid, None)

    # ---- collector jobs ---------------------------------------------------

    async def ensure_queue(self, job_id: str) -> asyncio.Queue:
        async with self.jobs_lock:
            self._sweep_locked(self.pending_jobs)
            q = self.pending_jobs.get(job_id)
            if q is None:
                q = asyncio.Queue()
                self.pending_jobs[job_id] = q
                self._created[job_id] = time.monotonic()
            return q

    async def get_queue_waiting(
        self, job_id: str, grace: float | None = None
    ) -> asyncio.Queue | None:
        """Queue for a job, waiting up to the init grace period for the
        master to create it (reference job_routes.py:314-333)."""
        grace = grace if grace is not None else constants.JOB_INIT_GRACE_PERIOD
        deadline = time.monotonic() + grace
        while True:
            async with self.jobs_lock:
                q = self.pending_jobs.get(job_id)
            if q is not None:
                return q
            if time.monotonic() >= deadline:
                return None
            await asyncio.sleep(0.2)

    async def drop_queue(self, job_id: str) -> None:
        async with self.jobs_lock:
            self.pending_jobs.pop(job_id, None)
            self._created.pop(job_id, None)

    # ---- tile jobs --------------------------------------------------------

    async def init_static_job(self, job_id: str, n_tiles: int,
                              batch_size: int = 1) -> TileJobState:
        job = TileJobState(job_id=job_id, total_tasks=n_tiles,
                           batch_size=batch_size)
        for t in range(n_tiles):
            job.pending_tasks.put_nowait(t)
        async with self.tile_jobs_lock:
            self._sweep_locked(self.tile_jobs)
            self.tile_jobs[job_id] = job
            self._created[job_id] = time.monotonic()
        return job

    async def init_dynamic_job(self, job_id: str, n_images: int) -> ImageJobState:
        job = ImageJobState(job_id=job_id, total_images=n_images)
        for i in range(n_images):
            job.pending_images.put_nowait(i)
        async with self.tile_jobs_lock:
            self._sweep_locked(self.tile_jobs)
            self.tile_jobs[job_id] = job
            self._created[job_id] = time.monotonic()
        return job

    async def get_tile_job(self, job_id: str):
        async with self.tile_jobs_lock:
            return self.tile_jobs.get(job_id)

    async def cleanup_job(self, job_id: str) -> None:
        async with self.tile_jobs_lock:
            self.tile_jobs.pop(job_id, None)
            self._created.pop(job_id, None)
