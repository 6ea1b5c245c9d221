"""Tunable constants, every one env-overridable.

Capability parity with the reference's utils/constants.py:1-68 (names and
defaults match its published behavior: 20-tile send batches, 10 s heartbeat
interval / 60 s timeout, 50 MB payload cap, orchestration concurrencies).
Env prefix is DISTGPU_ here; the reference's COMFYUI_* names are accepted
as fallbacks so drop-in deployments keep working.
"""

from __future__ import annotations

import os


def _env(name: str, default, cast=None):
    cast = cast or type(default)
    for key in (f"DISTGPU_{name}", f"COMFYUI_{name}"):
        raw = os.environ.get(key)
        if raw is not None:
            try:
                return cast(raw)
            except (TypeError, ValueError):
                pass
    return default


# --- transport / batching -------------------------------------------------
#: Tiles accumulated on a worker before a flush back to the master
#: (reference: MAX_BATCH=20, utils/constants.py:43).
MAX_BATCH: int = _env("MAX_BATCH", 20)

#: Hard cap on a single result payload, bytes (reference: 50 MB,
#: upscale/job_store.py:12).
MAX_PAYLOAD_SIZE: int = _env("MAX_PAYLOAD_SIZE", 50 * 1024 * 1024)

#: Headroom subtracted from MAX_PAYLOAD_SIZE when chunking tile batches.
PAYLOAD_HEADROOM: int = _env("PAYLOAD_HEADROOM", 1024 * 1024)

#: Cap on a decoded audio payload, bytes (reference: 256 MB,
#: utils/audio_payload.py:11-13).
MAX_AUDIO_PAYLOAD_BYTES: int = _env("MAX_AUDIO_PAYLOAD_BYTES", 256 * 1024 * 1024)

# --- fault tolerance ------------------------------------------------------
#: Seconds between worker heartbeats (reference: utils/constants.py:46).
HEARTBEAT_INTERVAL: float = _env("HEARTBEAT_INTERVAL", 10.0)

#: Seconds of heartbeat silence before a worker is suspected dead
#: (reference: utils/constants.py:47).
HEARTBEAT_TIMEOUT: float = _env("HEARTBEAT_TIMEOUT", 60.0)

#: Seconds job_complete waits for the collector queue to exist
#: (reference: 10 s grace loop, api/job_routes.py:314-333).
JOB_INIT_GRACE_PERIOD: float = _env("JOB_INIT_GRACE_PERIOD", 10.0)

#: Collector wait slice; interrupt/timeout checks run between slices
#: (reference: ~0.5 s, nodes/collector.py:332).
COLLECTOR_SLICE_TIMEOUT: float = _env("COLLECTOR_SLICE_TIMEOUT", 0.5)

#: Retry budget for worker->master sends (reference: worker_comms.py:88-104).
SEND_RETRY_ATTEMPTS: int = _env("SEND_RETRY_ATTEMPTS", 5)

#: Retry budget / total cap for work-item requests
#: (reference: worker_comms.py:124-169).
REQUEST_RETRY_ATTEMPTS: int = _env("REQUEST_RETRY_ATTEMPTS", 10)
REQUEST_RETRY_TOTAL_SECONDS: float = _env("REQUEST_RETRY_TOTAL_SECONDS", 30.0)

#: Seconds a pull-queue pop waits before reporting empty
#: (reference: 0.1 s, api/usdu_routes.py:193-212).
QUEUE_POP_WAIT: float = _env("QUEUE_POP_WAIT", 0.1)

#: Job-ready poll attempts x interval on workers
#: (reference: <=20 x 1 s, upscale/modes/static.py:33-47).
JOB_READY_POLL_ATTEMPTS: int = _env("JOB_READY_POLL_ATTEMPTS", 20)
JOB_READY_POLL_INTERVAL: float = _env("JOB_READY_POLL_INTERVAL", 1.0)

# --- orchestration concurrency -------------------------------------------
#: Concurrent worker probes (reference: settings.worker_probe_concurrency).
WORKER_PROBE_CONCURRENCY: int = _env("ORCHESTRATION_PROBE_CONCURRENCY", 8)
#: Concurrent per-worker prompt preparation.
WORKER_PREP_CONCURRENCY: int = _env("ORCHESTRATION_PREP_CONCURRENCY", 4)
#: Concurrent media-sync uploads.
MEDIA_SYNC_CONCURRENCY: int = _env("MEDIA_SYNC_CONCURRENCY", 4)
MEDIA_SYNC_TIMEOUT_SECONDS: float = _env("MEDIA_SYNC_TIMEOUT_SECONDS", 120.0)

# --- intra-node runtime ---------------------------------------------------
#: Default rendezvous address for single-node torch.distributed jobs.
DEFAULT_MASTER_ADDR: str = _env("MASTER_ADDR_DEFAULT", "127.0.0.1")

#: Seconds an actor may go without completing a tile before the scheduler
#: probes its stream health (intra-node analog of the HTTP heartbeat).
ACTOR_HEARTBEAT_TIMEOUT: float = _env("ACTOR_HEARTBEAT_TIMEOUT", 60.0)
