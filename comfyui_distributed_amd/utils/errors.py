"""Typed error hierarchy (parity: reference utils/exceptions.py:4-42)."""

from __future__ import annotations


class DistributedError(Exception):
    """Base error for the distributed runtime."""


class WorkerError(DistributedError):
    """A worker failed or returned an invalid result."""


class WorkerTimeoutError(WorkerError):
    """A worker stopped heartbeating and failed its liveness probe."""


class WorkerNotAvailableError(WorkerError):
    """No enabled worker could be reached."""


class JobQueueError(DistributedError):
    """Job queue state is missing or inconsistent."""


class TileCollectionError(DistributedError):
    """Tile results could not be collected/combined."""


class ProcessError(DistributedError):
    """Worker process lifecycle failure."""


class TunnelError(DistributedError):
    """Tunnel lifecycle failure."""


class KernelUnavailableError(DistributedError):
    """A HIP kernel extension is required on GPU but was not built/loaded.

    Raised loudly instead of silently falling back to eager PyTorch when
    running on an MI355X device (the CPU fallback exists only for CPU-side
    tests).
    """


class PromptValidationError(DistributedError):
    """A workflow graph failed validation; carries per-node errors."""

    def __init__(self, message: str, node_errors: dict | None = None):
        super().__init__(message)
        self.node_errors = node_errors or {}
