"""Execution tracing: trace-id prefixed logs + rocTX profiler ranges.

Reference has only log-prefix tracing (utils/trace_logger.py:4-13); this
adds rocprof-visible ranges around the compute phases (torch's nvtx maps to
rocTX on ROCm, so ``rocprofv3 --marker-trace`` shows them).
"""

from __future__ import annotations

from contextlib import contextmanager

import torch

_HAVE_NVTX = hasattr(torch.cuda, "nvtx")


@contextmanager
def trace_range(name: str):
    """rocTX range when on GPU; no-op otherwise."""
    if _HAVE_NVTX and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
