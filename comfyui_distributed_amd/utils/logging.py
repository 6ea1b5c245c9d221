"""Prefix logging + execution-scoped tracing.

Parity: reference utils/logging.py:15-43 ([Distributed] prefix, config-gated
debug with a TTL cache) and utils/trace_logger.py:4-13 ([exec:<id>] prefix).
"""

from __future__ import annotations

import os
import sys
import time

PREFIX = "[Distributed]"

#: in-memory ring buffer served by /distributed/local_log (reference keeps
#: an app.logger buffer, worker_routes.py:348-390)
from collections import deque as _deque

LOG_BUFFER: "_deque[str]" = _deque(maxlen=500)

_debug_cache = {"value": None, "ts": 0.0}
_DEBUG_TTL = 5.0


def is_debug_enabled() -> bool:
    """Config/env-gated debug flag with a 5 s TTL cache (reference parity)."""
    now = time.monotonic()
    if _debug_cache["value"] is None or now - _debug_cache["ts"] > _DEBUG_TTL:
        enabled = os.environ.get("DISTGPU_DEBUG", "") not in ("", "0", "false")
        if not enabled:
            try:
                from .config import load_config

                enabled = bool(load_config().get("settings", {}).get("debug", False))
            except Exception:
                enabled = False
        _debug_cache["value"] = enabled
        _debug_cache["ts"] = now
    return _debug_cache["value"]


def log(*args) -> None:
    line = " ".join(str(a) for a in args)
    LOG_BUFFER.append(f"{PREFIX} {line}")
    print(PREFIX, line, file=sys.stderr, flush=True)


def debug_log(*args) -> None:
    if is_debug_enabled():
        log(*args)


def trace_info(trace_id: str, *args) -> None:
    log(f"[exec:{trace_id}]", *args)


def trace_debug(trace_id: str, *args) -> None:
    debug_log(f"[exec:{trace_id}]", *args)
