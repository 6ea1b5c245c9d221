"""JSON cluster config.

Parity with reference utils/config.py (schema from SURVEY §5.6):
``gpu_config.json`` with ``master{host,port,cuda_device,extra_args}``,
``workers[{id,name,host,port,cuda_device,enabled,extra_args,type}]``,
``settings{...}``, ``tunnel{...}``, ``managed_processes{...}``.
Load is mtime-cached with a recursive default merge that preserves unknown
keys (reference config.py:47-97); saves are atomic tmp+fsync+rename
(config.py:99-116); ``config_transaction`` is an async lock with
snapshot-diff save (config.py:119-129).
"""

from __future__ import annotations

import asyncio
import copy
import json
import os
import tempfile
from contextlib import asynccontextmanager
from pathlib import Path
from typing import Any

DEFAULT_CONFIG: dict[str, Any] = {
    "master": {
        "host": "",
        "port": 8188,
        "cuda_device": 0,
        "extra_args": "",
    },
    "workers": [],
    "settings": {
        "debug": False,
        "auto_launch_workers": False,
        "stop_workers_on_master_exit": True,
        "master_delegate_only": False,
        "websocket_orchestration": True,
        "worker_timeout_seconds": None,
        "worker_probe_concurrency": 8,
        "worker_prep_concurrency": 4,
        "media_sync_concurrency": 4,
        "media_sync_timeout_seconds": 120,
        "has_auto_populated_workers": False,
    },
    "tunnel": {},
    "managed_processes": {},
}

WORKER_TYPES = ("local", "remote", "cloud")

_cache: dict[str, Any] = {"path": None, "mtime": None, "data": None}
_config_lock = asyncio.Lock()


def config_path() -> Path:
    override = os.environ.get("DISTGPU_CONFIG")
    if override:
        return Path(override)
    return Path(__file__).resolve().parent.parent / "gpu_config.json"


def _merge_defaults(defaults: dict, data: dict) -> dict:
    """Recursive default merge that keeps unknown keys from ``data``."""
    out = copy.deepcopy(data)
    for key, dval in defaults.items():
        if key not in out:
            out[key] = copy.deepcopy(dval)
        elif isinstance(dval, dict) and isinstance(out[key], dict):
            out[key] = _merge_defaults(dval, out[key])
    return out


def load_config(path: Path | None = None) -> dict[str, Any]:
    path = path or config_path()
    try:
        mtime = path.stat().st_mtime_ns
    except OSError:
        mtime = None
    if (
        _cache["data"] is not None
        and _cache["path"] == str(path)
        and _cache["mtime"] == mtime
    ):
        return copy.deepcopy(_cache["data"])
    data: dict[str, Any] = {}
    if mtime is not None:
        try:
            with open(path, "r", encoding="utf-8") as fh:
                data = json.load(fh)
        except (OSError, json.JSONDecodeError):
            data = {}
    merged = _merge_defaults(DEFAULT_CONFIG, data)
    _cache.update(path=str(path), mtime=mtime, data=copy.deepcopy(merged))
    return merged


def save_config(cfg: dict[str, Any], path: Path | None = None) -> None:
    """Atomic save: tmp file in the same dir + fsync + rename."""
    path = path or config_path()
    path.parent.mkdir(parents=True, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=str(path.parent), suffix=".tmp")
    try:
        with os.fdopen(fd, "w", encoding="utf-8") as fh:
            json.dump(cfg, fh, indent=2)
            fh.flush()
            os.fsync(fh.fileno())
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)
    _cache.update(path=str(path), mtime=None, data=None)  # force reload


@asynccontextmanager
async def config_transaction(path: Path | None = None):
    """Async-locked read-modify-write; saves only when the dict changed."""
    async with _config_lock:
        cfg = load_config(path)
        snapshot = copy.deepcopy(cfg)
        yield cfg
        if cfg != snapshot:
            save_config(cfg, path)


def get_worker_by_id(cfg: dict, worker_id) -> dict | None:
    wid = str(worker_id)
    for worker in cfg.get("workers", []):
        if str(worker.get("id")) == wid:
            return worker
    return None


def enabled_workers(cfg: dict) -> list[dict]:
    return [w for w in cfg.get("workers", []) if w.get("enabled")]


def get_worker_timeout_seconds(cfg: dict | None = None) -> float:
    """Unified timeout: UI setting > env > constant (reference config.py:141-157)."""
    from . import constants

    cfg = cfg or load_config()
    val = cfg.get("settings", {}).get("worker_timeout_seconds")
    if val is not None:
        try:
            val = float(val)
            if val > 0:
                return val
        except (TypeError, ValueError):
            pass
    return constants.HEARTBEAT_TIMEOUT


def is_master_delegate_only(cfg: dict | None = None) -> bool:
    cfg = cfg or load_config()
    return bool(cfg.get("settings", {}).get("master_delegate_only", False))
