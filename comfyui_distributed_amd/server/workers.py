"""Worker process lifecycle: launch with GPU pinning, watchdog, persistence.

Reference counterpart: workers/process/lifecycle.py (+ launch_builder,
persistence, worker_monitor). A worker here is another instance of this
framework's server (``python -m comfyui_distributed_amd.server.app``)
pinned to one GPU via HIP_VISIBLE_DEVICES (the ROCm equivalent of the
reference's CUDA_VISIBLE_DEVICES pinning, lifecycle.py:32-36).
"""

from __future__ import annotations

import os
import signal
import subprocess
import sys
import time
from pathlib import Path

from ..utils.config import config_transaction, load_config
from ..utils.errors import ProcessError
from ..utils.logging import log


def worker_log_path(worker_id: str) -> Path:
    logs = Path(os.environ.get("DISTGPU_LOG_DIR", "logs"))
    logs.mkdir(parents=True, exist_ok=True)
    stamp = time.strftime("%Y%m%d")
    return logs / f"worker_{worker_id}_{stamp}.log"


def build_launch_command(worker: dict) -> list[str]:
    """python -m ...server.app --port N [--listen ...] + shlex'd extra args
    (reference launch_builder.py:33-67, with its shell-metachar blacklist
    :133-143)."""
    cmd = [
        sys.executable, "-m", "comfyui_distributed_amd.server.app",
        "--port", str(worker.get("port", 8189)),
        "--worker",
    ]
    extra = worker.get("extra_args") or ""
    if extra:
        import shlex

        for token in shlex.split(extra):
            if any(ch in token for ch in ";&|`$<>\n"):
                raise ProcessError(f"refusing shell metacharacters in {token!r}")
            cmd.append(token)
    return cmd


def launch_worker(worker: dict, monitor: bool = True) -> subprocess.Popen:
    wid = str(worker.get("id"))
    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = str(worker.get("cuda_device", 0))
    env["CUDA_VISIBLE_DEVICES"] = str(worker.get("cuda_device", 0))
    env["DISTGPU_IS_WORKER"] = "1"
    env["DISTGPU_MASTER_PID"] = str(os.getpid())
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    cmd = build_launch_command(worker)
    if monitor:
        # wrap in the watchdog so orphaned workers die with the master
        # (reference lifecycle.py:67-76)
        cmd = [sys.executable, "-m",
               "comfyui_distributed_amd.server.worker_monitor",
               "--master-pid", str(os.getpid()), "--", *cmd]
    with open(worker_log_path(wid), "ab") as logf:
        proc = subprocess.Popen(
            cmd, env=env, stdout=logf, stderr=subprocess.STDOUT,
            start_new_session=True,
        )  # the child holds its own fd; the parent's copy closes here
    log(f"launched worker {wid} pid={proc.pid} on GPU "
        f"{worker.get('cuda_device', 0)}")
    persist_managed_pid(wid, proc.pid)
    return proc


def stop_worker(handle: subprocess.Popen | None, worker_id: str) -> None:
    """TERM then KILL the whole process group (reference
    lifecycle.py:210-293)."""
    pid = handle.pid if handle is not None else load_managed_pid(worker_id)
    if pid is None:
        return
    try:
        pgid = os.getpgid(pid)
        os.killpg(pgid, signal.SIGTERM)
        deadline = time.time() + 5.0
        while time.time() < deadline:
            if not is_process_alive(pid):
                break
            time.sleep(0.2)
        if is_process_alive(pid):
            os.killpg(pgid, signal.SIGKILL)
    except ProcessLookupError:
        pass
    finally:
        if handle is not None:
            try:
                handle.wait(timeout=2.0)  # reap so no zombie lingers
            except Exception:
                pass
        forget_managed_pid(worker_id)


def is_process_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False
    except PermissionError:
        return True


# ---- persistence (reference workers/process/persistence.py) ---------------


def persist_managed_pid(worker_id: str, pid: int) -> None:
    import asyncio

    async def txn():
        async with config_transaction() as cfg:
            cfg.setdefault("managed_processes", {})[str(worker_id)] = {
                "pid": pid, "started": time.time(), "launching": True,
            }

    _run(txn())


def forget_managed_pid(worker_id: str) -> None:
    import asyncio

    async def txn():
        async with config_transaction() as cfg:
            cfg.get("managed_processes", {}).pop(str(worker_id), None)

    _run(txn())


def load_managed_pid(worker_id: str) -> int | None:
    entry = load_config().get("managed_processes", {}).get(str(worker_id))
    return entry.get("pid") if entry else None


def adopt_or_cleanup_managed() -> dict[str, int]:
    """On master restart: re-adopt live managed workers, forget dead ones
    (reference persistence.py:11-29 + startup.py:27-32)."""
    alive = {}
    for wid, entry in list(load_config().get("managed_processes", {}).items()):
        pid = entry.get("pid")
        if pid and is_process_alive(pid):
            alive[wid] = pid
        else:
            forget_managed_pid(wid)
    return alive


def _run(coro):
    import asyncio

    try:
        loop = asyncio.get_running_loop()
    except RuntimeError:
        loop = None
    if loop is None:
        asyncio.run(coro)
    else:
        loop.create_task(coro)  # fire-and-forget from a route handler
