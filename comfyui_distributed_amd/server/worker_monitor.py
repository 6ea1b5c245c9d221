"""Standalone worker watchdog.

Reference counterpart: workers/worker_monitor.py:41-132 — spawns the worker
process, polls the master PID every 2 s, kills the worker when the master
dies, and forwards SIGTERM/SIGINT.

Usage: python -m comfyui_distributed_amd.server.worker_monitor \
           --master-pid P -- <worker command...>
"""

from __future__ import annotations

import argparse
import os
import signal
import subprocess
import sys
import time

POLL_INTERVAL = 2.0


def pid_alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except ProcessLookupError:
        return False
    except PermissionError:
        return True


def monitor_and_run(master_pid: int, cmd: list[str]) -> int:
    proc = subprocess.Popen(cmd)

    def forward(signum, _frame):
        try:
            proc.send_signal(signum)
        except ProcessLookupError:
            pass

    signal.signal(signal.SIGTERM, forward)
    signal.signal(signal.SIGINT, forward)

    while True:
        rc = proc.poll()
        if rc is not None:
            return rc
        if not pid_alive(master_pid):
            sys.stderr.write(
                f"[worker-monitor] master pid {master_pid} gone — killing worker\n"
            )
            proc.terminate()
            try:
                proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                proc.kill()
            return 1
        time.sleep(POLL_INTERVAL)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--master-pid", type=int, required=True)
    ap.add_argument("cmd", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        ap.error("no worker command given")
    sys.exit(monitor_and_run(args.master_pid, cmd))


if __name__ == "__main__":
    main()
