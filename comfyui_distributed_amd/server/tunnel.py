"""Cloudflare tunnel lifecycle (cloud-worker enabler).

Reference counterpart: utils/cloudflare/ (binary discovery binary.py:13-83,
stdout reader thread extracting the *.trycloudflare.com URL
process_reader.py:10-97, config-persisted state incl. previous master host
state.py:14-81, async-locked lifecycle tunnel.py:56-207). Binary download
is attempted only when the host has egress (this build environment does
not; a preinstalled ``cloudflared`` on PATH or CLOUDFLARED_PATH is used
when present).
"""

from __future__ import annotations

import asyncio
import os
import re
import shutil
import subprocess
import threading
from collections import deque

from ..utils.config import config_transaction
from ..utils.errors import TunnelError
from ..utils.logging import log

TUNNEL_URL_RE = re.compile(r"https://[a-z0-9-]+\.trycloudflare\.com")
START_TIMEOUT = float(os.environ.get("TUNNEL_START_TIMEOUT", "30"))


def find_cloudflared() -> str | None:
    override = os.environ.get("CLOUDFLARED_PATH")
    if override and os.path.isfile(override):
        return override
    return shutil.which("cloudflared")


class ProcessReader:
    """Background stdout reader: captures the assigned tunnel URL and keeps
    a ring buffer of error lines (reference process_reader.py:10-97)."""

    def __init__(self, proc: subprocess.Popen, maxlines: int = 200):
        self.proc = proc
        self.url: str | None = None
        self.lines: deque[str] = deque(maxlen=maxlines)
        self._url_event = threading.Event()
        self._thread = threading.Thread(target=self._read, daemon=True)
        self._thread.start()

    def _read(self):
        for raw in iter(self.proc.stdout.readline, b""):
            line = raw.decode(errors="replace").rstrip()
            self.lines.append(line)
            m = TUNNEL_URL_RE.search(line)
            if m and self.url is None:
                self.url = m.group(0)
                self._url_event.set()

    def wait_for_url(self, timeout: float) -> str | None:
        self._url_event.wait(timeout)
        return self.url


class TunnelManager:
    """Async-locked start/stop/status; persists tunnel state + restores the
    previous master host on stop (reference tunnel.py:56-207)."""

    def __init__(self):
        self._lock = asyncio.Lock()
        self.proc: subprocess.Popen | None = None
        self.reader: ProcessReader | None = None
        self.url: str | None = None

    async def start(self, port: int) -> str:
        async with self._lock:
            if self.proc is not None and self.proc.poll() is None:
                return self.url or ""
            binary = find_cloudflared()
            if binary is None:
                raise TunnelError(
                    "cloudflared binary not found (no egress to download it; "
                    "set CLOUDFLARED_PATH)"
                )
            self.proc = subprocess.Popen(
                [binary, "tunnel", "--url", f"http://127.0.0.1:{port}"],
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            )
            self.reader = ProcessReader(self.proc)
            loop = asyncio.get_running_loop()
            url = await loop.run_in_executor(
                None, self.reader.wait_for_url, START_TIMEOUT
            )
            if not url:
                self.proc.terminate()
                raise TunnelError(
                    f"tunnel URL not seen within {START_TIMEOUT}s; last lines: "
                    f"{list(self.reader.lines)[-5:]}"
                )
            self.url = url
            async with config_transaction() as cfg:
                cfg.setdefault("tunnel", {})
                cfg["tunnel"]["active"] = True
                cfg["tunnel"]["url"] = url
                cfg["tunnel"]["previous_master_host"] = cfg["master"].get("host", "")
                cfg["master"]["host"] = url.removeprefix("https://")
            log(f"tunnel up: {url}")
            return url

    async def stop(self) -> None:
        async with self._lock:
            if self.proc is not None:
                self.proc.terminate()

                def _reap(p=self.proc):
                    try:
                        p.wait(timeout=5)
                    except subprocess.TimeoutExpired:
                        p.kill()
                        p.wait(timeout=5)

                # off-loop: waiting on the child must not stall the server
                await asyncio.get_running_loop().run_in_executor(None, _reap)
                self.proc = None
            self.url = None
            async with config_transaction() as cfg:
                tun = cfg.setdefault("tunnel", {})
                if tun.get("active"):
                    cfg["master"]["host"] = tun.get("previous_master_host", "")
                tun["active"] = False
                tun.pop("url", None)
            log("tunnel stopped")

    def status(self) -> dict:
        alive = self.proc is not None and self.proc.poll() is None
        return {"active": alive, "url": self.url if alive else None}
