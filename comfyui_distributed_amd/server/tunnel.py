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


#: GitHub latest-release asset per platform (reference binary.py:13-46)
_CF_ASSETS = {
    ("linux", "x86_64"): "cloudflared-linux-amd64",
    ("linux", "aarch64"): "cloudflared-linux-arm64",
    ("darwin", "x86_64"): "cloudflared-darwin-amd64.tgz",
    ("darwin", "arm64"): "cloudflared-darwin-amd64.tgz",
    ("windows", "amd64"): "cloudflared-windows-amd64.exe",
}
_CF_RELEASE_URL = ("https://github.com/cloudflare/cloudflared/releases/"
                   "latest/download/{asset}")


def _download_dir() -> str:
    return os.path.join(os.path.dirname(__file__), "..", "..", ".cloudflared")


def download_cloudflared(fetch=None, dest_dir: str | None = None,
                         timeout: float = 60.0) -> str | None:
    """Fetch the platform's cloudflared binary from the GitHub latest
    release (reference utils/cloudflare/binary.py:47-83). ``fetch(url,
    timeout) -> bytes`` is injectable for tests; the default uses urllib
    and simply returns None when the host has no egress (this build
    environment does not — the tunnel then requires CLOUDFLARED_PATH)."""
    import platform

    key = (platform.system().lower(), platform.machine().lower())
    asset = _CF_ASSETS.get(key)
    if asset is None or asset.endswith((".tgz", ".exe")):
        return None  # only the plain linux binaries are auto-installed
    url = _CF_RELEASE_URL.format(asset=asset)
    if fetch is None:
        def fetch(u, t):
            import urllib.request

            with urllib.request.urlopen(u, timeout=t) as resp:
                return resp.read()
    try:
        data = fetch(url, timeout)
    except Exception as exc:  # no egress / fetch failure -> graceful None
        log(f"cloudflared download failed ({exc!r}); set CLOUDFLARED_PATH")
        return None
    if not data:
        return None
    dest_dir = dest_dir or _download_dir()
    os.makedirs(dest_dir, exist_ok=True)
    dest = os.path.join(dest_dir, "cloudflared")
    tmp = dest + ".part"
    with open(tmp, "wb") as fh:
        fh.write(data)
    os.chmod(tmp, 0o755)
    os.replace(tmp, dest)
    log(f"cloudflared downloaded to {dest}")
    return dest


def find_cloudflared(allow_download: bool = False, fetch=None) -> str | None:
    """Resolution order (reference binary.py:13-83): CLOUDFLARED_PATH ->
    PATH -> previously-downloaded copy -> (optional) fresh download."""
    override = os.environ.get("CLOUDFLARED_PATH")
    if override and os.path.isfile(override):
        return override
    on_path = shutil.which("cloudflared")
    if on_path:
        return on_path
    cached = os.path.join(_download_dir(), "cloudflared")
    if os.path.isfile(cached) and os.access(cached, os.X_OK):
        return cached
    if allow_download:
        return download_cloudflared(fetch=fetch)
    return None


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
            binary = find_cloudflared(allow_download=True)
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
