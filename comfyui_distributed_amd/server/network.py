"""Worker/master URL building + probing (parity: reference utils/network.py).

A pooled aiohttp session, host normalization, worker URL heuristics
(cloud/https), ``probe_worker`` = GET /prompt, master callback URLs (local
workers always call back on 127.0.0.1).
"""

from __future__ import annotations

import asyncio

from ..utils.logging import debug_log

_session = None
_session_lock = asyncio.Lock()


async def get_client_session():
    """Shared pooled session (reference network.py:14-26: 100 conns,
    30/host)."""
    global _session
    import aiohttp

    async with _session_lock:
        if _session is None or _session.closed:
            _session = aiohttp.ClientSession(
                connector=aiohttp.TCPConnector(limit=100, limit_per_host=30)
            )
        return _session


async def close_client_session():
    """Close the shared session (server shutdown / test teardown)."""
    global _session
    async with _session_lock:
        if _session is not None and not _session.closed:
            await _session.close()
        _session = None


def normalize_host(host: str) -> str:
    host = (host or "").strip()
    for prefix in ("http://", "https://"):
        if host.startswith(prefix):
            host = host[len(prefix):]
    return host.rstrip("/")


def split_host_port(hostport: str) -> tuple[str, int | None]:
    """IPv6-aware host:port split (reference network.py:57-85)."""
    h = normalize_host(hostport)
    if h.startswith("["):  # [v6]:port
        idx = h.find("]")
        host = h[1:idx]
        rest = h[idx + 1:]
        port = int(rest[1:]) if rest.startswith(":") else None
        return host, port
    if h.count(":") == 1:
        host, port_s = h.split(":")
        try:
            return host, int(port_s)
        except ValueError:
            return h, None
    return h, None


def is_cloud_host(host: str) -> bool:
    h = normalize_host(host)
    return any(
        token in h
        for token in ("trycloudflare.com", "proxy.runpod.net", "ngrok")
    )


def build_worker_url(worker: dict) -> str:
    """URL heuristics (reference network.py:88-105): cloud hosts use https
    without an explicit port; everything else http://host:port."""
    host = normalize_host(worker.get("host") or "localhost")
    port = worker.get("port")
    # a host entered as "1.2.3.4:9000" carries its own port
    bare, embedded = split_host_port(host)
    if embedded is not None:
        host, port = bare, embedded
    if is_cloud_host(host):
        return f"https://{host}"
    if not host:
        host = "localhost"
    return f"http://{host}:{port}"


def build_master_url(master: dict, default_port: int = 8188) -> str:
    host = normalize_host(master.get("host") or "")
    port = master.get("port") or default_port
    if host and is_cloud_host(host):
        return f"https://{host}"
    return f"http://{host or '127.0.0.1'}:{port}"


def build_master_callback_url(master: dict, worker: dict,
                              default_port: int = 8188) -> str:
    """Local workers always call back on loopback (reference
    network.py:139-201)."""
    if worker.get("type", "local") == "local" or not worker.get("host"):
        return f"http://127.0.0.1:{master.get('port') or default_port}"
    return build_master_url(master, default_port)


async def probe_worker(url: str, timeout: float = 5.0) -> dict | None:
    """GET <url>/prompt -> JSON dict, None when offline (reference
    network.py:108-136)."""
    import aiohttp

    try:
        session = await get_client_session()
        async with session.get(
            f"{url}/prompt", timeout=aiohttp.ClientTimeout(total=timeout)
        ) as resp:
            if resp.status != 200:
                return None
            return await resp.json()
    except Exception as exc:  # noqa: BLE001
        debug_log(f"probe {url} failed: {exc}")
        return None
