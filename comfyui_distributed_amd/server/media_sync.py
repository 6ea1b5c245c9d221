"""Media sync: push input files referenced by a prompt to remote workers.

Reference counterpart: api/orchestration/media_sync.py — scan prompt inputs
for media filename references (:14-33,70-81), md5-check against the
worker's /distributed/check_file and upload when missing/stale (:146-193),
rewrite prompt inputs to worker-local paths (:84-102) with cross-platform
separator conversion using the worker's /distributed/system_info
(:36-67,127-143). Local workers share the filesystem and skip all of this.
"""

from __future__ import annotations

import asyncio
import hashlib
import re
from pathlib import Path

from ..graph.prompt import PromptGraph, is_link
from ..utils import constants
from ..utils.logging import debug_log
from . import network

#: input keys that may carry media-file references
MEDIA_INPUT_KEYS = ("image", "video", "audio", "file")

#: filename with a media extension
_MEDIA_RE = re.compile(
    r".+\.(png|jpg|jpeg|webp|gif|bmp|tiff?|mp4|webm|mov|avi|mkv|wav|mp3|flac|ogg)$",
    re.IGNORECASE,
)


def find_media_references(prompt: dict) -> list[tuple[str, str, str]]:
    """(node_id, input_name, filename) for every media-file reference."""
    graph = PromptGraph(prompt)
    out = []
    for nid in graph.node_ids():
        for name, value in graph.inputs(nid).items():
            if name not in MEDIA_INPUT_KEYS or is_link(value):
                continue
            if isinstance(value, str) and _MEDIA_RE.match(value.strip()):
                if value.startswith("synthetic:"):
                    continue
                out.append((nid, name, value.strip()))
    return out


def md5_of(path: Path) -> str:
    h = hashlib.md5()
    with open(path, "rb") as fh:
        for chunk in iter(lambda: fh.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


def convert_path_for_platform(path: str, separator: str) -> str:
    """Rewrite path separators for the worker's platform (reference
    :36-67)."""
    if separator == "\\":
        return path.replace("/", "\\")
    return path.replace("\\", "/")


async def fetch_worker_system_info(worker_url: str) -> dict:
    import aiohttp

    session = await network.get_client_session()
    try:
        async with session.get(
            f"{worker_url}/distributed/system_info",
            timeout=aiohttp.ClientTimeout(total=10),
        ) as r:
            if r.status == 200:
                return await r.json()
    except Exception as exc:  # noqa: BLE001
        debug_log(f"system_info fetch failed: {exc}")
    return {"path_separator": "/"}


async def is_local_worker(worker: dict) -> bool:
    """Same-machine check (reference workers/detection.py:11-47): loopback
    host or type "local" is local by definition; a "cloud" worker may still
    be co-located (e.g. one container per GPU on one box) — compare its
    machine id from /distributed/system_info against ours."""
    host = network.normalize_host(worker.get("host", "")) or ""
    if host in ("", "localhost", "127.0.0.1", "0.0.0.0") or \
            worker.get("type", "local") == "local":
        return True
    if worker.get("type") == "cloud":
        import uuid

        info = await fetch_worker_system_info(network.build_worker_url(worker))
        return info.get("machine_id") == hex(uuid.getnode())
    return False


async def sync_worker_media(prompt: dict, worker: dict,
                            input_dir: str | Path = "input",
                            concurrency: int | None = None) -> dict:
    """Upload referenced media files the worker is missing; returns the
    prompt with rewritten worker-local paths. Workers on this machine
    (shared FS) are a no-op (reference behavior: sync only for
    genuinely-remote workers)."""
    refs = find_media_references(prompt)
    if not refs or await is_local_worker(worker):
        return prompt
    url = network.build_worker_url(worker)
    info = await fetch_worker_system_info(url)
    sep = info.get("path_separator", "/")
    input_dir = Path(input_dir)
    sem = asyncio.Semaphore(concurrency or constants.MEDIA_SYNC_CONCURRENCY)
    import aiohttp

    session = await network.get_client_session()

    async def sync_one(filename: str):
        local = input_dir / filename
        if not local.is_file():
            debug_log(f"media sync: missing local file {local}")
            return
        local_md5 = md5_of(local)
        async with sem:
            try:
                async with session.post(
                    f"{url}/distributed/check_file",
                    json={"path": str(local), "filename": filename},
                    timeout=aiohttp.ClientTimeout(total=15),
                ) as r:
                    remote = await r.json() if r.status == 200 else {}
            except Exception:  # noqa: BLE001
                remote = {}
            if remote.get("exists") and remote.get("md5") == local_md5:
                return
            data = aiohttp.FormData()
            # bytes, not an open handle: FormData would leak the fd
            data.add_field("image", local.read_bytes(), filename=filename)
            try:
                async with session.post(
                    f"{url}/upload/image", data=data,
                    timeout=aiohttp.ClientTimeout(
                        total=constants.MEDIA_SYNC_TIMEOUT_SECONDS),
                ) as r:
                    debug_log(f"media sync: uploaded {filename} -> {r.status}")
            except Exception as exc:  # noqa: BLE001
                debug_log(f"media sync: upload {filename} failed: {exc}")

    await asyncio.gather(*(sync_one(f) for _n, _k, f in set(refs)))

    out = PromptGraph(prompt).copy()
    for nid, name, filename in refs:
        out.inputs(nid)[name] = convert_path_for_platform(filename, sep)
    return out.raw
