"""DistributedCollector node — the DP gather for the HTTP (cross-node) path.

Signature + behavior parity with reference nodes/collector.py:24-469:
worker role serializes each image to base64 PNG and POSTs
``/distributed/job_complete`` envelopes (``is_last`` on the final one, audio
riding on it); master role drains the job's asyncio queue with sliced
waits, interrupt checks, activity-based timeout with busy-probe grace, then
deterministically reorders (master batch, then workers in enabled order,
then stragglers sorted) and concatenates.

Intra-node (8 GPUs, one process per GPU) this node is NOT used — the
collector is one RCCL gather (parallel/collector.py). This path serves
remote/cloud workers over the wire-compatible REST API.
"""

from __future__ import annotations

import asyncio
import json
import time

import torch

from ..utils import constants
from ..utils.audio import concat_audio, decode_audio_payload, encode_audio_payload
from ..utils.async_bridge import run_async_in_server_loop
from ..utils.image import decode_png_base64, encode_png_base64
from ..utils.logging import debug_log, log
from .runtime import get_runtime

EMPTY_AUDIO = {"waveform": torch.zeros(1, 2, 1), "sample_rate": 44100}


def _is_real_audio(audio) -> bool:
    """False for None and for the 1-sample EMPTY_AUDIO placeholder (its
    numel is 2, so a numel check would ship empty envelopes on image-only
    workflows)."""
    return audio is not None and audio["waveform"].shape[-1] > 1


class DistributedCollectorNode:
    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "images": ("IMAGE",),
                "load_balance": ("BOOLEAN", {"default": False}),
            },
            "optional": {"audio": ("AUDIO",)},
            "hidden": {
                "multi_job_id": ("STRING", {"default": ""}),
                "is_worker": ("BOOLEAN", {"default": False}),
                "master_url": ("STRING", {"default": ""}),
                "enabled_worker_ids": ("STRING", {"default": "[]"}),
                "worker_batch_size": ("INT", {"default": 1, "min": 1, "max": 1024}),
                "worker_id": ("STRING", {"default": ""}),
                "pass_through": ("BOOLEAN", {"default": False}),
                "delegate_only": ("BOOLEAN", {"default": False}),
            },
        }

    RETURN_TYPES = ("IMAGE", "AUDIO")
    RETURN_NAMES = ("images", "audio")
    FUNCTION = "run"
    CATEGORY = "image"

    def run(self, images, load_balance=False, audio=None, multi_job_id="",
            is_worker=False, master_url="", enabled_worker_ids="[]",
            worker_batch_size=1, worker_id="", pass_through=False,
            delegate_only=False):
        audio = audio if audio is not None else EMPTY_AUDIO
        if not multi_job_id or pass_through:
            return (images, audio)
        if is_worker:
            run_async_in_server_loop(
                self.send_batch_to_master(images, audio, multi_job_id,
                                          master_url, worker_id),
                timeout=300.0,
            )
            return (images, audio)
        out_images, out_audio = run_async_in_server_loop(
            self.collect_on_master(images, audio, multi_job_id,
                                   enabled_worker_ids, delegate_only),
            timeout=None,
        )
        return (out_images, out_audio)

    # ---- worker side ------------------------------------------------------

    async def send_batch_to_master(self, images, audio, job_id, master_url,
                                   worker_id):
        rt = get_runtime()
        url = f"{master_url}/distributed/job_complete"
        n = images.shape[0]
        has_audio = _is_real_audio(audio)
        if n == 0:
            payload = {
                "job_id": str(job_id), "worker_id": str(worker_id),
                "batch_idx": 0, "image": None, "is_last": True,
            }
            if has_audio:
                payload["audio"] = encode_audio_payload(audio)
            await rt.post_json(url, payload)
            return
        for i in range(n):
            payload = {
                "job_id": str(job_id),
                "worker_id": str(worker_id),
                "batch_idx": i,
                "image": encode_png_base64(images[i : i + 1].cpu()),
                "is_last": i == n - 1,
            }
            if has_audio and i == n - 1:
                payload["audio"] = encode_audio_payload(audio)
            await rt.post_json(url, payload)
        debug_log(f"collector worker {worker_id}: sent {n} images for {job_id}")

    # ---- master side ------------------------------------------------------

    async def collect_on_master(self, images, audio, job_id,
                                enabled_worker_ids, delegate_only):
        rt = get_runtime()
        enabled = [str(w) for w in json.loads(enabled_worker_ids or "[]")]
        queue = await rt.job_state.ensure_queue(job_id)
        local_images = None if delegate_only else images.cpu()

        worker_tensors: dict[str, dict[int, torch.Tensor]] = {}
        worker_audio: dict[str, dict] = {}
        workers_done: set[str] = set()
        expected = set(enabled)
        timeout = constants.HEARTBEAT_TIMEOUT
        last_activity = time.monotonic()

        try:
            while expected - workers_done:
                rt.throw_if_interrupted()
                try:
                    item = await asyncio.wait_for(
                        queue.get(), constants.COLLECTOR_SLICE_TIMEOUT
                    )
                except asyncio.TimeoutError:
                    idle = time.monotonic() - last_activity
                    if idle <= timeout:
                        continue
                    # probe the stragglers: busy workers get grace
                    still_busy = False
                    for wid in sorted(expected - workers_done):
                        info = await rt.probe_worker(wid)
                        if info and info.get("exec_info", {}).get("queue_remaining", 0):
                            still_busy = True
                    if still_busy:
                        last_activity = time.monotonic()
                        debug_log(f"collector {job_id}: stragglers busy — grace")
                        continue
                    log(f"collector {job_id}: timed out waiting for "
                        f"{sorted(expected - workers_done)} — continuing without")
                    break
                last_activity = time.monotonic()
                wid = str(item.get("worker_id"))
                tensor = item.get("tensor")
                if tensor is not None:
                    worker_tensors.setdefault(wid, {})[int(item.get("image_index", 0))] = tensor
                if item.get("audio") is not None:
                    worker_audio[wid] = item["audio"]
                if item.get("is_last"):
                    workers_done.add(wid)
        finally:
            await rt.job_state.drop_queue(job_id)

        images_out = self._reorder_and_combine(
            local_images, worker_tensors, enabled
        )
        audio_out = self._combine_audio(
            None if delegate_only else audio, worker_audio, enabled
        )
        return images_out, audio_out

    @staticmethod
    def _reorder_and_combine(local_images, worker_tensors, enabled_order):
        """Master batch first, then workers in enabled order, then stragglers
        sorted; each worker's images by ascending batch index
        (reference collector.py:193-236)."""
        parts = []
        if local_images is not None and local_images.shape[0] > 0:
            parts.append(local_images)
        seen = set()
        for wid in enabled_order:
            if wid in worker_tensors:
                seen.add(wid)
                imgs = worker_tensors[wid]
                parts.extend(imgs[i] for i in sorted(imgs))
        for wid in sorted(worker_tensors.keys()):
            if wid not in seen:
                imgs = worker_tensors[wid]
                parts.extend(imgs[i] for i in sorted(imgs))
        if not parts:
            return torch.zeros(0, 64, 64, 3)
        parts = [p if p.dim() == 4 else p[None] for p in parts]
        return torch.cat(parts, dim=0)

    @staticmethod
    def _combine_audio(local_audio, worker_audio, enabled_order):
        audios = []
        if _is_real_audio(local_audio):
            audios.append(local_audio)
        seen = set()
        for wid in enabled_order:
            if wid in worker_audio:
                seen.add(wid)
                audios.append(decode_audio_payload(worker_audio[wid]))
        for wid in sorted(worker_audio.keys()):
            if wid not in seen:
                audios.append(decode_audio_payload(worker_audio[wid]))
        if not audios:
            return EMPTY_AUDIO
        try:
            return concat_audio(audios)
        except Exception as exc:  # noqa: BLE001
            log(f"collector: audio combine failed: {exc}")
            return audios[0]


def decode_job_complete_envelope(payload: dict) -> dict:
    """Validate + decode a /distributed/job_complete envelope into the queue
    item the master loop consumes (reference job_routes.py:273-307)."""
    for key in ("job_id", "worker_id"):
        if key not in payload:
            raise ValueError(f"job_complete missing '{key}'")
    item = {
        "worker_id": str(payload["worker_id"]),
        "image_index": int(payload.get("batch_idx", 0)),
        "is_last": bool(payload.get("is_last", False)),
        "tensor": None,
        "audio": payload.get("audio"),
    }
    if payload.get("image"):
        item["tensor"] = decode_png_base64(payload["image"])
    return item
