"""The REST server: wire-compatible /distributed/* + /prompt endpoints.

Endpoint parity with the reference's aiohttp routes:
job_routes.py (queue, job_complete, prepare_job, clear_memory, check_file),
usdu_routes.py (heartbeat, submit_tiles, submit_image, request_image,
job_status), config_routes.py (config CRUD), worker_routes.py
(network_info, system_info, launch/stop/managed workers, log tail) and the
ComfyUI-side GET/POST /prompt that probes and dispatch rely on.
"""

from __future__ import annotations

import asyncio
import hashlib
import os
import platform
import socket
import time
import uuid

from aiohttp import web

from ..graph.executor import Executor, validate_prompt
from ..nodes.collector import decode_job_complete_envelope
from ..nodes.runtime import NodeRuntime, set_runtime
from ..server.job_state import JobState
from ..utils import constants
from ..utils.async_bridge import ServerLoop
from ..utils.config import (
    WORKER_TYPES,
    config_transaction,
    get_worker_by_id,
    load_config,
)
from ..utils.errors import PromptValidationError
from ..utils.logging import log
from . import network
from .orchestration import orchestrate_distributed_execution
from .queue_request import QueueRequestError, parse_queue_request_payload


def _int_query(request, name: str, default: int, lo: int, hi: int) -> int:
    try:
        return min(max(int(request.query.get(name, default)), lo), hi)
    except (TypeError, ValueError):
        return default


def _err(message: str, status: int = 400):
    return web.json_response({"error": message}, status=status)


class DistributedServer:
    """Application state + routes. One instance per process (master or
    worker role — the role only differs in config and env)."""

    def __init__(self, is_worker: bool = False, device: str | None = None):
        self.job_state = JobState()
        self.is_worker = is_worker
        self.device = device
        self.prompt_queue: asyncio.Queue = asyncio.Queue()
        self.executing = False
        self.history: dict[str, dict] = {}
        self.executor = Executor(context={"device": device} if device else {})
        self.registry = self.executor.registry
        self._exec_task: asyncio.Task | None = None
        self._log_buffer: list[str] = []
        self.managed = {}  # worker_id -> process handle

        rt = NodeRuntime(self.job_state)
        rt.probe_worker = self._probe_worker_by_id  # type: ignore
        set_runtime(rt)

    async def _probe_worker_by_id(self, worker_id: str):
        cfg = load_config()
        worker = get_worker_by_id(cfg, worker_id)
        if worker is None:
            return None
        return await network.probe_worker(network.build_worker_url(worker))

    # ------------------------------------------------------------------ app

    def build_app(self) -> web.Application:
        import json as _json

        @web.middleware
        async def bad_json_middleware(request, handler):
            """Malformed JSON bodies are a client error, not a 500
            (reference utils/network.py handle_api_error envelope)."""
            try:
                return await handler(request)
            except (_json.JSONDecodeError, UnicodeDecodeError):
                return _err("invalid JSON body")

        app = web.Application(client_max_size=constants.MAX_PAYLOAD_SIZE + 2**20,
                              middlewares=[bad_json_middleware])
        r = app.router
        r.add_get("/", self.get_panel)
        r.add_get("/panel", self.get_panel)
        r.add_get("/static/{tail:.+}", self.get_static)
        r.add_get("/prompt", self.get_prompt)
        r.add_post("/prompt", self.post_prompt)
        r.add_get("/object_info", self.get_object_info)
        r.add_post("/distributed/queue", self.post_queue)
        r.add_post("/distributed/job_complete", self.post_job_complete)
        r.add_post("/distributed/prepare_job", self.post_prepare_job)
        r.add_post("/distributed/clear_memory", self.post_clear_memory)
        r.add_post("/distributed/heartbeat", self.post_heartbeat)
        r.add_post("/distributed/submit_tiles", self.post_submit_tiles)
        r.add_post("/distributed/submit_image", self.post_submit_image)
        r.add_post("/distributed/request_image", self.post_request_image)
        r.add_get("/distributed/job_status/{job_id}", self.get_job_status)
        r.add_post("/distributed/job_status", self.post_job_status)
        r.add_get("/distributed/queue_status/{job_id}", self.get_queue_status)
        r.add_get("/distributed/config", self.get_config)
        r.add_post("/distributed/config/update_worker", self.post_update_worker)
        r.add_post("/distributed/config/delete_worker", self.post_delete_worker)
        r.add_post("/distributed/config/update_setting", self.post_update_setting)
        r.add_post("/distributed/config/update_master", self.post_update_master)
        r.add_post("/distributed/check_file", self.post_check_file)
        r.add_post("/upload/image", self.post_upload_image)
        r.add_get("/view", self.get_view)
        r.add_get("/history", self.get_history)
        r.add_get("/distributed/workflow_examples", self.get_workflow_examples)
        r.add_get("/history/{prompt_id}", self.get_history)
        r.add_get("/distributed/network_info", self.get_network_info)
        r.add_get("/distributed/system_info", self.get_system_info)
        r.add_post("/distributed/tunnel/start", self.post_tunnel_start)
        r.add_post("/distributed/tunnel/stop", self.post_tunnel_stop)
        r.add_get("/distributed/tunnel/status", self.get_tunnel_status)
        r.add_post("/distributed/launch_worker", self.post_launch_worker)
        r.add_post("/distributed/stop_worker", self.post_stop_worker)
        r.add_get("/distributed/managed_workers", self.get_managed_workers)
        r.add_get("/distributed/worker_status", self.get_worker_status)
        r.add_get("/distributed/local-worker-status",
                  self.get_local_worker_status)
        r.add_get("/distributed/remote_worker_log/{worker_id}",
                  self.get_remote_worker_log)
        r.add_post("/distributed/worker/clear_launching",
                   self.post_clear_launching)
        r.add_post("/distributed/auto_populate_workers",
                   self.post_auto_populate_workers)
        r.add_get("/distributed/worker_log", self.get_worker_log)
        r.add_get("/distributed/local_log", self.get_local_log)
        r.add_post("/distributed/load_image", self.post_load_image)
        r.add_post("/interrupt", self.post_interrupt)
        r.add_get("/distributed/ws", self.ws_handler)
        app.on_startup.append(self._on_startup)
        return app

    async def _on_startup(self, app):
        ServerLoop.set(asyncio.get_running_loop())
        self._exec_task = asyncio.create_task(self._execution_loop())

    # ----------------------------------------------------------- execution

    async def _execution_loop(self):
        loop = asyncio.get_running_loop()
        while True:
            prompt, client_id, prompt_id = await self.prompt_queue.get()
            self.executing = True
            from ..nodes.runtime import get_runtime

            get_runtime().clear_interrupt()  # interrupts are per-prompt
            saved = self.executor.context.setdefault("saved_images", [])
            n_before = len(saved)
            try:
                await loop.run_in_executor(None, self.executor.execute, prompt)
                log(f"prompt {prompt_id} done (client {client_id})")
                self._record_history(prompt_id, "success",
                                     saved[n_before:], None)
            except Exception as exc:  # noqa: BLE001
                log(f"prompt {prompt_id} FAILED: {exc!r}")
                self._record_history(prompt_id, "error",
                                     saved[n_before:], repr(exc))
            finally:
                self.executing = False

    HISTORY_LIMIT = 200

    def _record_history(self, prompt_id, status, new_files, error):
        """ComfyUI /history parity: clients poll it to learn a prompt's
        outcome and fetch outputs via /view."""
        entry = {
            "status": {"status_str": status,
                       "completed": status == "success",
                       "messages": [error] if error else []},
            "outputs": {"images": [
                self._output_entry(f) for f in new_files
            ]},
        }
        self.history[str(prompt_id)] = entry
        while len(self.history) > self.HISTORY_LIMIT:
            self.history.pop(next(iter(self.history)))

    def _output_entry(self, path) -> dict:
        """filename + subfolder relative to the output dir (the pair /view
        takes), mirroring ComfyUI's history format."""
        out_dir = os.path.abspath(
            str(self.executor.context.get("output_dir", "output")))
        ap = os.path.abspath(str(path))
        sub = ""
        if ap.startswith(out_dir + os.sep):
            rel = os.path.relpath(ap, out_dir)
            sub = os.path.dirname(rel)
        return {"filename": os.path.basename(ap), "subfolder": sub,
                "type": "output"}

    async def enqueue_local(self, prompt: dict, client_id: str) -> str:
        validate_prompt(prompt, self.registry)
        prompt_id = uuid.uuid4().hex
        await self.prompt_queue.put((prompt, client_id, prompt_id))
        return prompt_id

    def queue_remaining(self) -> int:
        return self.prompt_queue.qsize() + (1 if self.executing else 0)

    # -------------------------------------------------------------- routes

    async def get_panel(self, request):
        from pathlib import Path

        path = Path(__file__).parent / "static" / "panel.html"
        return web.Response(text=path.read_text(), content_type="text/html")

    async def get_static(self, request):
        """Panel assets (static/js/*). Confined to the static dir."""
        from pathlib import Path

        base = (Path(__file__).parent / "static").resolve()
        cand = (base / request.match_info["tail"]).resolve()
        if not str(cand).startswith(str(base) + os.sep) or not cand.is_file():
            raise web.HTTPNotFound
        ctype = ("application/javascript" if cand.suffix == ".js"
                 else "text/css" if cand.suffix == ".css" else "text/plain")
        return web.Response(text=cand.read_text(), content_type=ctype)

    async def get_prompt(self, request):
        return web.json_response(
            {"exec_info": {"queue_remaining": self.queue_remaining()}}
        )

    async def get_object_info(self, request):
        """Node schemas (ComfyUI /object_info parity: the UI and remote
        tooling discover node classes + input types here)."""
        out = {}
        for name in self.registry.names():
            cls = self.registry.get(name)
            try:
                inputs = cls.INPUT_TYPES() if hasattr(cls, "INPUT_TYPES") else {}
            except Exception:  # noqa: BLE001
                inputs = {}
            out[name] = {
                "input": inputs,
                "output": [str(t) for t in getattr(cls, "RETURN_TYPES", ())],
                "output_name": [str(n) for n in getattr(cls, "RETURN_NAMES", ())],
                "category": getattr(cls, "CATEGORY", ""),
                "output_node": bool(getattr(cls, "OUTPUT_NODE", False)),
            }
        return web.json_response(out)

    async def post_prompt(self, request):
        data = await request.json()
        prompt = data.get("prompt")
        if not isinstance(prompt, dict):
            return _err("missing prompt")
        try:
            prompt_id = await self.enqueue_local(prompt, data.get("client_id", ""))
        except PromptValidationError as exc:
            return web.json_response(
                {"error": str(exc), "node_errors": exc.node_errors}, status=400
            )
        return web.json_response({"prompt_id": prompt_id, "number": self.prompt_queue.qsize()})

    async def post_queue(self, request):
        try:
            payload = parse_queue_request_payload(await request.json())
        except QueueRequestError as exc:
            return _err(str(exc))
        result = await orchestrate_distributed_execution(
            payload, self.job_state, self.enqueue_local
        )
        # superset of the reference's response shape
        # (docs/comfyui-distributed-api.md): prompt_id + worker_count +
        # auto_prepare_supported alongside our participants/job_ids
        result.setdefault("prompt_id", result.get("master_prompt_id", ""))
        result["worker_count"] = sum(
            1 for p in result.get("participants", []) if p != "master"
        )
        result["auto_prepare_supported"] = True
        return web.json_response(result)

    async def post_job_complete(self, request):
        data = await request.json()
        try:
            item = decode_job_complete_envelope(data)
        except (ValueError, KeyError) as exc:
            return _err(str(exc))
        q = await self.job_state.get_queue_waiting(str(data["job_id"]))
        if q is None:
            return _err("unknown job (queue never created)", status=404)
        await q.put(item)
        return web.json_response({"status": "ok"})

    async def post_prepare_job(self, request):
        data = await request.json()
        job_id = data.get("job_id")
        if not job_id:
            return _err("missing job_id")
        await self.job_state.ensure_queue(str(job_id))
        return web.json_response({"status": "ready"})

    async def post_clear_memory(self, request):
        """Unload cached models + free the allocator (reference
        job_routes.py:160-203 unloads ComfyUI models on demand)."""
        import gc

        import torch

        from ..graph.builtin_nodes import _STACK_CACHE, _STACK_LOCK

        with _STACK_LOCK:
            n = len(_STACK_CACHE)
            _STACK_CACHE.clear()
        gc.collect()
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
        relayed = await self._fanout_to_workers(request,
                                                "/distributed/clear_memory")
        return web.json_response({"status": "ok", "unloaded_models": n,
                                  **relayed})

    # ---- USDU tile endpoints ---------------------------------------------

    async def post_heartbeat(self, request):
        data = await request.json()
        job = await self.job_state.get_tile_job(str(data.get("job_id", "")))
        if job is None:
            return _err("unknown job", status=404)
        job.worker_status[str(data.get("worker_id", ""))] = time.time()
        return web.json_response({"status": "ok"})

    async def post_submit_tiles(self, request):
        from .usdu_http import decode_tile_submission

        if request.content_type.startswith("multipart/"):
            # reference wire format: PNG file parts + tiles_metadata JSON
            # (upscale/payload_parsers.py:7-64)
            data = await self._parse_multipart_tiles(request)
        else:
            data = await request.json()
        job = await self.job_state.get_tile_job(str(data.get("job_id", "")))
        if job is None:
            return _err("unknown job", status=404)
        try:
            items = decode_tile_submission(data)
        except (TypeError, ValueError, KeyError) as exc:
            return _err(f"bad tile submission: {exc}")
        for item in items:
            await job.results.put(item)
        if data.get("is_last"):
            job.finished_workers.add(str(data.get("worker_id", "")))
        job.worker_status[str(data.get("worker_id", ""))] = time.time()
        return web.json_response({"status": "ok", "received": len(items)})

    async def _parse_multipart_tiles(self, request) -> dict:
        import base64
        import json as _json

        reader = await request.multipart()
        fields: dict = {}
        blobs: list[bytes] = []
        while True:
            part = await reader.next()
            if part is None:
                break
            if part.filename:
                blobs.append(await part.read())
            else:
                fields[part.name] = (await part.read()).decode()
        meta = _json.loads(fields.get("tiles_metadata", "[]"))
        tiles = []
        for i, m in enumerate(meta):
            if i < len(blobs):
                tiles.append({
                    "tile_idx": m.get("tile_idx", m.get("global_idx", i)),
                    "batch_idx": m.get("batch_idx", 0),
                    "image": base64.b64encode(blobs[i]).decode("ascii"),
                })
        return {
            "job_id": fields.get("job_id", ""),
            "worker_id": fields.get("worker_id", ""),
            "is_last": fields.get("is_last", "").lower() in ("1", "true"),
            "tiles": tiles,
        }

    async def post_submit_image(self, request):
        from ..utils.image import decode_png_base64

        data = await request.json()
        job = await self.job_state.get_tile_job(str(data.get("job_id", "")))
        if job is None:
            return _err("unknown job", status=404)
        try:
            item = {
                "image_idx": int(data.get("image_idx", 0)),
                "tensor": decode_png_base64(data["image"]),
                "worker_id": str(data.get("worker_id", "")),
                "is_last": bool(data.get("is_last", False)),
            }
        except (TypeError, ValueError, KeyError) as exc:
            return _err(f"bad submit_image payload: {exc}")
        await job.results.put(item)
        if data.get("is_last"):
            job.finished_workers.add(str(data.get("worker_id", "")))
        return web.json_response({"status": "ok"})

    async def post_request_image(self, request):
        """The pull scheduler (reference usdu_routes.py:168-215)."""
        data = await request.json()
        job = await self.job_state.get_tile_job(str(data.get("job_id", "")))
        if job is None:
            return _err("unknown job", status=404)
        wid = str(data.get("worker_id", ""))
        job.worker_status[wid] = time.time()
        pending = getattr(job, "pending_tasks", None)
        key = "tile_idx"
        if pending is None:
            pending = job.pending_images
            key = "image_idx"
        try:
            idx = await asyncio.wait_for(pending.get(), constants.QUEUE_POP_WAIT)
        except asyncio.TimeoutError:
            return web.json_response({key: None, "estimated_remaining": 0})
        job.assigned_to_workers[idx] = wid
        return web.json_response({
            key: idx,
            "estimated_remaining": pending.qsize(),
            "batched_static": getattr(job, "batched_static", False),
        })

    async def get_job_status(self, request):
        job_id = request.match_info["job_id"]
        job = await self.job_state.get_tile_job(job_id)
        return web.json_response({"ready": job is not None})

    async def post_job_status(self, request):
        data = await request.json()
        job = await self.job_state.get_tile_job(str(data.get("job_id", "")))
        return web.json_response({"ready": job is not None})

    async def get_queue_status(self, request):
        job_id = request.match_info["job_id"]
        async with self.job_state.jobs_lock:
            q = self.job_state.pending_jobs.get(job_id)
        return web.json_response(
            {"exists": q is not None, "pending": q.qsize() if q else 0}
        )

    # ---- config CRUD ------------------------------------------------------

    async def get_config(self, request):
        return web.json_response(load_config())

    async def post_update_worker(self, request):
        data = await request.json()
        wid = data.get("id")
        if wid is None:
            return _err("missing worker id")
        async with config_transaction() as cfg:
            worker = get_worker_by_id(cfg, wid)
            fields = {k: v for k, v in data.items() if k in (
                "name", "host", "port", "cuda_device", "enabled", "extra_args", "type")}
            if worker is None:
                required = {"name", "port"}
                if not required.issubset(data):
                    return _err("new worker needs name and port")
                worker = {"id": str(wid), "enabled": False, "type": "local",
                          "host": "", "cuda_device": 0, "extra_args": ""}
                worker.update(fields)
                if worker.get("type") not in WORKER_TYPES:
                    return _err(f"bad worker type {worker.get('type')!r}")
                cfg["workers"].append(worker)
            else:
                if "type" in fields and fields["type"] not in WORKER_TYPES:
                    return _err(f"bad worker type {fields['type']!r}")
                worker.update(fields)
        return web.json_response({"status": "ok"})

    async def post_delete_worker(self, request):
        data = await request.json()
        wid = str(data.get("id"))
        async with config_transaction() as cfg:
            before = len(cfg["workers"])
            cfg["workers"] = [w for w in cfg["workers"] if str(w.get("id")) != wid]
            if len(cfg["workers"]) == before:
                return _err("unknown worker", status=404)
        return web.json_response({"status": "ok"})

    ALLOWED_SETTINGS = (
        "debug", "auto_launch_workers", "stop_workers_on_master_exit",
        "master_delegate_only", "websocket_orchestration",
        "worker_timeout_seconds", "worker_probe_concurrency",
        "worker_prep_concurrency", "media_sync_concurrency",
        "media_sync_timeout_seconds", "has_auto_populated_workers",
    )

    async def post_update_setting(self, request):
        data = await request.json()
        key, value = data.get("key"), data.get("value")
        if key not in self.ALLOWED_SETTINGS:
            return _err(f"setting {key!r} not allowed")
        async with config_transaction() as cfg:
            cfg["settings"][key] = value
        return web.json_response({"status": "ok"})

    async def post_update_master(self, request):
        data = await request.json()
        async with config_transaction() as cfg:
            for k in ("host", "port", "cuda_device", "extra_args"):
                if k in data:
                    cfg["master"][k] = data[k]
        return web.json_response({"status": "ok"})

    # ---- introspection ----------------------------------------------------

    async def post_check_file(self, request):
        """md5-check a synced media file. ``filename`` resolves against THIS
        server's input dir (the cross-machine case — the sender's absolute
        path means nothing here); ``path`` is the same-filesystem fallback."""
        data = await request.json()
        candidates = []
        name = data.get("filename")
        input_dir = str(self.executor.context.get("input_dir", "input"))
        output_dir = str(self.executor.context.get("output_dir", "output"))
        if name:
            candidates.append(os.path.join(
                input_dir, os.path.basename(str(name))))
        if data.get("path"):
            # same-filesystem fallback, but never an arbitrary-path md5
            # oracle: the path must resolve inside the input or output dir
            p = os.path.realpath(str(data["path"]))
            for base in (input_dir, output_dir):
                rbase = os.path.realpath(base)
                if os.path.commonpath([rbase, p]) == rbase:
                    candidates.append(p)
                    break
        path = next((p for p in candidates if os.path.isfile(p)), None)
        if path is None:
            return web.json_response({"exists": False})
        h = hashlib.md5()
        with open(path, "rb") as fh:
            for chunk in iter(lambda: fh.read(1 << 20), b""):
                h.update(chunk)
        return web.json_response({"exists": True, "md5": h.hexdigest()})

    async def post_upload_image(self, request):
        """Multipart upload into the input dir (ComfyUI /upload/image
        parity — media sync pushes files here)."""
        from pathlib import Path

        reader = await request.multipart()
        input_dir = Path(self.executor.context.get("input_dir", "input"))
        input_dir.mkdir(parents=True, exist_ok=True)
        saved = []
        while True:
            part = await reader.next()
            if part is None:
                break
            if part.name in ("image", "file"):
                fname = os.path.basename(part.filename or "upload.bin")
                dest = input_dir / fname
                with open(dest, "wb") as fh:
                    while True:
                        chunk = await part.read_chunk(1 << 20)
                        if not chunk:
                            break
                        fh.write(chunk)
                saved.append(fname)
        return web.json_response({"saved": saved})

    async def get_workflow_examples(self, request):
        """List the shipped example workflows (repo ``workflows/`` dir or
        ``DISTGPU_WORKFLOWS_DIR``); ``?name=`` returns one prompt JSON."""
        import json as _json
        from pathlib import Path

        wf_dir = Path(os.environ.get("DISTGPU_WORKFLOWS_DIR", "workflows"))
        name = request.query.get("name")
        if name:
            path = wf_dir / os.path.basename(str(name))
            if not path.is_file() or path.suffix != ".json":
                return _err("not found", status=404)
            try:
                data = _json.loads(path.read_text())
            except ValueError:
                return _err("invalid workflow JSON", status=500)
            data.pop("_comment", None)
            return web.json_response({"name": path.name, "prompt": data})
        names = sorted(p.name for p in wf_dir.glob("*.json")) \
            if wf_dir.is_dir() else []
        return web.json_response({"workflows": names})

    async def get_history(self, request):
        pid = request.match_info.get("prompt_id")
        if pid is None:
            return web.json_response(self.history)
        entry = self.history.get(str(pid))
        return web.json_response({str(pid): entry} if entry else {})

    async def get_view(self, request):
        """Serve a saved output (ComfyUI GET /view parity: the panel and
        external clients fetch results by filename). ``type=input`` reads
        the input dir instead."""
        from pathlib import Path

        name = os.path.basename(str(request.query.get("filename", "")))
        if not name:
            return _err("missing filename")
        kind = request.query.get("type", "output")
        key = "input_dir" if kind == "input" else "output_dir"
        sub = Path(str(request.query.get("subfolder", "")))
        if sub.is_absolute() or ".." in sub.parts:
            return _err("bad subfolder")
        path = Path(self.executor.context.get(key, kind)) / sub / name
        if not path.is_file():
            return _err("not found", status=404)
        ext = name.lower().rsplit(".", 1)[-1]
        ctype = {"png": "image/png", "webp": "image/webp",
                 "wav": "audio/wav", "jpg": "image/jpeg",
                 "jpeg": "image/jpeg"}.get(ext, "application/octet-stream")
        return web.Response(body=path.read_bytes(), content_type=ctype)

    async def get_network_info(self, request):
        import torch

        hostname = socket.gethostname()
        try:
            ip = socket.gethostbyname(hostname)
        except OSError:
            ip = "127.0.0.1"
        return web.json_response({
            "hostname": hostname,
            "ips": [ip],
            "cuda_device_count": torch.cuda.device_count()
            if torch.cuda.is_available() else 0,
            "master_cuda_device": load_config()["master"].get("cuda_device", 0),
        })

    async def get_system_info(self, request):
        return web.json_response({
            "platform": platform.system().lower(),
            "path_separator": os.sep,
            "machine_id": hex(uuid.getnode()),
            "is_docker": os.path.exists("/.dockerenv"),
            "is_runpod": bool(os.environ.get("RUNPOD_POD_ID")),
            "is_worker": self.is_worker,
        })

    # ---- tunnel ------------------------------------------------------------

    @property
    def tunnel(self):
        if not hasattr(self, "_tunnel"):
            from .tunnel import TunnelManager

            self._tunnel = TunnelManager()
        return self._tunnel

    async def post_tunnel_start(self, request):
        from ..utils.errors import TunnelError

        data = await request.json()
        port = int(data.get("port", load_config()["master"].get("port", 8188)))
        try:
            url = await self.tunnel.start(port)
        except TunnelError as exc:
            return _err(str(exc), status=500)
        return web.json_response({"status": "started", "url": url})

    async def post_tunnel_stop(self, request):
        await self.tunnel.stop()
        return web.json_response({"status": "stopped"})

    async def get_tunnel_status(self, request):
        return web.json_response(self.tunnel.status())

    async def get_worker_status(self, request):
        """Master-side worker probe for the panel (browsers can't
        cross-origin-probe workers; reference worker_routes.py:536-603
        serves the same role)."""
        wid = request.query.get("id")
        worker = get_worker_by_id(load_config(), wid) if wid else None
        if worker is None:
            return _err("unknown worker", status=404)
        from .workers import is_process_alive, load_managed_pid

        info = await network.probe_worker(network.build_worker_url(worker))
        pid = load_managed_pid(str(wid))
        return web.json_response({
            "id": str(wid),
            "online": info is not None,
            "queue_remaining": (info or {}).get("exec_info", {}).get(
                "queue_remaining"),
            "managed": pid is not None,
            "pid_alive": bool(pid and is_process_alive(pid)),
        })

    async def get_local_worker_status(self, request):
        """Status of every LOCAL worker in one call (reference
        worker_routes.py:536-603: the panel polls this instead of N probes).
        Disabled workers are reported without being probed."""
        cfg = load_config()
        local = [w for w in cfg.get("workers", [])
                 if network.normalize_host(w.get("host", "")) in
                 ("", "localhost", "127.0.0.1")]

        async def one(w):
            wid = str(w.get("id"))
            if not w.get("enabled", False):
                return wid, {"online": False, "enabled": False,
                             "processing": False, "queue_count": 0}
            info = await network.probe_worker(network.build_worker_url(w),
                                              timeout=2.0)
            if info is None:
                return wid, {"online": False, "enabled": True,
                             "processing": False, "queue_count": 0,
                             "error": "Unavailable"}
            qr = info.get("exec_info", {}).get("queue_remaining", 0)
            return wid, {"online": True, "enabled": True,
                         "processing": qr > 0, "queue_count": qr}

        results = await asyncio.gather(*(one(w) for w in local))
        return web.json_response({"status": "success",
                                  "worker_statuses": dict(results)})

    async def get_remote_worker_log(self, request):
        """Proxy a remote worker's in-memory log (reference
        worker_routes.py:649-695); local workers use the file-tail
        endpoint instead."""
        wid = str(request.match_info["worker_id"]).strip()
        worker = get_worker_by_id(load_config(), wid)
        if worker is None:
            return _err("unknown worker", status=404)
        if not network.normalize_host(worker.get("host", "")):
            return _err(f"worker {wid} is local; use /distributed/worker_log",
                        status=400)
        lines = _int_query(request, "lines", 300, 1, 3000)
        import aiohttp

        session = await network.get_client_session()
        url = network.build_worker_url(worker) + "/distributed/local_log"
        try:
            async with session.get(
                url, params={"lines": str(lines)},
                timeout=aiohttp.ClientTimeout(total=5),
            ) as resp:
                if resp.status >= 400:
                    return _err(f"worker returned HTTP {resp.status}",
                                status=resp.status)
                return web.json_response(await resp.json())
        except Exception as exc:
            return _err(f"worker unreachable: {exc}", status=502)

    async def post_clear_launching(self, request):
        """Clear a worker's 'launching' marker once it is confirmed up
        (reference worker_routes.py:115-135)."""
        data = await request.json()
        wid = data.get("worker_id")
        if wid is None:
            return _err("missing worker_id")
        if get_worker_by_id(load_config(), wid) is None:
            return _err("unknown worker", status=404)
        async with config_transaction() as cfg:
            entry = cfg.get("managed_processes", {}).get(str(wid))
            if entry:
                entry.pop("launching", None)
        return web.json_response({"status": "ok"})

    async def post_auto_populate_workers(self, request):
        """One local worker per GPU beyond the master's, first run only
        (reference web/masterDetection.js:3-147 does this client-side on
        first launch; here it is a server action the panel calls)."""
        import torch

        data = await request.json() if request.can_read_body else {}
        force = bool(data.get("force"))
        n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
        created = []
        async with config_transaction() as cfg:
            if cfg["settings"].get("has_auto_populated_workers") and not force:
                return web.json_response({"status": "already_populated",
                                          "created": []})
            master_dev = cfg["master"].get("cuda_device", 0)
            base_port = int(cfg["master"].get("port", 8188))
            existing_devs = {w.get("cuda_device") for w in cfg["workers"]}
            for dev in range(n_gpus):
                if dev == master_dev or dev in existing_devs:
                    continue
                wid = f"auto_gpu{dev}"
                cfg["workers"].append({
                    "id": wid, "name": f"GPU {dev}", "host": "",
                    "port": base_port + 1 + dev, "cuda_device": dev,
                    "enabled": True, "type": "local", "extra_args": "",
                })
                created.append(wid)
            cfg["settings"]["has_auto_populated_workers"] = True
        return web.json_response({"status": "ok", "created": created,
                                  "gpu_count": n_gpus})

    # ---- worker process management ----------------------------------------

    async def post_launch_worker(self, request):
        from .workers import launch_worker

        data = await request.json()
        cfg = load_config()
        worker = get_worker_by_id(cfg, data.get("id"))
        if worker is None:
            return _err("unknown worker", status=404)
        try:
            handle = launch_worker(worker)
        except Exception as exc:  # noqa: BLE001
            return _err(f"launch failed: {exc}", status=500)
        self.managed[str(worker["id"])] = handle
        return web.json_response({"status": "launched", "pid": handle.pid})

    async def post_stop_worker(self, request):
        from .workers import stop_worker

        data = await request.json()
        wid = str(data.get("id"))
        handle = self.managed.pop(wid, None)
        stop_worker(handle, wid)
        return web.json_response({"status": "stopped"})

    async def get_managed_workers(self, request):
        out = {}
        for wid, handle in self.managed.items():
            out[wid] = {"pid": handle.pid, "alive": handle.poll() is None}
        return web.json_response({"managed": out})

    async def get_worker_log(self, request):
        """Tail a managed worker's log file (reference
        worker_routes.py:606-646 behavior: efficient tail-seek)."""
        from .workers import worker_log_path

        wid = request.query.get("id", "")
        lines = _int_query(request, "lines", 100, 1, 10000)
        path = worker_log_path(wid)
        if not path.exists():
            return _err("no log for worker", status=404)
        with open(path, "rb") as fh:
            fh.seek(0, os.SEEK_END)
            size = fh.tell()
            fh.seek(max(0, size - 64 * 1024))
            tail = fh.read().decode(errors="replace").splitlines()[-lines:]
        return web.json_response({"log": "\n".join(tail)})

    async def get_local_log(self, request):
        from ..utils.logging import LOG_BUFFER

        lines = _int_query(request, "lines", 100, 1, 10000)
        return web.json_response({"log": "\n".join(list(LOG_BUFFER)[-lines:])})

    async def post_load_image(self, request):
        """Return an input image as a base64 PNG (reference
        job_routes.py:238-258: workers fetch inputs from the master)."""
        from pathlib import Path

        from ..utils.image import decode_png_bytes, encode_png_base64

        data = await request.json()
        name = os.path.basename(str(data.get("filename", "")))
        path = Path(self.executor.context.get("input_dir", "input")) / name
        if not path.is_file():
            return _err("not found", status=404)
        tensor = decode_png_bytes(path.read_bytes())
        return web.json_response({"image": encode_png_base64(tensor)})

    async def post_interrupt(self, request):
        """User interrupt: flags the node runtime so every wait loop raises
        (reference checks comfy.model_management interrupts in each loop).
        With {"fanout": true} the master also relays the interrupt to every
        enabled worker (the reference's interrupt-all lives client-side in
        web/workerUtils.js:4-125; here the master fans out server-side)."""
        from ..nodes.runtime import get_runtime

        get_runtime().interrupt()
        relayed = await self._fanout_to_workers(request, "/interrupt")
        return web.json_response({"status": "interrupted", **relayed})

    async def _fanout_to_workers(self, request, path: str) -> dict:
        """POST ``path`` to every enabled worker when the request body asks
        for fanout; returns {} otherwise."""
        try:
            data = await request.json() if request.can_read_body else {}
        except Exception:
            data = {}
        if not data.get("fanout"):
            return {}
        import aiohttp

        session = await network.get_client_session()

        async def one(w):
            try:
                url = network.build_worker_url(w) + path
                async with session.post(
                    url, json={}, timeout=aiohttp.ClientTimeout(total=5)
                ) as resp:
                    return str(w.get("id")), resp.status == 200
            except Exception:
                return str(w.get("id")), False

        workers = [w for w in load_config().get("workers", [])
                   if w.get("enabled")]
        results = await asyncio.gather(*(one(w) for w in workers))
        return {"fanout": dict(results)}

    async def ws_handler(self, request):
        """WebSocket orchestration endpoint: accepts dispatch_prompt
        messages, replies dispatch_ack with the request id (reference
        worker_routes.py:43-112 + dispatch.py:62-95)."""
        import json as _json

        ws = web.WebSocketResponse()
        await ws.prepare(request)
        async for msg in ws:
            if msg.type != web.WSMsgType.TEXT:
                continue
            try:
                data = _json.loads(msg.data)
            except ValueError:
                continue
            if data.get("type") == "dispatch_prompt":
                rid = data.get("request_id")
                try:
                    await self.enqueue_local(data.get("prompt") or {},
                                             data.get("client_id", ""))
                    await ws.send_json({"type": "dispatch_ack",
                                        "request_id": rid, "ok": True})
                except PromptValidationError as exc:
                    await ws.send_json({"type": "dispatch_ack",
                                        "request_id": rid, "ok": False,
                                        "error": str(exc)})
            elif data.get("type") == "ping":
                await ws.send_json({"type": "pong"})
        return ws


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8188)
    ap.add_argument("--listen", default="0.0.0.0")
    ap.add_argument("--worker", action="store_true",
                    default=os.environ.get("DISTGPU_IS_WORKER") == "1")
    args = ap.parse_args()

    import torch

    device = "cuda:0" if torch.cuda.is_available() else None
    server = DistributedServer(is_worker=args.worker, device=device)
    app = server.build_app()
    role = "worker" if args.worker else "master"

    if not args.worker:
        # adopt/clean managed workers from a previous run, then auto-launch
        # (reference workers/startup.py: delayed timer + signal/atexit)
        from .workers import adopt_or_cleanup_managed, launch_worker, stop_worker

        adopted = adopt_or_cleanup_managed()
        if adopted:
            log(f"re-adopted managed workers: {adopted}")
        cfg = load_config()
        if cfg["settings"].get("auto_launch_workers"):
            import threading

            def delayed():
                for w in cfg.get("workers", []):
                    if w.get("enabled") and w.get("type", "local") == "local" \
                            and not w.get("host"):
                        try:
                            server.managed[str(w["id"])] = launch_worker(w)
                        except Exception as exc:  # noqa: BLE001
                            log(f"auto-launch of {w.get('id')} failed: {exc}")

            threading.Timer(2.0, delayed).start()

        import atexit
        import signal as _signal

        def cleanup(*_a):
            if load_config()["settings"].get("stop_workers_on_master_exit", True):
                for wid, handle in list(server.managed.items()):
                    stop_worker(handle, wid)

        atexit.register(cleanup)
        for sig in (_signal.SIGTERM, _signal.SIGINT, _signal.SIGHUP):
            try:
                _signal.signal(sig, lambda *_a: (cleanup(), os._exit(0)))
            except (ValueError, OSError):
                pass

    log(f"starting {role} server on {args.listen}:{args.port} (device={device})")
    web.run_app(app, host=args.listen, port=args.port, print=None)


if __name__ == "__main__":
    main()
