"""Master-side queue orchestration (parity: reference
api/queue_orchestration.py:200-418 + api/orchestration/dispatch.py).

Flow: resolve enabled workers from config -> probe concurrently under a
semaphore -> honor delegate-only -> (load_balance) pick one least-busy
participant -> per-node job ids -> pre-create collector queues -> build
per-participant prompts (prune + overrides) -> dispatch POST /prompt to
workers -> queue the master's own prompt locally.
"""

from __future__ import annotations

import asyncio
import itertools

from ..graph import PromptGraph, transform
from ..graph.prompt import (DISTRIBUTED_OUTPUT_CLASSES, NODE_CLASS_COLLECTOR,
                            NODE_CLASS_UPSCALE)
from ..utils import constants
from ..utils.config import enabled_workers, is_master_delegate_only, load_config
from ..utils.logging import log, trace_debug
from . import network
from .queue_request import QueueRequestPayload

_round_robin = itertools.count()


async def probe_workers(workers: list[dict],
                        concurrency: int | None = None) -> dict[str, dict | None]:
    """Probe all workers concurrently; returns {worker_id: probe_json|None}."""
    sem = asyncio.Semaphore(concurrency or constants.WORKER_PROBE_CONCURRENCY)

    async def one(w):
        async with sem:
            return str(w["id"]), await network.probe_worker(network.build_worker_url(w))

    results = await asyncio.gather(*(one(w) for w in workers))
    return dict(results)


def select_least_busy(candidates: list[tuple[str, dict | None]]) -> str:
    """Pick the participant with the smallest queue_remaining; round-robin
    among idle ones (reference dispatch.py:225-268)."""
    depths = []
    for pid, info in candidates:
        depth = (info or {}).get("exec_info", {}).get("queue_remaining", 0)
        depths.append((depth, pid))
    min_depth = min(d for d, _ in depths)
    idle = [pid for d, pid in depths if d == min_depth]
    return idle[next(_round_robin) % len(idle)]


async def dispatch_worker_prompt(worker: dict, prompt: dict, client_id: str,
                                 timeout: float = 30.0,
                                 use_websocket: bool | None = None) -> bool:
    """Dispatch to a worker: WS dispatch_prompt/dispatch_ack when
    websocket_orchestration is on (reference dispatch.py:62-95), falling
    back to POST /prompt (:98-141)."""
    import secrets

    import aiohttp

    url = network.build_worker_url(worker)
    if use_websocket is None:
        use_websocket = bool(
            load_config().get("settings", {}).get("websocket_orchestration", True)
        )
    if use_websocket:
        try:
            session = await network.get_client_session()
            rid = secrets.token_hex(8)
            async with session.ws_connect(
                f"{url}/distributed/ws",
                timeout=aiohttp.ClientWSTimeout(ws_close=timeout),
            ) as ws:
                await ws.send_json({"type": "dispatch_prompt", "request_id": rid,
                                    "prompt": prompt, "client_id": client_id})
                async for msg in ws:
                    if msg.type != aiohttp.WSMsgType.TEXT:
                        break
                    data = msg.json()
                    if (data.get("type") == "dispatch_ack"
                            and data.get("request_id") == rid):
                        return bool(data.get("ok"))
        except Exception as exc:  # noqa: BLE001
            log(f"WS dispatch to {url} failed ({exc}); falling back to POST")
    try:
        session = await network.get_client_session()
        async with session.post(
            f"{url}/prompt", json={"prompt": prompt, "client_id": client_id},
            timeout=aiohttp.ClientTimeout(total=timeout),
        ) as resp:
            return resp.status == 200
    except Exception as exc:  # noqa: BLE001
        log(f"dispatch to {url} failed: {exc}")
        return False


async def orchestrate_distributed_execution(
    payload: QueueRequestPayload,
    job_state,
    enqueue_local,
    config: dict | None = None,
) -> dict:
    """Returns {"status": ..., "participants": [...], "job_ids": {...}}.

    ``enqueue_local`` is an async callable(prompt, client_id) that queues the
    master's own prompt into the local executor.
    """
    cfg = config or load_config()
    trace = payload.trace_execution_id or "queue"
    graph = PromptGraph(payload.prompt)

    # ---- resolve + probe workers -----------------------------------------
    requested = set(payload.enabled_worker_ids)
    workers = [w for w in enabled_workers(cfg) if str(w["id"]) in requested]
    probes = await probe_workers(workers)
    online = [w for w in workers if probes.get(str(w["id"])) is not None]
    offline = [str(w["id"]) for w in workers if probes.get(str(w["id"])) is None]
    if offline:
        trace_debug(trace, f"offline workers skipped: {offline}")

    delegate = payload.delegate_master or is_master_delegate_only(cfg)
    if delegate and not online:
        trace_debug(trace, "delegate-only with zero online workers — master fallback")
        delegate = False
    if delegate and graph.nodes_of_class(NODE_CLASS_UPSCALE):
        # reference limitation kept: delegate-only does not support USDU
        # (the master must coordinate the tile job); fall back to full
        # master participation (docs/comfyui-distributed-api.md)
        trace_debug(trace, "delegate-only with USDU — master fallback")
        delegate = False
    enabled_ids = [str(w["id"]) for w in online]

    # ---- load balancing ---------------------------------------------------
    load_balance = any(
        graph.inputs(nid).get("load_balance")
        for nid in graph.nodes_of_class(NODE_CLASS_COLLECTOR)
    )
    if load_balance and online:
        candidates: list[tuple[str, dict | None]] = [
            (str(w["id"]), probes.get(str(w["id"]))) for w in online
        ]
        if not delegate:
            candidates.insert(0, ("master", {"exec_info": {"queue_remaining": 0}}))
        chosen = select_least_busy(candidates)
        trace_debug(trace, f"load_balance chose {chosen}")
        if chosen == "master":
            pid = await enqueue_local(payload.prompt, payload.client_id)
            return {"status": "queued", "participants": ["master"],
                    "job_ids": {}, "master_prompt_id": pid}
        worker = next(w for w in online if str(w["id"]) == chosen)
        pruned = transform.prune_prompt_for_worker(graph)
        job_id_map = transform.generate_job_id_map(graph, prefix=None)
        wp = transform.apply_participant_overrides(
            pruned, is_master=False, participant_id=chosen,
            enabled_worker_ids=[chosen], job_id_map=job_id_map,
            master_url=network.build_master_callback_url(cfg["master"], worker),
        )
        # master still collects
        for jid in job_id_map.values():
            await job_state.ensure_queue(jid)
        master_prompt = transform.apply_participant_overrides(
            transform.prepare_delegate_master_prompt(
                graph, graph.nodes_of_class(NODE_CLASS_COLLECTOR)),
            is_master=True, participant_id="master",
            enabled_worker_ids=[chosen], job_id_map=job_id_map,
        )
        ok = await dispatch_worker_prompt(worker, wp.raw, payload.client_id)
        if not ok:
            pid = await enqueue_local(payload.prompt, payload.client_id)
            return {"status": "queued", "participants": ["master"],
                    "job_ids": {}, "master_prompt_id": pid}
        pid = await enqueue_local(master_prompt.raw, payload.client_id)
        return {"status": "queued", "participants": [chosen],
                "master_prompt_id": pid,
                "job_ids": job_id_map}

    # ---- regular fan-out --------------------------------------------------
    job_id_map = transform.generate_job_id_map(graph, prefix=None)
    if not graph.nodes_of_class(*DISTRIBUTED_OUTPUT_CLASSES) or not online:
        # nothing distributed (or nobody to distribute to): run locally
        pid = await enqueue_local(payload.prompt, payload.client_id)
        return {"status": "queued", "participants": ["master"], "job_ids": {},
                "master_prompt_id": pid}

    for jid in job_id_map.values():
        await job_state.ensure_queue(jid)

    sem = asyncio.Semaphore(constants.WORKER_PREP_CONCURRENCY)

    async def prep_and_dispatch(worker):
        async with sem:
            wid = str(worker["id"])
            pruned = transform.prune_prompt_for_worker(graph)
            wp = transform.apply_participant_overrides(
                pruned, is_master=False, participant_id=wid,
                enabled_worker_ids=enabled_ids, job_id_map=job_id_map,
                master_url=network.build_master_callback_url(cfg["master"], worker),
            )
            ok = await dispatch_worker_prompt(worker, wp.raw, payload.client_id)
            return wid, ok

    dispatched = await asyncio.gather(*(prep_and_dispatch(w) for w in online))
    failed = [wid for wid, ok in dispatched if not ok]
    if failed:
        log(f"orchestration: dispatch failed for {failed}")
    ok_ids = [wid for wid, ok in dispatched if ok]

    # Build the master prompt AFTER dispatch so its collectors only expect
    # workers that actually received the job — a mid-dispatch failure must
    # not cost the collector its full straggler timeout.
    if delegate and not ok_ids:
        trace_debug(trace, "delegate-only but every dispatch failed — "
                           "master runs the original prompt")
        pid = await enqueue_local(payload.prompt, payload.client_id)
        return {"status": "queued", "participants": ["master"], "job_ids": {},
                "master_prompt_id": pid}
    master_graph = graph
    if delegate:
        master_graph = transform.prepare_delegate_master_prompt(
            graph, graph.nodes_of_class(NODE_CLASS_COLLECTOR)
        )
    master_prompt = transform.apply_participant_overrides(
        master_graph, is_master=True, participant_id="master",
        enabled_worker_ids=ok_ids, job_id_map=job_id_map,
    )
    pid = await enqueue_local(master_prompt.raw, payload.client_id)
    participants = (["master"] if not delegate else []) + ok_ids
    return {"status": "queued", "participants": participants,
            "job_ids": job_id_map, "master_prompt_id": pid}
