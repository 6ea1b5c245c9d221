"""Orchestration logic tests (probes + dispatch mocked; reference tests its
selection logic the same way, tests/test_dispatch_selection.py)."""

import asyncio
import json

import pytest

from comfyui_distributed_amd.server import orchestration
from comfyui_distributed_amd.server.job_state import JobState
from comfyui_distributed_amd.server.queue_request import (
    QueueRequestError,
    QueueRequestPayload,
    parse_queue_request_payload,
)


def dist_prompt():
    return {
        "1": {"class_type": "DistributedSeed", "inputs": {"seed": 7}},
        "2": {"class_type": "KSampler", "inputs": {"seed": ["1", 0]}},
        "3": {"class_type": "DistributedCollector", "inputs": {"images": ["2", 0]}},
    }


@pytest.fixture()
def two_worker_config(tmp_config):
    from comfyui_distributed_amd.utils.config import load_config, save_config

    cfg = load_config()
    cfg["workers"] = [
        {"id": "w1", "name": "gpu1", "host": "", "port": 8189, "cuda_device": 1,
         "enabled": True, "type": "local"},
        {"id": "w2", "name": "gpu2", "host": "", "port": 8190, "cuda_device": 2,
         "enabled": True, "type": "local"},
    ]
    save_config(cfg)
    return cfg


def run_orch(payload, monkeypatch, probes, dispatch_ok=True):
    """Run orchestration with fake probe/dispatch; returns (result, calls)."""
    calls = {"dispatched": [], "local": []}

    async def fake_probe(url, timeout=5.0):
        # url contains the port
        for wid, info in probes.items():
            if wid in url:
                return info
        return None

    async def fake_probe_by_cfg(workers, concurrency=None):
        return {str(w["id"]): probes.get(str(w["id"])) for w in workers}

    async def fake_dispatch(worker, prompt, client_id, timeout=30.0,
                            use_websocket=None):
        calls["dispatched"].append((str(worker["id"]), prompt))
        if isinstance(dispatch_ok, bool):
            return dispatch_ok
        return str(worker["id"]) in dispatch_ok

    async def enqueue_local(prompt, client_id):
        calls["local"].append(prompt)

    monkeypatch.setattr(orchestration, "probe_workers", fake_probe_by_cfg)
    monkeypatch.setattr(orchestration, "dispatch_worker_prompt", fake_dispatch)
    state = JobState()
    result = asyncio.run(
        orchestration.orchestrate_distributed_execution(payload, state, enqueue_local)
    )
    return result, calls, state


def test_parse_payload_schema():
    with pytest.raises(QueueRequestError):
        parse_queue_request_payload({})
    with pytest.raises(QueueRequestError):
        parse_queue_request_payload({"prompt": dist_prompt()})  # no workers
    p = parse_queue_request_payload({
        "workflow": {"prompt": dist_prompt()},
        "workers": ["w1"], "client_id": "c",
    })
    assert p.enabled_worker_ids == ["w1"]


def test_fanout_dispatches_to_online_workers(two_worker_config, monkeypatch):
    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 0}},
              "w2": {"exec_info": {"queue_remaining": 0}}}
    result, calls, state = run_orch(payload, monkeypatch, probes)
    assert sorted(w for w, _ in calls["dispatched"]) == ["w1", "w2"]
    assert len(calls["local"]) == 1  # master's own prompt
    # collector queues pre-created
    assert len(state.pending_jobs) == 1
    # worker prompts have overrides
    wprompt = calls["dispatched"][0][1]
    col = next(n for n in wprompt.values()
               if n["class_type"] == "DistributedCollector")
    assert col["inputs"]["is_worker"] is True
    assert json.loads(col["inputs"]["enabled_worker_ids"]) == ["w1", "w2"]


def test_offline_workers_skipped(two_worker_config, monkeypatch):
    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 0}}}  # w2 offline
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    assert [w for w, _ in calls["dispatched"]] == ["w1"]
    assert "master" in result["participants"] and "w1" in result["participants"]


def test_all_offline_falls_back_to_master(two_worker_config, monkeypatch):
    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    result, calls, _ = run_orch(payload, monkeypatch, probes={})
    assert calls["dispatched"] == []
    assert result["participants"] == ["master"]


def test_delegate_master_prompt_pruned(two_worker_config, monkeypatch):
    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["w1"],
                                  delegate_master=True)
    probes = {"w1": {"exec_info": {"queue_remaining": 0}}}
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    master_prompt = calls["local"][0]
    classes = {n["class_type"] for n in master_prompt.values()}
    assert "KSampler" not in classes  # upstream stripped
    assert "DistributedCollector" in classes
    assert "master" not in result["participants"]


def test_load_balance_picks_least_busy(two_worker_config, monkeypatch):
    prompt = dist_prompt()
    prompt["3"]["inputs"]["load_balance"] = True
    payload = QueueRequestPayload(prompt=prompt, client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 5}},
              "w2": {"exec_info": {"queue_remaining": 5}}}
    # master idle (0) -> master wins
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    assert result["participants"] == ["master"]
    assert calls["dispatched"] == []


def test_load_balance_routes_to_idle_worker(two_worker_config, monkeypatch, tmp_config):
    from comfyui_distributed_amd.utils.config import load_config, save_config

    cfg = load_config()
    cfg["settings"]["master_delegate_only"] = True
    save_config(cfg)
    prompt = dist_prompt()
    prompt["3"]["inputs"]["load_balance"] = True
    payload = QueueRequestPayload(prompt=prompt, client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 3}},
              "w2": {"exec_info": {"queue_remaining": 0}}}
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    assert result["participants"] == ["w2"]
    assert [w for w, _ in calls["dispatched"]] == ["w2"]


def test_select_least_busy_round_robin():
    cands = [("a", {"exec_info": {"queue_remaining": 0}}),
             ("b", {"exec_info": {"queue_remaining": 0}})]
    picks = {orchestration.select_least_busy(cands) for _ in range(10)}
    assert picks == {"a", "b"}  # round-robin among idle


def test_auto_prepare_field_validation():
    with pytest.raises(QueueRequestError):
        parse_queue_request_payload({
            "prompt": dist_prompt(), "client_id": "c",
            "enabled_worker_ids": [], "auto_prepare": "yes",
        })
    p = parse_queue_request_payload({
        "prompt": dist_prompt(), "client_id": "c",
        "enabled_worker_ids": [], "auto_prepare": False,
    })
    assert p.auto_prepare is False


def test_delegate_with_usdu_falls_back_to_master(two_worker_config, monkeypatch):
    prompt = dist_prompt()
    prompt["9"] = {"class_type": "UltimateSDUpscaleDistributed",
                   "inputs": {"upscaled_image": ["2", 0], "seed": 1}}
    payload = QueueRequestPayload(prompt=prompt, client_id="c",
                                  enabled_worker_ids=["w1"],
                                  delegate_master=True)
    probes = {"w1": {"exec_info": {"queue_remaining": 0}}}
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    # master participates despite delegate flag (reference limitation kept)
    assert "master" in result["participants"]
    assert "master_prompt_id" in result


def test_failed_dispatch_excluded_from_master_collector(two_worker_config,
                                                        monkeypatch):
    """If a worker accepts the probe but the dispatch fails, the master's
    collector must not wait for it: the master prompt is built after
    dispatch with only the successful ids."""
    import json

    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["w1", "w2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 0}},
              "w2": {"exec_info": {"queue_remaining": 0}}}
    result, calls, _ = run_orch(payload, monkeypatch, probes,
                                dispatch_ok={"w1"})
    assert result["participants"] == ["master", "w1"]
    assert len(calls["local"]) == 1
    master_prompt = calls["local"][0]
    collector = next(v for v in master_prompt.values()
                     if v["class_type"] == "DistributedCollector")
    assert json.loads(collector["inputs"]["enabled_worker_ids"]) == ["w1"]


def test_probe_concurrency_is_bounded(monkeypatch):
    """probe_workers must respect the concurrency semaphore (reference
    tests/test_dispatch_selection.py:167)."""
    import asyncio

    peak = {"now": 0, "max": 0}

    async def slow_probe(url, timeout=5.0):
        peak["now"] += 1
        peak["max"] = max(peak["max"], peak["now"])
        await asyncio.sleep(0.02)
        peak["now"] -= 1
        return {"exec_info": {"queue_remaining": 0}}

    monkeypatch.setattr(orchestration.network, "probe_worker", slow_probe)
    workers = [{"id": f"w{i}", "host": "", "port": 8000 + i}
               for i in range(12)]
    res = asyncio.run(orchestration.probe_workers(workers, concurrency=3))
    assert len(res) == 12 and all(v is not None for v in res.values())
    assert peak["max"] <= 3


def test_unknown_worker_ids_fall_back_to_master(two_worker_config, monkeypatch):
    payload = QueueRequestPayload(prompt=dist_prompt(), client_id="c",
                                  enabled_worker_ids=["ghost1", "ghost2"])
    probes = {"w1": {"exec_info": {"queue_remaining": 0}}}
    result, calls, _ = run_orch(payload, monkeypatch, probes)
    assert result["participants"] == ["master"]
    assert calls["dispatched"] == []
    assert len(calls["local"]) == 1
