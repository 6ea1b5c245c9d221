import json

from comfyui_distributed_amd.graph import PromptGraph, transform


def make_gen_prompt():
    """checkpoint -> sampler -> vae-decode -> collector -> save."""
    return {
        "1": {"class_type": "CheckpointLoader", "inputs": {"name": "sd15"}},
        "2": {"class_type": "DistributedSeed", "inputs": {"seed": 7}},
        "3": {
            "class_type": "KSampler",
            "inputs": {"model": ["1", 0], "seed": ["2", 0], "steps": 20},
        },
        "4": {"class_type": "VAEDecode", "inputs": {"samples": ["3", 0], "vae": ["1", 2]}},
        "5": {
            "class_type": "DistributedCollector",
            "inputs": {"images": ["4", 0], "load_balance": False},
        },
        "6": {"class_type": "SaveImage", "inputs": {"images": ["5", 0]}},
    }


def make_usdu_prompt():
    return {
        "1": {"class_type": "CheckpointLoader", "inputs": {"name": "sd15"}},
        "2": {"class_type": "LoadImage", "inputs": {"image": "in.png"}},
        "3": {
            "class_type": "UltimateSDUpscaleDistributed",
            "inputs": {"upscaled_image": ["2", 0], "model": ["1", 0], "seed": 1},
        },
        "4": {
            "class_type": "DistributedCollector",
            "inputs": {"images": ["3", 0]},
        },
        "5": {"class_type": "SaveImage", "inputs": {"images": ["4", 0]}},
    }


def test_upstream_downstream_closures():
    g = PromptGraph(make_gen_prompt())
    up = g.upstream_closure(["5"])
    assert up == {"1", "2", "3", "4", "5"}
    down = g.downstream_closure(["5"])
    assert down == {"5", "6"}


def test_job_id_map_format():
    g = PromptGraph(make_gen_prompt())
    m = transform.generate_job_id_map(g)
    assert set(m.keys()) == {"5"}
    assert m["5"].startswith("exec_") and m["5"].endswith("_5")
    m2 = transform.generate_job_id_map(g, prefix="exec_1_abc")
    assert m2["5"] == "exec_1_abc_5"


def test_prune_removes_downstream_and_adds_sink():
    g = PromptGraph(make_gen_prompt())
    pruned = transform.prune_prompt_for_worker(g)
    assert not pruned.nodes_of_class("SaveImage")  # downstream pruned
    assert "5" in pruned.raw
    # a PreviewImage sink was appended and feeds from the collector
    previews = pruned.nodes_of_class("PreviewImage")
    assert len(previews) == 1
    assert pruned.inputs(previews[0])["images"] == ["5", 0]


def test_prune_keeps_existing_consumer_without_sink():
    prompt = make_gen_prompt()
    # make the collector feed the USDU node, both distributed
    prompt["7"] = {
        "class_type": "UltimateSDUpscaleDistributed",
        "inputs": {"upscaled_image": ["5", 0], "model": ["1", 0]},
    }
    pruned = transform.prune_prompt_for_worker(PromptGraph(prompt))
    # collector has a surviving consumer (the USDU node) -> no sink for it
    sinks = pruned.nodes_of_class("PreviewImage")
    assert all(pruned.inputs(s)["images"][0] != "5" for s in sinks)


def test_prune_non_distributed_graph_is_identity():
    prompt = {"1": {"class_type": "LoadImage", "inputs": {}}}
    pruned = transform.prune_prompt_for_worker(PromptGraph(prompt))
    assert pruned.raw == prompt


def test_delegate_master_prompt():
    g = PromptGraph(make_gen_prompt())
    out = transform.prepare_delegate_master_prompt(g, ["5"])
    # only collector + downstream + placeholder remain
    classes = {out.class_of(n) for n in out.node_ids()}
    assert "KSampler" not in classes and "VAEDecode" not in classes
    assert "SaveImage" in classes
    assert "DistributedEmptyImage" in classes
    col_inputs = out.inputs("5")
    assert col_inputs["delegate_only"] is True
    placeholder = out.nodes_of_class("DistributedEmptyImage")[0]
    assert col_inputs["images"] == [placeholder, 0]


def test_master_overrides():
    g = PromptGraph(make_gen_prompt())
    jm = transform.generate_job_id_map(g, prefix="exec_1_abc")
    out = transform.apply_participant_overrides(
        g, is_master=True, participant_id="master",
        enabled_worker_ids=["w0", "w1"], job_id_map=jm,
    )
    seed_inputs = out.inputs("2")
    assert seed_inputs["is_worker"] is False and seed_inputs["worker_id"] == ""
    col = out.inputs("5")
    assert col["multi_job_id"] == "exec_1_abc_5"
    assert json.loads(col["enabled_worker_ids"]) == ["w0", "w1"]
    assert "worker_id" not in col


def test_worker_overrides_positional_index():
    g = PromptGraph(make_gen_prompt())
    jm = transform.generate_job_id_map(g, prefix="exec_1_abc")
    out = transform.apply_participant_overrides(
        g, is_master=False, participant_id="w1",
        enabled_worker_ids=["w0", "w1"], job_id_map=jm,
        master_url="http://127.0.0.1:8188",
    )
    # Seed node gets the positional id (index 1 -> "worker_1")
    assert out.inputs("2")["worker_id"] == "worker_1"
    assert out.inputs("2")["is_worker"] is True
    col = out.inputs("5")
    assert col["worker_id"] == "w1"
    assert col["master_url"] == "http://127.0.0.1:8188"
    assert col["is_worker"] is True


def test_collector_downstream_of_usdu_gets_pass_through():
    g = PromptGraph(make_usdu_prompt())
    jm = transform.generate_job_id_map(g, prefix="exec_1_abc")
    out = transform.apply_participant_overrides(
        g, is_master=True, participant_id="master",
        enabled_worker_ids=["w0"], job_id_map=jm,
    )
    assert out.inputs("4")["pass_through"] is True
    assert "pass_through" not in out.inputs("3")
