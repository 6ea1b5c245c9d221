import pytest
import torch

from comfyui_distributed_amd.graph.executor import Executor, default_registry, validate_prompt
from comfyui_distributed_amd.utils.errors import PromptValidationError


def txt2img_prompt(model="tiny", w=16, h=16):
    return {
        "1": {"class_type": "CheckpointLoader", "inputs": {"ckpt_name": model}},
        "2": {"class_type": "CLIPTextEncode", "inputs": {"text": "a cat", "clip": ["1", 1]}},
        "3": {"class_type": "CLIPTextEncode", "inputs": {"text": "", "clip": ["1", 1]}},
        "4": {"class_type": "EmptyLatentImage",
              "inputs": {"width": w, "height": h, "batch_size": 1}},
        "5": {"class_type": "KSampler", "inputs": {
            "model": ["1", 0], "seed": 3, "steps": 2, "cfg": 2.0,
            "sampler_name": "euler", "scheduler": "karras",
            "positive": ["2", 0], "negative": ["3", 0],
            "latent_image": ["4", 0], "denoise": 1.0}},
        "6": {"class_type": "VAEDecode", "inputs": {"samples": ["5", 0], "vae": ["1", 2]}},
        "7": {"class_type": "PreviewImage", "inputs": {"images": ["6", 0]}},
    }


def test_validate_rejects_unknown_class():
    reg = default_registry()
    with pytest.raises(PromptValidationError) as ei:
        validate_prompt({"1": {"class_type": "NoSuchNode", "inputs": {}}}, reg)
    assert "1" in ei.value.node_errors


def test_validate_rejects_missing_link_and_cycle():
    reg = default_registry()
    with pytest.raises(PromptValidationError):
        validate_prompt(
            {"1": {"class_type": "PreviewImage", "inputs": {"images": ["9", 0]}}},
            reg,
        )
    with pytest.raises(PromptValidationError):
        validate_prompt(
            {
                "1": {"class_type": "VAEDecode", "inputs": {"samples": ["2", 0], "vae": ["2", 0]}},
                "2": {"class_type": "VAEDecode", "inputs": {"samples": ["1", 0], "vae": ["1", 0]}},
            },
            reg,
        )


def test_txt2img_graph_executes():
    previews = []
    ex = Executor(context={"preview_images": previews, "device": "cpu"})
    cache = ex.execute(txt2img_prompt())
    assert previews and previews[0].shape == (1, 16, 16, 3)
    assert torch.isfinite(previews[0]).all()
    # KSampler output cached
    assert "5" in cache


def test_distributed_seed_node_in_graph():
    prompt = {
        "1": {"class_type": "DistributedSeed",
              "inputs": {"seed": 100, "is_worker": True, "worker_id": "worker_2"}},
    }
    ex = Executor()
    cache = ex.execute(prompt)
    assert cache["1"] == (103,)  # 100 + index 2 + 1


def test_batch_divider_in_graph():
    import comfyui_distributed_amd.graph.builtin_nodes  # noqa: F401

    prompt = {
        "1": {"class_type": "LoadImage", "inputs": {"image": "synthetic:8x8"}},
        "2": {"class_type": "ImageBatchDivider",
              "inputs": {"images": ["1", 0], "divide_by": 2}},
    }
    ex = Executor()
    cache = ex.execute(prompt)
    outs = cache["2"]
    assert len(outs) == 10
    assert outs[0].shape[0] == 1 and outs[1].shape[0] == 0


def test_cross_run_output_caching():
    """ComfyUI-style execution caching: a re-submitted workflow reuses
    unchanged nodes' outputs; changing a widget re-executes the node and
    its downstream; output nodes always re-run."""
    from comfyui_distributed_amd.graph.executor import Executor, NodeRegistry

    calls = {"a": 0, "sink": 0}

    class Producer:
        RETURN_TYPES = ("INT",)
        FUNCTION = "run"

        def run(self, value):
            calls["a"] += 1
            return (value * 2,)

    class Sink:
        RETURN_TYPES = ()
        OUTPUT_NODE = True
        FUNCTION = "run"

        def run(self, x):
            calls["sink"] += 1
            return ()

    reg = NodeRegistry()
    reg.register("Producer", Producer)
    reg.register("Sink", Sink)
    ex = Executor(registry=reg)
    prompt = {"1": {"class_type": "Producer", "inputs": {"value": 3}},
              "2": {"class_type": "Sink", "inputs": {"x": ["1", 0]}}}
    out1 = ex.execute(prompt)
    assert out1["1"] == (6,) and calls == {"a": 1, "sink": 1}
    out2 = ex.execute(prompt)  # producer cached, sink re-runs
    assert out2["1"] == (6,) and calls == {"a": 1, "sink": 2}
    prompt["1"]["inputs"]["value"] = 5  # widget change invalidates
    out3 = ex.execute(prompt)
    assert out3["1"] == (10,) and calls["a"] == 2
    # fingerprint is positional spec, not node id: renumbering still hits
    renamed = {"9": {"class_type": "Producer", "inputs": {"value": 5}},
               "8": {"class_type": "Sink", "inputs": {"x": ["9", 0]}}}
    ex.execute(renamed)
    assert calls["a"] == 2


def test_validation_rejects_missing_required_input():
    from comfyui_distributed_amd.graph.executor import (
        default_registry, validate_prompt)
    from comfyui_distributed_amd.utils.errors import PromptValidationError

    # KSampler without its model input: caught at validation, not at
    # execution (ComfyUI-parity structured node_errors)
    bad = {"1": {"class_type": "KSampler", "inputs": {"seed": 1}}}
    with pytest.raises(PromptValidationError) as exc:
        validate_prompt(bad, default_registry())
    msgs = exc.value.node_errors["1"]
    assert any("model" in m for m in msgs)
    # defaults still allowed to be absent
    ok = {"1": {"class_type": "DistributedSeed", "inputs": {}}}
    validate_prompt(ok, default_registry())


def test_validation_rejects_out_of_range_output_index():
    from comfyui_distributed_amd.graph.executor import (
        default_registry, validate_prompt)
    from comfyui_distributed_amd.utils.errors import PromptValidationError

    bad = {
        "1": {"class_type": "DistributedSeed", "inputs": {"seed": 1}},
        "2": {"class_type": "PreviewImage", "inputs": {"images": ["1", 5]}},
    }
    with pytest.raises(PromptValidationError) as exc:
        validate_prompt(bad, default_registry())
    assert any("output 5" in m for m in exc.value.node_errors["2"])


def test_is_changed_nodes_never_cached_across_runs():
    from comfyui_distributed_amd.graph.executor import Executor, NodeRegistry

    calls = {"n": 0}

    class Volatile:
        RETURN_TYPES = ("INT",)
        FUNCTION = "run"

        @classmethod
        def IS_CHANGED(cls, **kw):
            return float("nan")

        def run(self):
            calls["n"] += 1
            return (calls["n"],)

    class Sink:
        RETURN_TYPES = ()
        OUTPUT_NODE = True
        FUNCTION = "run"

        def run(self, x):
            return ()

    reg = NodeRegistry()
    reg.register("Volatile", Volatile)
    reg.register("Sink", Sink)
    ex = Executor(registry=reg)
    prompt = {"1": {"class_type": "Volatile", "inputs": {}},
              "2": {"class_type": "Sink", "inputs": {"x": ["1", 0]}}}
    ex.execute(prompt)
    ex.execute(prompt)
    assert calls["n"] == 2  # re-executed despite identical fingerprint


def test_auto_populate_creates_workers_for_fake_gpus(tmp_config, monkeypatch):
    import asyncio

    import torch
    from aiohttp.test_utils import TestClient, TestServer

    from comfyui_distributed_amd.server.app import DistributedServer

    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)
    monkeypatch.setattr(torch.cuda, "device_count", lambda: 4)

    async def go():
        srv = DistributedServer()
        cl = TestClient(TestServer(srv.build_app()))
        await cl.start_server()
        try:
            r = await cl.post("/distributed/auto_populate_workers", json={})
            body = await r.json()
            assert body["gpu_count"] == 4
            # master sits on GPU 0 -> workers for GPUs 1..3
            assert body["created"] == ["auto_gpu1", "auto_gpu2", "auto_gpu3"]
            cfg = await (await cl.get("/distributed/config")).json()
            devs = sorted(w["cuda_device"] for w in cfg["workers"])
            assert devs == [1, 2, 3]
            ports = sorted(w["port"] for w in cfg["workers"])
            assert ports == [cfg["master"]["port"] + 2,
                             cfg["master"]["port"] + 3,
                             cfg["master"]["port"] + 4]
        finally:
            await cl.close()

    asyncio.run(go())


def test_load_image_rejects_path_escape(tmp_path):
    """LoadImage must not read files outside the input dir."""
    from comfyui_distributed_amd.graph.builtin_nodes import LoadImage

    secret = tmp_path / "secret.png"
    secret.write_bytes(b"x")
    input_dir = tmp_path / "input"
    input_dir.mkdir()
    node = LoadImage()
    node.set_context({"input_dir": str(input_dir)})
    for name in ("../secret.png", str(secret), "a/../../secret.png"):
        with pytest.raises(ValueError, match="escapes"):
            node.load(image=name)


def test_load_image_cache_busts_on_file_change(tmp_path):
    """a re-uploaded file under the same name re-executes the
    load (IS_CHANGED content mark in the fingerprint)."""
    import os
    import time

    import torch

    from comfyui_distributed_amd.graph.executor import Executor
    from comfyui_distributed_amd.utils.image import encode_png_bytes

    input_dir = tmp_path / "input"
    input_dir.mkdir()
    img1 = torch.zeros(1, 4, 4, 3)
    img2 = torch.ones(1, 4, 4, 3)
    p = input_dir / "x.png"
    p.write_bytes(encode_png_bytes(img1))
    ex = Executor(context={"input_dir": str(input_dir),
                           "output_dir": str(tmp_path / "out")})
    prompt = {"1": {"class_type": "LoadImage", "inputs": {"image": "x.png"}},
              "2": {"class_type": "PreviewImage", "inputs": {"images": ["1", 0]}}}
    out1 = ex.execute(prompt)["1"][0]
    assert float(out1.sum()) == 0.0
    # unchanged file: cached (same tensor object back)
    assert ex.execute(prompt)["1"][0] is out1
    # replace content under the same name -> must re-read
    p.write_bytes(encode_png_bytes(img2))
    st = p.stat()
    os.utime(p, ns=(st.st_atime_ns, st.st_mtime_ns + 1_000_000))
    out2 = ex.execute(prompt)["1"][0]
    assert float(out2.mean()) > 0.9
