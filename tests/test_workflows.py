"""The shipped example workflows must validate and (downsized) execute."""

import json
from pathlib import Path

import pytest

from comfyui_distributed_amd.graph.executor import Executor, default_registry, validate_prompt

WF_DIR = Path(__file__).resolve().parent.parent / "workflows"


def load_wf(name):
    data = json.loads((WF_DIR / name).read_text())
    data.pop("_comment", None)
    return data


@pytest.mark.parametrize("name", [
    "seed_parallel_txt2img.json",
    "distributed_upscale.json",
    "distributed_wan_video.json",
    "distributed_upscale_video.json",
    "distributed_flux_txt2img.json",
    "parameter_sweep.json",
    "distributed_audio_collect.json",
    "inpaint.json",
    "hires_fix.json",
    "wan_image_to_video.json",
])
def test_workflows_validate(name):
    validate_prompt(load_wf(name), default_registry())


def test_txt2img_workflow_executes_downsized(tmp_path):
    wf = load_wf("seed_parallel_txt2img.json")
    wf["1"]["inputs"]["ckpt_name"] = "tiny"
    wf["5"]["inputs"].update(width=16, height=16)
    wf["6"]["inputs"].update(steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 1
    assert (tmp_path / Path(saved[0]).name).exists()


def test_wan_workflow_executes_downsized(tmp_path):
    wf = load_wf("distributed_wan_video.json")
    wf["1"]["inputs"]["ckpt_name"] = "wan_tiny"
    wf["4"]["inputs"].update(width=16, height=16, frames=5, steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    # 5 frames split 3/2 across the two dividers' save nodes, plus the
    # animated webp of the full clip
    assert len(saved) == 6
    assert sum(1 for p in saved if p.endswith(".webp")) == 1


def test_hires_fix_workflow_executes_downsized(tmp_path):
    wf = load_wf("hires_fix.json")
    wf["1"]["inputs"]["ckpt_name"] = "tiny"
    wf["4"]["inputs"].update(width=16, height=16)
    wf["5"]["inputs"].update(steps=1)
    wf["6"]["inputs"].update(width=32, height=32)
    wf["7"]["inputs"].update(steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 1
    import torch as _t

    from comfyui_distributed_amd.utils.image import decode_png_bytes

    img = decode_png_bytes(Path(saved[0]).read_bytes())
    assert img.shape == (1, 32, 32, 3)  # the upscaled size


def test_inpaint_workflow_executes_downsized(tmp_path):
    wf = load_wf("inpaint.json")
    wf["1"]["inputs"]["ckpt_name"] = "tiny"
    wf["4"]["inputs"]["image"] = "synthetic:32x32"
    wf["8"]["inputs"].update(steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 1


def test_flux_workflow_executes_downsized(tmp_path):
    wf = load_wf("distributed_flux_txt2img.json")
    wf["1"]["inputs"]["ckpt_name"] = "flux_tiny"
    wf["4"]["inputs"].update(width=16, height=16, steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 1


def test_audio_workflow_executes_downsized(tmp_path):
    wf = load_wf("distributed_audio_collect.json")
    wf["2"]["inputs"].update(seconds=0.01, sample_rate=8000)
    saved = []
    previews = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "preview_images": previews, "device": "cpu"})
    ex.execute(wf)
    assert any(p.endswith(".wav") for p in saved)
    assert previews


def test_i2v_workflow_executes_downsized(tmp_path):
    wf = load_wf("wan_image_to_video.json")
    wf["1"]["inputs"]["ckpt_name"] = "wan_tiny"
    wf["3"]["inputs"]["image"] = "synthetic:16x16"
    wf["4"]["inputs"].update(width=16, height=16, frames=5, steps=1)
    saved = []
    previews = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "preview_images": previews, "device": "cpu"})
    ex.execute(wf)
    assert any(p.endswith(".webp") for p in saved)


def test_parameter_sweep_workflow_executes_downsized(tmp_path):
    wf = load_wf("parameter_sweep.json")
    wf["1"]["inputs"]["ckpt_name"] = "tiny"
    wf["5"]["inputs"].update(width=16, height=16)
    wf["6"]["inputs"].update(steps=1)
    saved = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 1


def test_upscale_video_workflow_executes_downsized(tmp_path):
    wf = load_wf("distributed_upscale_video.json")
    wf["1"]["inputs"]["ckpt_name"] = "wan_tiny"
    wf["3"]["inputs"].update(width=16, height=16, frames=5, steps=1)
    wf["4"]["inputs"]["ckpt_name"] = "tiny"
    wf["7"]["inputs"].update(steps=1, tile_width=16, tile_height=16,
                             padding=16, mask_blur=2)
    saved = []
    previews = []
    ex = Executor(context={"output_dir": str(tmp_path), "saved_images": saved,
                           "preview_images": previews, "device": "cpu"})
    ex.execute(wf)
    assert len(saved) == 5  # five upscaled frames
