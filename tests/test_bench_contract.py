"""The driver's bench contract: `python bench.py` must print ONE JSON line
with the exact keys BASELINE.json's harness expects. Guarded here so no
refactor silently breaks round-end measurement."""

import json
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(600)
def test_bench_json_contract_tiny():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--src-size", "64",
         "--scale", "2", "--tile", "32", "--steps", "1", "--warmup", "0",
         "--sampler-steps", "1", "--tile-batch", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=540,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    json_lines = [ln for ln in proc.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, proc.stdout
    data = json.loads(json_lines[0])
    assert REQUIRED_KEYS.issubset(data), REQUIRED_KEYS - set(data)
    assert data["value"] > 0 and data["ms_per_step"] > 0
    assert data["n_gpus"] == 1 and data["steps"] == 1
    assert data["higher_is_better"] is True
    assert data["scaling"] == "weak"
    assert data["unit"] == "tiles/s"
    assert data["data"] == "synthetic"
    cfg = data["config"]
    for key in ("model", "global_batch", "parallelism"):
        assert key in cfg, key


@pytest.mark.timeout(600)
def test_bench_generation_branch_contract():
    """The seed-parallel generation branch (gen-sdxl / gen-flux / wan-t2v
    presets) also honors the JSON contract."""
    proc = subprocess.run(
        [sys.executable, "bench.py", "--config", "gen-flux", "--model",
         "flux_tiny", "--steps", "1", "--warmup", "0",
         "--sampler-steps", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=540,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    json_lines = [ln for ln in proc.stdout.splitlines() if ln.startswith("{")]
    assert len(json_lines) == 1, proc.stdout
    data = json.loads(json_lines[0])
    assert REQUIRED_KEYS.issubset(data)
    assert data["unit"] == "images/s" and data["value"] > 0
