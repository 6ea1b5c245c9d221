"""The browser-logic test suite (tests/web/run_tests.js) runs under plain
node — the JS counterpart of the reference's vitest suite (web/tests/).
Covered areas: URL-building parity with server/network.py, the submission
interceptor decision (preflight fallback + dead-tunnel blocking), the
DistributedValue per-worker widget model, divider socket morphing."""

import shutil
import subprocess
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.skipif(shutil.which("node") is None, reason="node not installed")
def test_web_js_suite_passes():
    res = subprocess.run(
        ["node", str(REPO / "tests" / "web" / "run_tests.js")],
        capture_output=True, text=True, timeout=60,
    )
    assert res.returncode == 0, f"js tests failed:\n{res.stdout}\n{res.stderr}"
    assert "18/18 passed" in res.stdout


def test_js_url_parity_with_python():
    """Cross-language spot check: the JS buildWorkerUrl must agree with the
    python implementation on a shared case table."""
    if shutil.which("node") is None:
        pytest.skip("node not installed")
    import json

    from comfyui_distributed_amd.server.network import build_worker_url

    cases = [
        {"host": "10.0.0.2", "port": 8189},
        {"host": "", "port": 8189},
        {"host": "http://box/", "port": 9000},
        {"host": "1.2.3.4:9000", "port": 8189},
        {"host": "abc.trycloudflare.com", "port": 8189},
        {"host": "[::1]:9100", "port": 8189},
    ]
    script = (
        "const D=require(process.argv[1]);"
        "const cases=JSON.parse(process.argv[2]);"
        "console.log(JSON.stringify(cases.map(c=>D.buildWorkerUrl(c))));"
    )
    res = subprocess.run(
        ["node", "-e", script,
         str(REPO / "comfyui_distributed_amd/server/static/js/distributed.js"),
         json.dumps(cases)],
        capture_output=True, text=True, timeout=30,
    )
    assert res.returncode == 0, res.stderr
    js_urls = json.loads(res.stdout)
    py_urls = [build_worker_url(c) for c in cases]
    assert js_urls == py_urls
