"""REST API tests (aiohttp test client, no sockets beyond loopback)."""

import asyncio
import json

import pytest
import torch
from aiohttp.test_utils import TestClient, TestServer

from comfyui_distributed_amd.server.app import DistributedServer
from comfyui_distributed_amd.utils.image import encode_png_base64


@pytest.fixture()
def client(tmp_config, event_loop=None):
    async def make():
        srv = DistributedServer()
        app = srv.build_app()
        client = TestClient(TestServer(app))
        await client.start_server()
        return srv, client

    loop = asyncio.new_event_loop()
    srv, cl = loop.run_until_complete(make())
    yield srv, cl, loop
    loop.run_until_complete(cl.close())
    loop.close()


def run(loop, coro):
    return loop.run_until_complete(coro)


def test_get_prompt_probe(client):
    srv, cl, loop = client

    async def go():
        r = await cl.get("/prompt")
        assert r.status == 200
        data = await r.json()
        assert "exec_info" in data and "queue_remaining" in data["exec_info"]

    run(loop, go())


def test_post_prompt_validation(client):
    srv, cl, loop = client

    async def go():
        r = await cl.post("/prompt", json={"prompt": {"1": {"class_type": "Nope", "inputs": {}}}})
        assert r.status == 400
        data = await r.json()
        assert "node_errors" in data
        r = await cl.post("/prompt", json={"prompt": {
            "1": {"class_type": "DistributedSeed", "inputs": {"seed": 5}}}})
        assert r.status == 200
        assert "prompt_id" in await r.json()

    run(loop, go())


def test_queue_requires_fields(client):
    srv, cl, loop = client

    async def go():
        r = await cl.post("/distributed/queue", json={})
        assert r.status == 400
        r = await cl.post("/distributed/queue", json={
            "prompt": {"1": {"class_type": "DistributedSeed", "inputs": {"seed": 1}}},
            "client_id": "c1",
        })
        assert r.status == 400  # missing enabled_worker_ids

    run(loop, go())


def test_queue_local_fallback_no_workers(client):
    srv, cl, loop = client

    async def go():
        r = await cl.post("/distributed/queue", json={
            "prompt": {"1": {"class_type": "DistributedSeed", "inputs": {"seed": 1}}},
            "client_id": "c1",
            "enabled_worker_ids": [],
        })
        assert r.status == 200
        data = await r.json()
        assert data["participants"] == ["master"]

    run(loop, go())


def test_job_complete_roundtrip(client):
    srv, cl, loop = client

    async def go():
        await srv.job_state.ensure_queue("jobX")
        img = torch.rand(1, 8, 8, 3)
        r = await cl.post("/distributed/job_complete", json={
            "job_id": "jobX", "worker_id": "w1", "batch_idx": 0,
            "image": encode_png_base64(img), "is_last": True,
        })
        assert r.status == 200
        q = srv.job_state.pending_jobs["jobX"]
        item = q.get_nowait()
        assert item["worker_id"] == "w1" and item["is_last"] is True
        assert item["tensor"].shape == (1, 8, 8, 3)
        # invalid envelope
        r = await cl.post("/distributed/job_complete", json={"job_id": "jobX"})
        assert r.status == 400

    run(loop, go())


def test_tile_job_endpoints(client):
    srv, cl, loop = client

    async def go():
        await srv.job_state.init_static_job("tj", n_tiles=2, batch_size=1)
        # worker polls ready
        r = await cl.get("/distributed/job_status/tj")
        assert (await r.json())["ready"] is True
        r = await cl.get("/distributed/job_status/nope")
        assert (await r.json())["ready"] is False
        # pull both tiles then exhaustion
        got = []
        for _ in range(2):
            r = await cl.post("/distributed/request_image",
                              json={"job_id": "tj", "worker_id": "w1"})
            got.append((await r.json())["tile_idx"])
        assert sorted(got) == [0, 1]
        r = await cl.post("/distributed/request_image",
                          json={"job_id": "tj", "worker_id": "w1"})
        assert (await r.json())["tile_idx"] is None
        # heartbeat + submit
        r = await cl.post("/distributed/heartbeat",
                          json={"job_id": "tj", "worker_id": "w1"})
        assert r.status == 200
        tile_png = encode_png_base64(torch.rand(1, 8, 8, 3))
        r = await cl.post("/distributed/submit_tiles", json={
            "job_id": "tj", "worker_id": "w1", "is_last": True,
            "tiles": [{"tile_idx": 0, "batch_idx": 0, "image": tile_png}],
        })
        assert (await r.json())["received"] == 1
        job = await srv.job_state.get_tile_job("tj")
        assert "w1" in job.finished_workers
        item = job.results.get_nowait()
        assert item["tile_idx"] == 0 and item["tensor"].shape == (1, 8, 8, 3)

    run(loop, go())


def test_config_crud(client):
    srv, cl, loop = client

    async def go():
        r = await cl.post("/distributed/config/update_worker", json={
            "id": "w9", "name": "gpu9", "port": 8199, "cuda_device": 1,
            "enabled": True, "type": "local",
        })
        assert r.status == 200
        r = await cl.get("/distributed/config")
        cfg = await r.json()
        assert any(w["id"] == "w9" for w in cfg["workers"])
        r = await cl.post("/distributed/config/update_setting",
                          json={"key": "debug", "value": True})
        assert r.status == 200
        r = await cl.post("/distributed/config/update_setting",
                          json={"key": "evil", "value": 1})
        assert r.status == 400
        r = await cl.post("/distributed/config/delete_worker", json={"id": "w9"})
        assert r.status == 200
        r = await cl.post("/distributed/config/delete_worker", json={"id": "w9"})
        assert r.status == 404

    run(loop, go())


def test_info_endpoints(client):
    srv, cl, loop = client

    async def go():
        r = await cl.get("/distributed/network_info")
        data = await r.json()
        assert "hostname" in data and "cuda_device_count" in data
        r = await cl.get("/distributed/system_info")
        data = await r.json()
        assert data["platform"] and "machine_id" in data
        r = await cl.post("/distributed/check_file", json={"path": "/nonexistent"})
        assert (await r.json())["exists"] is False

    run(loop, go())


def test_prompt_execution_end_to_end(client):
    """POST /prompt with a DistributedSeed graph actually executes."""
    srv, cl, loop = client

    async def go():
        r = await cl.post("/prompt", json={"prompt": {
            "1": {"class_type": "DistributedSeed",
                  "inputs": {"seed": 5, "is_worker": True, "worker_id": "worker_0"}},
        }})
        assert r.status == 200
        # wait for the execution loop to drain
        for _ in range(50):
            if srv.prompt_queue.qsize() == 0 and not srv.executing:
                break
            await asyncio.sleep(0.05)
        assert srv.prompt_queue.qsize() == 0

    run(loop, go())


def test_object_info_and_clear_memory(client):
    srv, cl, loop = client

    async def go():
        r = await cl.get("/object_info")
        info = await r.json()
        assert "DistributedCollector" in info
        assert "UltimateSDUpscaleDistributed" in info
        assert info["DistributedSeed"]["output"] == ["INT"]
        assert "hidden" in info["DistributedCollector"]["input"]
        r = await cl.post("/distributed/clear_memory", json={})
        data = await r.json()
        assert data["status"] == "ok" and "unloaded_models" in data

    run(loop, go())


def test_worker_status_and_auto_populate(client):
    srv, cl, loop = client

    async def go():
        # unknown worker -> 404
        r = await cl.get("/distributed/worker_status?id=ghost")
        assert r.status == 404
        # create an offline worker
        r = await cl.post("/distributed/config/update_worker", json={
            "id": "ws1", "name": "ws1", "port": 1, "host": "127.0.0.1",
            "type": "remote", "enabled": True})
        assert r.status == 200
        r = await cl.get("/distributed/worker_status?id=ws1")
        assert r.status == 200
        st = await r.json()
        assert st["online"] is False and st["managed"] is False
        # a worker pointing at this very server probes online
        r = await cl.post("/distributed/config/update_worker", json={
            "id": "ws2", "name": "self", "port": cl.server.port,
            "host": "127.0.0.1", "type": "remote", "enabled": True})
        assert r.status == 200
        r = await cl.get("/distributed/worker_status?id=ws2")
        st = await r.json()
        assert st["online"] is True and st["queue_remaining"] == 0

        # auto-populate: 0 GPUs on CPU box -> nothing created, flag set
        r = await cl.post("/distributed/auto_populate_workers", json={})
        body = await r.json()
        assert body["status"] == "ok" and body["created"] == []
        r = await cl.post("/distributed/auto_populate_workers", json={})
        body = await r.json()
        assert body["status"] == "already_populated"
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()

    run(loop, go())


def test_local_worker_status_and_clear_launching(client):
    srv, cl, loop = client

    async def go():
        # disabled local + self-pointing enabled local + remote (excluded)
        for body in (
            {"id": "la", "name": "a", "port": 1, "host": "", "enabled": False},
            {"id": "lb", "name": "b", "port": cl.server.port,
             "host": "127.0.0.1", "enabled": True},
            {"id": "rc", "name": "c", "port": 9, "host": "10.9.9.9",
             "type": "remote", "enabled": True},
        ):
            r = await cl.post("/distributed/config/update_worker", json=body)
            assert r.status == 200
        r = await cl.get("/distributed/local-worker-status")
        assert r.status == 200
        st = (await r.json())["worker_statuses"]
        assert set(st) == {"la", "lb"}  # remote excluded
        assert st["la"] == {"online": False, "enabled": False,
                            "processing": False, "queue_count": 0}
        assert st["lb"]["online"] is True and st["lb"]["queue_count"] == 0

        # remote log proxy rejects local workers; unknown -> 404
        r = await cl.get("/distributed/remote_worker_log/la")
        assert r.status == 400
        r = await cl.get("/distributed/remote_worker_log/ghost")
        assert r.status == 404
        r = await cl.get("/distributed/remote_worker_log/rc")
        assert r.status == 502  # unreachable remote

        # clear_launching drops the marker set by launch_worker persistence
        from comfyui_distributed_amd.utils.config import (
            config_transaction, load_config)

        async with config_transaction() as cfg:
            cfg.setdefault("managed_processes", {})["lb"] = {
                "pid": 1, "launching": True}
        r = await cl.post("/distributed/worker/clear_launching",
                          json={"worker_id": "lb"})
        assert r.status == 200
        assert "launching" not in load_config()["managed_processes"]["lb"]
        r = await cl.post("/distributed/worker/clear_launching",
                          json={"worker_id": "ghost"})
        assert r.status == 404
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()

    run(loop, go())


def test_panel_served_with_expected_controls(client):
    srv, cl, loop = client

    async def go():
        for path in ("/", "/panel"):
            r = await cl.get(path)
            assert r.status == 200
            html = await r.text()
            for needle in ("worker_status", "auto_populate_workers",
                           "update_worker", "delete_worker", "update_setting",
                           "tunnel", "local_log", "distributed/queue",
                           "/history", "/view?filename="):
                assert needle in html, f"panel missing wiring for {needle}"

    run(loop, go())


def test_interrupt_fanout_and_view(client, tmp_path):
    srv, cl, loop = client

    async def go():
        # a worker pointing at this server: fanout interrupt reaches it
        r = await cl.post("/distributed/config/update_worker", json={
            "id": "fo", "name": "fo", "port": cl.server.port,
            "host": "127.0.0.1", "type": "remote", "enabled": True})
        assert r.status == 200
        r = await cl.post("/interrupt", json={"fanout": True})
        body = await r.json()
        assert body["status"] == "interrupted"
        assert body["fanout"] == {"fo": True}
        from comfyui_distributed_amd.nodes.runtime import get_runtime

        get_runtime().clear_interrupt()
        # no fanout flag -> no relay key
        r = await cl.post("/interrupt")
        body = await r.json()
        assert "fanout" not in body
        get_runtime().clear_interrupt()

        # /view serves saved outputs, rejects traversal, 404s missing
        out_dir = srv.executor.context.setdefault("output_dir", str(tmp_path))
        from pathlib import Path

        Path(out_dir).mkdir(parents=True, exist_ok=True)
        (Path(out_dir) / "res.png").write_bytes(b"\x89PNG fake")
        r = await cl.get("/view?filename=res.png")
        assert r.status == 200
        assert r.content_type == "image/png"
        assert await r.read() == b"\x89PNG fake"
        r = await cl.get("/view?filename=../res.png")
        assert r.status in (200, 404)  # basename-sanitized, never escapes
        r = await cl.get("/view?filename=nope.png")
        assert r.status == 404
        from comfyui_distributed_amd.server.network import close_client_session

        await close_client_session()

    run(loop, go())


def test_history_records_prompt_outcomes(client, tmp_path):
    srv, cl, loop = client

    async def go():
        srv.executor.context["output_dir"] = str(tmp_path)
        srv.executor.context["device"] = "cpu"
        good = {
            "1": {"class_type": "LoadImage",
                  "inputs": {"image": "synthetic:8x8"}},
            "2": {"class_type": "SaveImage",
                  "inputs": {"images": ["1", 0], "filename_prefix": "hist"}},
        }
        r = await cl.post("/prompt", json={"prompt": good, "client_id": "h"})
        assert r.status == 200
        pid = (await r.json())["prompt_id"]
        for _ in range(100):
            r = await cl.get(f"/history/{pid}")
            body = await r.json()
            if body:
                break
            await asyncio.sleep(0.05)
        entry = body[pid]
        assert entry["status"]["completed"] is True
        images = entry["outputs"]["images"]
        assert images and images[0]["filename"].startswith("hist")
        # the recorded output is fetchable via /view
        r = await cl.get(f"/view?filename={images[0]['filename']}")
        assert r.status == 200

        bad = {"1": {"class_type": "LoadImage",
                     "inputs": {"image": "does-not-exist.png"}},
               "2": {"class_type": "SaveImage",
                     "inputs": {"images": ["1", 0]}}}
        r = await cl.post("/prompt", json={"prompt": bad, "client_id": "h"})
        pid2 = (await r.json())["prompt_id"]
        for _ in range(100):
            body = await (await cl.get(f"/history/{pid2}")).json()
            if body:
                break
            await asyncio.sleep(0.05)
        assert body[pid2]["status"]["completed"] is False
        # full listing contains both
        full = await (await cl.get("/history")).json()
        assert pid in full and pid2 in full

    run(loop, go())


def test_workflow_examples_endpoint(client, tmp_path, monkeypatch):
    srv, cl, loop = client
    (tmp_path / "demo.json").write_text(
        '{"_comment": "x", "1": {"class_type": "DistributedSeed",'
        ' "inputs": {"seed": 1}}}')
    monkeypatch.setenv("DISTGPU_WORKFLOWS_DIR", str(tmp_path))

    async def go():
        r = await cl.get("/distributed/workflow_examples")
        assert (await r.json())["workflows"] == ["demo.json"]
        r = await cl.get("/distributed/workflow_examples?name=demo.json")
        body = await r.json()
        assert "_comment" not in body["prompt"]
        assert body["prompt"]["1"]["class_type"] == "DistributedSeed"
        r = await cl.get("/distributed/workflow_examples?name=../etc/passwd")
        assert r.status == 404

    run(loop, go())


def test_panel_javascript_syntax():
    """Extract the panel's inline script and syntax-check it with node
    (the reference ships a vitest suite for its JS; this is the
    equivalent guard for the single-page panel)."""
    import shutil
    import subprocess
    import tempfile
    from pathlib import Path

    if shutil.which("node") is None:
        pytest.skip("node not available")
    html = (Path(__file__).resolve().parent.parent /
            "comfyui_distributed_amd/server/static/panel.html").read_text()
    js = html.split("<script>")[1].split("</script>")[0]
    with tempfile.NamedTemporaryFile("w", suffix=".js", delete=False) as fh:
        fh.write(js)
        path = fh.name
    proc = subprocess.run(["node", "--check", path],
                          capture_output=True, text=True, timeout=30)
    assert proc.returncode == 0, proc.stderr


def test_malformed_json_returns_400_everywhere(client):
    srv, cl, loop = client
    posts = ["/prompt", "/distributed/queue", "/distributed/job_complete",
             "/distributed/heartbeat", "/distributed/submit_tiles",
             "/distributed/request_image", "/distributed/job_status",
             "/distributed/config/update_worker", "/distributed/check_file",
             "/distributed/worker/clear_launching"]

    async def go():
        for p in posts:
            r = await cl.post(p, data=b"{not json",
                              headers={"Content-Type": "application/json"})
            assert r.status == 400, (p, r.status)
        # /interrupt treats the body as optional: still succeeds
        r = await cl.post("/interrupt", data=b"{not json",
                          headers={"Content-Type": "application/json"})
        assert r.status == 200
        from comfyui_distributed_amd.nodes.runtime import get_runtime

        get_runtime().clear_interrupt()

    run(loop, go())


def test_bad_media_payloads_are_client_errors(client):
    srv, cl, loop = client

    async def go():
        # valid base64 that is not a PNG -> 400, not a PIL 500
        r = await cl.post("/distributed/job_complete", json={
            "job_id": "x", "worker_id": "w", "image": "QUJD",
            "is_last": True})
        assert r.status == 400
        assert "invalid image payload" in (await r.text())
        await cl.post("/distributed/prepare_job", json={"job_id": "si"})
        # garbage image_idx on a live dynamic job -> 400
        from comfyui_distributed_amd.nodes.runtime import get_runtime

        await get_runtime().job_state.init_dynamic_job("dj", 1)
        r = await cl.post("/distributed/submit_image", json={
            "job_id": "dj", "worker_id": "w", "image_idx": "nope",
            "image": "QUJD"})
        assert r.status == 400

    run(loop, go())


def test_view_subfolder(client, tmp_path):
    srv, cl, loop = client

    async def go():
        from pathlib import Path

        out = Path(srv.executor.context.setdefault("output_dir",
                                                   str(tmp_path)))
        (out / "runB").mkdir(parents=True, exist_ok=True)
        (out / "runB" / "x.png").write_bytes(b"\x89PNGdata")
        r = await cl.get("/view?filename=x.png&subfolder=runB")
        assert r.status == 200
        r = await cl.get("/view?filename=x.png&subfolder=../etc")
        assert r.status == 400

    run(loop, go())


def test_history_subfolder_outputs_fetchable(client, tmp_path):
    srv, cl, loop = client

    async def go():
        srv.executor.context["output_dir"] = str(tmp_path)
        srv.executor.context["device"] = "cpu"
        prompt = {
            "1": {"class_type": "LoadImage",
                  "inputs": {"image": "synthetic:8x8"}},
            "2": {"class_type": "SaveImage",
                  "inputs": {"images": ["1", 0],
                             "filename_prefix": "runX/img"}},
        }
        r = await cl.post("/prompt", json={"prompt": prompt, "client_id": "s"})
        pid = (await r.json())["prompt_id"]
        for _ in range(100):
            body = await (await cl.get(f"/history/{pid}")).json()
            if body:
                break
            await asyncio.sleep(0.05)
        im = body[pid]["outputs"]["images"][0]
        assert im["subfolder"] == "runX"
        r = await cl.get(f"/view?filename={im['filename']}"
                         f"&subfolder={im['subfolder']}")
        assert r.status == 200

    run(loop, go())


def test_view_content_types(client, tmp_path):
    srv, cl, loop = client

    async def go():
        from pathlib import Path

        out = Path(srv.executor.context.setdefault("output_dir",
                                                   str(tmp_path)))
        out.mkdir(parents=True, exist_ok=True)
        for name, ctype in (("a.png", "image/png"), ("b.webp", "image/webp"),
                            ("c.wav", "audio/wav")):
            (out / name).write_bytes(b"data")
            r = await cl.get(f"/view?filename={name}")
            assert r.status == 200 and r.content_type == ctype, name

    run(loop, go())


def test_static_js_served_and_confined(tmp_config):
    """The panel's shared JS module is served from /static/, and path
    escapes are rejected."""
    import asyncio

    from aiohttp.test_utils import TestClient, TestServer

    from comfyui_distributed_amd.server.app import DistributedServer

    async def go():
        cl = TestClient(TestServer(DistributedServer().build_app()))
        await cl.start_server()
        try:
            r = await cl.get("/static/js/distributed.js")
            assert r.status == 200
            body = await r.text()
            assert "decideSubmission" in body and "valueWidgetModel" in body
            assert r.headers["Content-Type"].startswith("application/javascript")
            r = await cl.get("/static/../app.py")
            assert r.status == 404
        finally:
            await cl.close()

    asyncio.run(go())
