from comfyui_distributed_amd.server.media_sync import (
    convert_path_for_platform,
    find_media_references,
)


def test_find_media_references():
    prompt = {
        "1": {"class_type": "LoadImage", "inputs": {"image": "cat.png"}},
        "2": {"class_type": "LoadVideo", "inputs": {"video": "clip.mp4"}},
        "3": {"class_type": "KSampler", "inputs": {"image": ["1", 0], "seed": 3}},
        "4": {"class_type": "LoadImage", "inputs": {"image": "synthetic:64x64"}},
        "5": {"class_type": "Note", "inputs": {"text": "not_a_file.png_but_text"}},
    }
    refs = find_media_references(prompt)
    assert ("1", "image", "cat.png") in refs
    assert ("2", "video", "clip.mp4") in refs
    assert all(nid not in ("3", "4", "5") for nid, _k, _f in refs)


def test_convert_path_separators():
    assert convert_path_for_platform("a/b/c.png", "\\") == "a\\b\\c.png"
    assert convert_path_for_platform("a\\b\\c.png", "/") == "a/b/c.png"


def test_sync_local_worker_is_noop():
    import asyncio

    from comfyui_distributed_amd.server.media_sync import sync_worker_media

    prompt = {"1": {"class_type": "LoadImage", "inputs": {"image": "cat.png"}}}
    out = asyncio.run(sync_worker_media(prompt, {"id": "w", "type": "local"}))
    assert out == prompt


def test_check_file_resolves_filename_against_input_dir(tmp_config, tmp_path):
    """The cross-machine md5 skip: the worker resolves `filename` against
    its OWN input dir instead of trusting the sender's absolute path."""
    import asyncio
    import hashlib

    from aiohttp.test_utils import TestClient, TestServer

    from comfyui_distributed_amd.server.app import DistributedServer

    async def go():
        srv = DistributedServer()
        srv.executor.context["input_dir"] = str(tmp_path)
        (tmp_path / "cat.png").write_bytes(b"pngdata")
        cl = TestClient(TestServer(srv.build_app()))
        await cl.start_server()
        try:
            r = await cl.post("/distributed/check_file",
                              json={"filename": "cat.png",
                                    "path": "/nonexistent/master/cat.png"})
            body = await r.json()
            assert body["exists"] is True
            assert body["md5"] == hashlib.md5(b"pngdata").hexdigest()
            r = await cl.post("/distributed/check_file",
                              json={"filename": "missing.png"})
            assert (await r.json())["exists"] is False
        finally:
            await cl.close()

    asyncio.run(go())


def test_is_local_worker_heuristics(tmp_config):
    import asyncio
    import uuid

    from comfyui_distributed_amd.server import media_sync

    run = asyncio.run
    assert run(media_sync.is_local_worker({"host": "", "type": "remote"}))
    assert run(media_sync.is_local_worker({"host": "127.0.0.1",
                                           "type": "remote"}))
    assert run(media_sync.is_local_worker({"host": "10.0.0.9",
                                           "type": "local"}))
    assert not run(media_sync.is_local_worker({"host": "10.0.0.9",
                                               "type": "remote"}))

    # cloud worker: machine-id comparison decides (faked system_info)
    async def same_info(url):
        return {"machine_id": hex(uuid.getnode())}

    async def other_info(url):
        return {"machine_id": "0xdeadbeef"}

    w = {"host": "198.51.100.4", "port": 8189, "type": "cloud"}
    orig = media_sync.fetch_worker_system_info
    try:
        media_sync.fetch_worker_system_info = same_info
        assert run(media_sync.is_local_worker(w)) is True
        media_sync.fetch_worker_system_info = other_info
        assert run(media_sync.is_local_worker(w)) is False
    finally:
        media_sync.fetch_worker_system_info = orig


def test_check_file_rejects_out_of_tree_path(tmp_config, tmp_path):
    """the raw `path` fallback must not be an arbitrary-file
    md5 oracle — only paths inside the input/output dirs are answered."""
    import asyncio

    from aiohttp.test_utils import TestClient, TestServer

    from comfyui_distributed_amd.server.app import DistributedServer

    secret = tmp_path / "secret.bin"
    secret.write_bytes(b"topsecret")
    input_dir = tmp_path / "input"
    input_dir.mkdir()
    inside = input_dir / "ok.png"
    inside.write_bytes(b"pngdata")

    async def go():
        srv = DistributedServer()
        srv.executor.context["input_dir"] = str(input_dir)
        srv.executor.context["output_dir"] = str(tmp_path / "out")
        cl = TestClient(TestServer(srv.build_app()))
        await cl.start_server()
        try:
            # outside path: existence/hash not disclosed
            r = await cl.post("/distributed/check_file",
                              json={"path": str(secret)})
            assert (await r.json())["exists"] is False
            # inside path still works (same-filesystem fast path)
            r = await cl.post("/distributed/check_file",
                              json={"path": str(inside)})
            assert (await r.json())["exists"] is True
        finally:
            await cl.close()

    asyncio.run(go())
