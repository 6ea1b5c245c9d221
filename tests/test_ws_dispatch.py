"""WebSocket dispatch_prompt/dispatch_ack protocol test (reference
dispatch.py:62-95 semantics) against the real server app."""

import asyncio

from aiohttp.test_utils import TestClient, TestServer

from comfyui_distributed_amd.server.app import DistributedServer


def test_ws_dispatch_roundtrip(tmp_config):
    async def go():
        srv = DistributedServer()
        client = TestClient(TestServer(srv.build_app()))
        await client.start_server()
        try:
            ws = await client.ws_connect("/distributed/ws")
            await ws.send_json({
                "type": "dispatch_prompt", "request_id": "r1",
                "client_id": "c",
                "prompt": {"1": {"class_type": "DistributedSeed",
                                 "inputs": {"seed": 5}}},
            })
            ack = await ws.receive_json(timeout=10)
            assert ack == {"type": "dispatch_ack", "request_id": "r1", "ok": True}
            # invalid prompt -> nack
            await ws.send_json({
                "type": "dispatch_prompt", "request_id": "r2",
                "client_id": "c",
                "prompt": {"1": {"class_type": "Nope", "inputs": {}}},
            })
            ack = await ws.receive_json(timeout=10)
            assert ack["request_id"] == "r2" and ack["ok"] is False
            # ping/pong
            await ws.send_json({"type": "ping"})
            assert (await ws.receive_json(timeout=10)) == {"type": "pong"}
            await ws.close()
        finally:
            await client.close()

    asyncio.run(go())


def test_dispatch_worker_prompt_ws_fallback_to_post(tmp_config, monkeypatch):
    """When the worker has no WS endpoint the dispatcher falls back to
    POST /prompt (covered by orchestration fanout tests via the mock); here
    we exercise the real fallback against a server that serves both."""

    async def go():
        srv = DistributedServer()
        client = TestClient(TestServer(srv.build_app()))
        await client.start_server()
        try:
            from comfyui_distributed_amd.server import orchestration

            worker = {"id": "w", "host": "127.0.0.1",
                      "port": client.server.port, "type": "local"}
            ok = await orchestration.dispatch_worker_prompt(
                worker,
                {"1": {"class_type": "DistributedSeed", "inputs": {"seed": 2}}},
                "client", use_websocket=True,
            )
            assert ok is True
            ok = await orchestration.dispatch_worker_prompt(
                worker,
                {"1": {"class_type": "DistributedSeed", "inputs": {"seed": 2}}},
                "client", use_websocket=False,
            )
            assert ok is True
        finally:
            from comfyui_distributed_amd.server.network import close_client_session

            await close_client_session()
            await client.close()

    asyncio.run(go())
