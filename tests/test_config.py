import asyncio
import json

from comfyui_distributed_amd.utils import config as config_mod


def test_defaults_when_missing(tmp_config):
    cfg = config_mod.load_config()
    assert cfg["master"]["port"] == 8188
    assert cfg["workers"] == []
    assert cfg["settings"]["websocket_orchestration"] is True


def test_merge_preserves_unknown_keys(tmp_config):
    tmp_config.write_text(json.dumps({"settings": {"debug": True, "custom": 42}}))
    cfg = config_mod.load_config()
    assert cfg["settings"]["debug"] is True
    assert cfg["settings"]["custom"] == 42
    assert cfg["settings"]["auto_launch_workers"] is False  # default filled in


def test_atomic_save_roundtrip(tmp_config):
    cfg = config_mod.load_config()
    cfg["workers"].append(
        {"id": "w0", "name": "GPU1", "host": "", "port": 8189,
         "cuda_device": 1, "enabled": True, "type": "local"}
    )
    config_mod.save_config(cfg)
    cfg2 = config_mod.load_config()
    assert cfg2["workers"][0]["id"] == "w0"
    assert config_mod.enabled_workers(cfg2)[0]["port"] == 8189


def test_mtime_cache_invalidation(tmp_config):
    cfg = config_mod.load_config()
    assert cfg["settings"]["debug"] is False
    # external write with different mtime
    tmp_config.write_text(json.dumps({"settings": {"debug": True}}))
    import os

    os.utime(tmp_config, ns=(1, 1))
    cfg = config_mod.load_config()
    assert cfg["settings"]["debug"] is True


def test_transaction_saves_only_on_change(tmp_config):
    async def run():
        async with config_mod.config_transaction() as cfg:
            cfg["settings"]["debug"] = True
        assert config_mod.load_config()["settings"]["debug"] is True
        # no-op transaction must not rewrite
        mtime = tmp_config.stat().st_mtime_ns
        async with config_mod.config_transaction():
            pass
        assert tmp_config.stat().st_mtime_ns == mtime

    asyncio.run(run())


def test_worker_timeout_resolution(tmp_config):
    from comfyui_distributed_amd.utils import constants

    cfg = config_mod.load_config()
    assert config_mod.get_worker_timeout_seconds(cfg) == constants.HEARTBEAT_TIMEOUT
    cfg["settings"]["worker_timeout_seconds"] = 5
    assert config_mod.get_worker_timeout_seconds(cfg) == 5.0
    cfg["settings"]["worker_timeout_seconds"] = "bogus"
    assert config_mod.get_worker_timeout_seconds(cfg) == constants.HEARTBEAT_TIMEOUT


def test_get_worker_by_id(tmp_config):
    cfg = config_mod.load_config()
    cfg["workers"] = [{"id": 1, "enabled": False}, {"id": "2", "enabled": True}]
    assert config_mod.get_worker_by_id(cfg, "1") == {"id": 1, "enabled": False}
    assert config_mod.get_worker_by_id(cfg, 2)["enabled"] is True
    assert config_mod.get_worker_by_id(cfg, "missing") is None
