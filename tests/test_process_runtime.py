"""Worker process runtime: launch command building, watchdog, persistence,
tunnel reader (fake cloudflared)."""

import os
import subprocess
import sys

import pytest

from comfyui_distributed_amd.server import workers
from comfyui_distributed_amd.utils.errors import ProcessError


def test_build_launch_command_basic():
    cmd = workers.build_launch_command({"id": "w1", "port": 8190})
    assert cmd[0] == sys.executable
    assert "--port" in cmd and "8190" in cmd
    assert "--worker" in cmd


def test_build_launch_command_rejects_shell_metachars():
    with pytest.raises(ProcessError):
        workers.build_launch_command({"id": "w", "port": 1, "extra_args": "--x; rm -rf /"})


def test_extra_args_pass_through():
    cmd = workers.build_launch_command(
        {"id": "w", "port": 1, "extra_args": "--listen 0.0.0.0"}
    )
    assert "--listen" in cmd and "0.0.0.0" in cmd


def test_pid_persistence_roundtrip(tmp_config):
    workers.persist_managed_pid("w7", 12345)
    assert workers.load_managed_pid("w7") == 12345
    workers.forget_managed_pid("w7")
    assert workers.load_managed_pid("w7") is None


def test_adopt_or_cleanup(tmp_config):
    workers.persist_managed_pid("alive", os.getpid())
    workers.persist_managed_pid("dead", 99999999)
    adopted = workers.adopt_or_cleanup_managed()
    assert adopted == {"alive": os.getpid()}
    assert workers.load_managed_pid("dead") is None
    workers.forget_managed_pid("alive")


def test_worker_monitor_kills_orphan():
    """Watchdog kills the worker when the (fake) master pid dies."""
    master = subprocess.Popen([sys.executable, "-c", "import time; time.sleep(2)"])
    mon = subprocess.Popen([
        sys.executable, "-m", "comfyui_distributed_amd.server.worker_monitor",
        "--master-pid", str(master.pid), "--",
        sys.executable, "-c", "import time; time.sleep(60)",
    ])
    master.wait()
    rc = mon.wait(timeout=30)
    assert rc == 1  # monitor reports master-death shutdown


def test_tunnel_reader_extracts_url(tmp_path):
    from comfyui_distributed_amd.server.tunnel import ProcessReader

    script = tmp_path / "fake_cloudflared.py"
    script.write_text(
        "import sys, time\n"
        "print('INF starting tunnel')\n"
        "print('INF https://abc-def.trycloudflare.com registered')\n"
        "sys.stdout.flush()\n"
        "time.sleep(5)\n"
    )
    proc = subprocess.Popen([sys.executable, str(script)], stdout=subprocess.PIPE,
                            stderr=subprocess.STDOUT)
    reader = ProcessReader(proc)
    url = reader.wait_for_url(10)
    assert url == "https://abc-def.trycloudflare.com"
    proc.terminate()


def test_tunnel_manager_with_fake_binary(tmp_path, tmp_config, monkeypatch):
    import asyncio

    from comfyui_distributed_amd.server import tunnel as tunnel_mod

    fake = tmp_path / "cloudflared"
    fake.write_text(
        "#!/usr/bin/env python3\n"
        "import time\n"
        "print('https://xyz.trycloudflare.com up', flush=True)\n"
        "time.sleep(30)\n"
    )
    fake.chmod(0o755)
    monkeypatch.setenv("CLOUDFLARED_PATH", str(fake))

    async def go():
        mgr = tunnel_mod.TunnelManager()
        url = await mgr.start(8188)
        assert url == "https://xyz.trycloudflare.com"
        assert mgr.status()["active"] is True
        from comfyui_distributed_amd.utils.config import load_config

        cfg = load_config()
        assert cfg["master"]["host"] == "xyz.trycloudflare.com"
        await mgr.stop()
        cfg = load_config()
        assert cfg["master"]["host"] == ""  # restored
        assert mgr.status()["active"] is False

    asyncio.run(go())


def test_cloudflared_download_with_faked_fetch(tmp_path, monkeypatch):
    """Reference utils/cloudflare/binary.py:47-83 parity: the binary is
    fetched from the GitHub latest release, chmod +x'd, and found by the
    resolver afterwards; a fetch failure (no egress) degrades to None."""
    import os

    from comfyui_distributed_amd.server import tunnel as tmod

    monkeypatch.delenv("CLOUDFLARED_PATH", raising=False)
    fetched = {}

    def fake_fetch(url, timeout):
        fetched["url"] = url
        return b"#!/bin/sh\necho fake-cloudflared\n"

    dest = tmod.download_cloudflared(fetch=fake_fetch, dest_dir=str(tmp_path))
    assert dest and os.path.isfile(dest)
    assert os.access(dest, os.X_OK)
    assert "cloudflared/releases/latest/download" in fetched["url"]

    # resolver picks up the downloaded copy
    monkeypatch.setattr(tmod, "_download_dir", lambda: str(tmp_path))
    monkeypatch.setattr(tmod.shutil, "which", lambda name: None)
    assert tmod.find_cloudflared() == dest

    # fetch failure (no egress) -> graceful None
    def broken_fetch(url, timeout):
        raise OSError("no route to host")

    assert tmod.download_cloudflared(
        fetch=broken_fetch, dest_dir=str(tmp_path / "x")) is None
