from comfyui_distributed_amd.server import network


def test_normalize_host():
    assert network.normalize_host(" http://foo.com/ ") == "foo.com"
    assert network.normalize_host("https://bar.io") == "bar.io"
    assert network.normalize_host("") == ""


def test_split_host_port():
    assert network.split_host_port("1.2.3.4:8188") == ("1.2.3.4", 8188)
    assert network.split_host_port("host") == ("host", None)
    assert network.split_host_port("[::1]:9000") == ("::1", 9000)
    assert network.split_host_port("[fe80::2]") == ("fe80::2", None)
    assert network.split_host_port("http://h:1") == ("h", 1)


def test_build_worker_url_heuristics():
    assert network.build_worker_url({"host": "", "port": 8189}) == "http://localhost:8189"
    assert network.build_worker_url({"host": "10.0.0.2", "port": 8190}) == "http://10.0.0.2:8190"
    assert (network.build_worker_url({"host": "abc.trycloudflare.com", "port": 443})
            == "https://abc.trycloudflare.com")
    assert (network.build_worker_url({"host": "x-8188.proxy.runpod.net", "port": 8188})
            == "https://x-8188.proxy.runpod.net")


def test_master_callback_url_local_loopback():
    master = {"host": "203.0.113.5", "port": 8188}
    local_worker = {"id": "w", "type": "local", "host": ""}
    remote_worker = {"id": "r", "type": "remote", "host": "10.1.1.1"}
    assert network.build_master_callback_url(master, local_worker) == "http://127.0.0.1:8188"
    assert network.build_master_callback_url(master, remote_worker) == "http://203.0.113.5:8188"


def test_master_url_tunnel():
    assert (network.build_master_url({"host": "t.trycloudflare.com", "port": 8188})
            == "https://t.trycloudflare.com")
    assert network.build_master_url({"host": "", "port": 9000}) == "http://127.0.0.1:9000"


def test_build_worker_url_host_with_embedded_port():
    assert (network.build_worker_url({"host": "10.0.0.2:9000", "port": 8190})
            == "http://10.0.0.2:9000")
