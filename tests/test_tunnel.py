"""Controller TCP tunnel (reference: websocket_tunnel.py): an external
client reaches an in-cluster service through the controller's one public
port. Here: an HTTP store service is the target; the client does a full
put/get round trip with KT_STORE_URL pointed at the tunnel."""
import os
import socket
import threading
import time

import httpx
import pytest
import uvicorn

os.environ.setdefault("KT_LOCAL_MODE", "true")
os.environ.setdefault("KT_USERNAME", "tuntest")

pytestmark = pytest.mark.flaky_retry


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.mark.timeout(180)
def test_http_store_through_tunnel(tmp_path, monkeypatch):
    from kubetorch_amd.data_store import server as store_server

    port = _free_port()
    store_server.DATA_ROOT = str(tmp_path / "root")
    os.makedirs(store_server.DATA_ROOT, exist_ok=True)
    server = uvicorn.Server(uvicorn.Config(store_server.app, host="127.0.0.1",
                                           port=port, log_level="error"))
    threading.Thread(target=server.run, daemon=True).start()
    deadline = time.time() + 15
    while time.time() < deadline:
        try:
            if httpx.get(f"http://127.0.0.1:{port}/health",
                         timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)

    from kubetorch_amd.client.tunnel import TcpTunnel

    # explicit host:port target (BYO form — the store isn't a registered
    # workload in this test)
    with TcpTunnel(f"127.0.0.1:{port}", port) as t:
        base = f"http://127.0.0.1:{t.local_port}"
        # plain HTTP through the tunnel
        r = httpx.get(base + "/health", timeout=30)
        assert r.status_code == 200
        # full data-plane round trip through the tunnel
        monkeypatch.setenv("KT_STORE_URL", base)
        from kubetorch_amd.data_store import commands as ds

        src = tmp_path / "src"
        src.mkdir()
        (src / "a.txt").write_text("tunnel-payload")
        ds.put("tun/key", src)
        dest = tmp_path / "dest"
        ds.get("tun/key", dest)
        assert (dest / "a.txt").read_text() == "tunnel-payload"
        # several sequential connections (the pool reuses + reopens)
        for _ in range(3):
            assert httpx.get(base + "/health",
                             timeout=30).status_code == 200
    server.should_exit = True


@pytest.mark.timeout(240)
def test_tunnel_to_deployed_service(tmp_path):
    """Tunnel by SERVICE NAME: the controller resolves the workload's pods
    and bridges to the first one — an external client calls a deployed
    fn through the tunnel without knowing any pod address."""
    import sys as _sys

    _sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets",
                                     "summer"))
    os.environ["KT_USERNAME"] = "tunsvc"
    import kubetorch_amd as kt
    from tests.assets.summer import summer as summer_mod

    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        from kubetorch_amd.client.tunnel import TcpTunnel

        with TcpTunnel(f.name, 32300) as t:
            base = f"http://127.0.0.1:{t.local_port}"
            r = httpx.get(base + "/health", timeout=30)
            assert r.status_code == 200
            r = httpx.post(f"{base}/call/summer",
                           json={"args": [20, 22], "kwargs": {}}, timeout=60)
            assert r.status_code == 200 and r.json()["result"] == 42
    finally:
        f.teardown()


def test_tunnel_refuses_nonloopback_hostport():
    """Explicit host:port targets must not turn the controller into a
    network pivot (403 for anything but loopback on the local driver)."""
    from kubetorch_amd.globals import controller_client

    cc = controller_client()
    r = httpx.post(cc.base_url + "/controller/tunnel/open",
                   json={"service": "10.0.0.5:22", "port": 22}, timeout=10)
    assert r.status_code == 403
