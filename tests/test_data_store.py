"""Data-store tests: file store (local + HTTP server), tensor store through
the pod-data-server (same-node + cross-server gloo broadcast on CPU), and
the hipIpc GPU path (-m gpu). Reference parity model: test_store.py +
test_gpu_store.py in the reference suite."""
import multiprocessing as mp
import os
import socket
import subprocess
import sys
import tempfile
import time

import pytest
import torch

# daemon-spawning tests are sensitive to CI host load spikes
pytestmark = pytest.mark.flaky_retry

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture()
def tmp_store(tmp_path, monkeypatch):
    monkeypatch.setenv("KT_STORE_ROOT", str(tmp_path / "store"))
    monkeypatch.delenv("KT_STORE_URL", raising=False)
    import kubetorch_amd.data_store.commands as cmds

    monkeypatch.setattr(cmds, "LOCAL_STORE_ROOT", str(tmp_path / "store"))
    return tmp_path


def test_file_put_get_local(tmp_store):
    from kubetorch_amd.data_store import commands as ds

    src = tmp_store / "src"
    src.mkdir()
    (src / "a.txt").write_text("hello")
    (src / "sub").mkdir()
    (src / "sub" / "b.txt").write_text("world")
    ds.put("proj/code", src=str(src))
    keys = {e["key"] for e in ds.ls("proj")}
    assert "proj/code/a.txt" in keys and "proj/code/sub/b.txt" in keys
    dest = tmp_store / "dest"
    ds.get("proj/code", dest=str(dest))
    assert (dest / "a.txt").read_text() == "hello"
    ds.rm("proj/code")
    assert ds.ls("proj") == []


def test_file_store_http_server(tmp_store):
    """Full HTTP store service: upload dir as tar, download, ls, rm, meta."""
    import httpx
    import uvicorn

    from kubetorch_amd.data_store import server as store_server

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    store_server.DATA_ROOT = str(tmp_store / "httproot")
    os.makedirs(store_server.DATA_ROOT, exist_ok=True)
    config = uvicorn.Config(store_server.app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    import threading

    threading.Thread(target=server.run, daemon=True).start()
    url = f"http://127.0.0.1:{port}"
    deadline = time.time() + 15
    while time.time() < deadline:
        try:
            if httpx.get(url + "/health", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)

    os.environ["KT_STORE_URL"] = url
    try:
        from kubetorch_amd.data_store import commands as ds

        src = tmp_store / "src2"
        src.mkdir()
        (src / "model.txt").write_text("weights")
        ds.put("ns/run1", src=str(src))
        dest = tmp_store / "down"
        ds.get("ns/run1", dest=str(dest))
        assert (dest / "model.txt").read_text() == "weights"
        assert any(e["key"].endswith("model.txt") for e in ds.ls("ns"))
        # delta sync: change one file -> only it is re-uploaded
        import time as _t

        (src / "model.txt").write_text("weights-v2")
        (src / "extra.txt").write_text("new")
        _t.sleep(0.01)
        out = ds.put("ns/run1", src=str(src))
        assert out.get("changed") == 2, out
        out2 = ds.put("ns/run1", src=str(src))
        assert out2.get("changed") == 0, out2
        os.remove(src / "extra.txt")
        out3 = ds.put("ns/run1", src=str(src))
        assert out3.get("removed") == 1, out3
        dest2 = tmp_store / "down2"
        ds.get("ns/run1", dest=str(dest2))
        assert (dest2 / "model.txt").read_text() == "weights-v2"
        assert not (dest2 / "extra.txt").exists()
        # metadata endpoints
        httpx.post(url + "/meta/ns/k1", json={"host": "1.2.3.4"}).raise_for_status()
        assert httpx.get(url + "/meta/ns/k1").json()["host"] == "1.2.3.4"
        # log store
        httpx.post(url + "/logs/push", json={
            "service": "svc", "entries": [{"ts": time.time(), "line": "hi",
                                           "request_id": "r1"}]})
        got = httpx.get(url + "/logs/tail", params={"service": "svc"}).json()
        assert got["entries"][0]["line"] == "hi"
        ds.rm("ns/run1")
    finally:
        os.environ.pop("KT_STORE_URL", None)
        server.should_exit = True


class TestTensorStore:
    @pytest.fixture()
    def pd_env(self, tmp_path, monkeypatch):
        sock = str(tmp_path / "pd.sock")
        monkeypatch.setenv("KT_GPU_DATA_SOCK", sock)
        monkeypatch.setenv("KT_STORE_ROOT", str(tmp_path / "store"))
        monkeypatch.delenv("KT_STORE_URL", raising=False)
        import kubetorch_amd.data_store.pod_data_server as pds

        monkeypatch.setattr(pds, "SOCK_PATH", sock)
        monkeypatch.setattr(pds, "LOCK_PATH", sock + ".lock")
        import kubetorch_amd.data_store.gpu_store as gs

        gs._client = None
        yield sock
        gs._client = None

    def test_tensor_put_get_same_node_cpu(self, pd_env):
        from kubetorch_amd.data_store import gpu_store

        t = torch.randn(16, 8)
        gpu_store.put("w1", t)
        dest = torch.zeros(16, 8)
        gpu_store.get("w1", dest)
        torch.testing.assert_close(dest, t)
        gpu_store.rm("w1")

    def test_state_dict_put_get_packed(self, pd_env):
        from kubetorch_amd.data_store import gpu_store
        from kubetorch_amd.data_store.types import BroadcastWindow

        sd = {"a": torch.randn(4, 4), "b": torch.randn(8)}
        gpu_store.put("ckpt", sd, window=BroadcastWindow(pack=True))
        dest = {"a": torch.zeros(4, 4), "b": torch.zeros(8)}
        gpu_store.get("ckpt", dest)
        torch.testing.assert_close(dest["a"], sd["a"])
        torch.testing.assert_close(dest["b"], sd["b"])

    def test_state_dict_put_get_unpacked(self, pd_env):
        from kubetorch_amd.data_store import gpu_store

        sd = {"x": torch.randn(3), "y": torch.randn(5, 2)}
        gpu_store.put("ckpt2", sd)
        dest = {"x": torch.zeros(3), "y": torch.zeros(5, 2)}
        gpu_store.get("ckpt2", dest)
        torch.testing.assert_close(dest["y"], sd["y"])

    def test_cross_server_broadcast_cpu(self, pd_env, tmp_path):
        """Two daemons on localhost = two 'nodes'; transfer via a 2-rank
        gloo process group (the RCCL path on GPU nodes)."""
        from kubetorch_amd.data_store.pod_data_server import (
            PodDataClient,
            ensure_server,
        )

        sock_a = str(tmp_path / "a.sock")
        sock_b = pd_env
        port_a = _free_port()
        ensure_server(sock_a, tcp_port=port_a)
        cli_a = PodDataClient(sock_a, autostart=False)
        t = torch.arange(32, dtype=torch.float32)
        cli_a.request({"cmd": "register", "key": "remote_w",
                       "payload": _export(t)})

        cli_b = PodDataClient(sock_b)
        dest = torch.zeros(32)
        cli_b.fetch_remote("remote_w", dest, f"127.0.0.1:{port_a}")
        torch.testing.assert_close(dest, t)


def _export(t):
    from kubetorch_amd.data_store.pod_data_server import export_tensor

    return export_tensor(t)


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


# ---------------------------------------------------------------------------
# GPU: hipIpc zero-copy path between two processes on one MI355X
# ---------------------------------------------------------------------------
def _gpu_putter(sock, q):
    os.environ["KT_GPU_DATA_SOCK"] = sock
    import kubetorch_amd.data_store.pod_data_server as pds

    pds.SOCK_PATH = sock
    pds.LOCK_PATH = sock + ".lock"
    import torch

    from kubetorch_amd.data_store import gpu_store
    gpu_store._client = None

    t = torch.arange(1024, dtype=torch.bfloat16, device="cuda") * 0.5
    gpu_store.put("gpu_w", t)
    q.put("published")
    time.sleep(600)  # keep the owning process (and its HBM) alive


@pytest.mark.gpu
def test_gpu_ipc_put_get(tmp_path):
    """Publish a GPU tensor from process A; process B receives it through
    the daemon's hipIpc device-to-device copy (no RCCL on same node)."""
    sock = str(tmp_path / "gpu_pd.sock")
    os.environ["KT_GPU_DATA_SOCK"] = sock
    os.environ["KT_STORE_ROOT"] = str(tmp_path / "store")
    import kubetorch_amd.data_store.pod_data_server as pds

    pds.SOCK_PATH = sock
    pds.LOCK_PATH = sock + ".lock"
    import kubetorch_amd.data_store.gpu_store as gs

    gs._client = None
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_gpu_putter, args=(sock, q), daemon=True)
    p.start()
    try:
        assert q.get(timeout=420) == "published"
        from kubetorch_amd.data_store import gpu_store

        dest = torch.zeros(1024, dtype=torch.bfloat16, device="cuda")
        gpu_store.get("gpu_w", dest)
        expected = torch.arange(1024, dtype=torch.bfloat16, device="cuda") * 0.5
        torch.testing.assert_close(dest, expected)
    finally:
        p.terminate()
        p.join(10)


def _bcast_receiver(sock, store_root, rank_tag, q):
    os.environ["KT_GPU_DATA_SOCK"] = sock
    os.environ["KT_STORE_ROOT"] = store_root
    os.environ.pop("KT_STORE_URL", None)
    import kubetorch_amd.data_store.pod_data_server as pds

    pds.SOCK_PATH = sock
    pds.LOCK_PATH = sock + ".lock"
    import kubetorch_amd.data_store.gpu_store as gs

    gs._client = None
    import torch as t

    from kubetorch_amd.data_store.types import BroadcastWindow

    dest = {"w": t.zeros(64), "b": t.zeros(8)}
    gs.get("bc/sd", dest, window=BroadcastWindow(world_size=3, timeout=60))
    q.put((rank_tag, dest["w"].sum().item(), dest["b"].sum().item()))


def test_three_party_broadcast_cpu(tmp_path, monkeypatch):
    """1 source + 2 receivers through ONE gloo broadcast group (the RCCL
    path on GPU nodes; BroadcastWindow coordination via the metadata
    layer)."""
    sock = str(tmp_path / "bc.sock")
    store_root = str(tmp_path / "store")
    monkeypatch.setenv("KT_GPU_DATA_SOCK", sock)
    monkeypatch.setenv("KT_STORE_ROOT", store_root)
    monkeypatch.delenv("KT_STORE_URL", raising=False)
    import kubetorch_amd.data_store.pod_data_server as pds

    monkeypatch.setattr(pds, "SOCK_PATH", sock)
    monkeypatch.setattr(pds, "LOCK_PATH", sock + ".lock")
    import kubetorch_amd.data_store.gpu_store as gs

    gs._client = None
    import torch as t

    from kubetorch_amd.data_store.types import BroadcastWindow

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    recs = [ctx.Process(target=_bcast_receiver,
                        args=(sock, store_root, i, q), daemon=True)
            for i in range(2)]
    sd = {"w": t.ones(64) * 2, "b": t.ones(8) * 3}
    gs.put("bc/sd", sd, window=BroadcastWindow(world_size=3, timeout=60))
    for p in recs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for _tag, wsum, bsum in results:
        assert wsum == 128.0 and bsum == 24.0, results
    for p in recs:
        p.join(30)
    gs._client = None


def test_daemon_failure_auto_restart(tmp_path, monkeypatch):
    """Kill the pod-data-server; the client's next request restarts it
    (reference parity: test_gpu_store.py failure injection + auto-restart)."""
    sock = str(tmp_path / "fi.sock")
    monkeypatch.setenv("KT_GPU_DATA_SOCK", sock)
    monkeypatch.setenv("KT_STORE_ROOT", str(tmp_path / "store"))
    import kubetorch_amd.data_store.pod_data_server as pds

    monkeypatch.setattr(pds, "SOCK_PATH", sock)
    monkeypatch.setattr(pds, "LOCK_PATH", sock + ".lock")
    cli = pds.PodDataClient(sock)
    assert cli.ping()["ok"]

    # find and kill the daemon by its exact pid (match on our socket path)
    import subprocess

    out = subprocess.run(["pgrep", "-af", "pod_data_server"],
                         capture_output=True, text=True).stdout
    pids = [line.split()[0] for line in out.splitlines() if sock in line]
    assert pids, out
    for pid in pids:
        subprocess.run(["kill", "-9", pid], check=True)
    time.sleep(1.0)

    # client retries + auto-restart -> service recovers (registry is fresh)
    t = torch.randn(8)
    cli.register("recovered", t)
    r = cli.request({"cmd": "list"})
    assert "recovered" in r["keys"]


# ---------------------------------------------------------------------------
# Filesystem tree broadcast (reference: join_fs_broadcast rolling tree)
# ---------------------------------------------------------------------------
@pytest.fixture()
def http_store(tmp_store):
    """A live HTTP store service with KT_STORE_URL exported."""
    import threading

    import httpx
    import uvicorn

    from kubetorch_amd.data_store import server as store_server

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    store_server.DATA_ROOT = str(tmp_store / "fsbroot")
    os.makedirs(store_server.DATA_ROOT, exist_ok=True)
    config = uvicorn.Config(store_server.app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    threading.Thread(target=server.run, daemon=True).start()
    url = f"http://127.0.0.1:{port}"
    deadline = time.time() + 15
    while time.time() < deadline:
        try:
            if httpx.get(url + "/health", timeout=1).status_code == 200:
                break
        except Exception:
            time.sleep(0.05)
    old = os.environ.get("KT_STORE_URL")
    os.environ["KT_STORE_URL"] = url
    yield url
    server.should_exit = True
    if old is None:
        os.environ.pop("KT_STORE_URL", None)
    else:
        os.environ["KT_STORE_URL"] = old


def test_fs_broadcast_tree_chain(http_store, tmp_store):
    """fanout=1: the 2nd and 3rd joiners must be fed by completed PEERS,
    not the store (store egress stays O(1))."""
    import httpx

    from kubetorch_amd.data_store import commands as ds
    from kubetorch_amd.data_store import fileserve

    src = tmp_store / "bsrc"
    src.mkdir()
    (src / "w.bin").write_bytes(b"x" * 4096)
    (src / "cfg.json").write_text("{}")
    ds.put("ns/bcast1", src=str(src))

    # joiner A: store is the only source
    a_dest = str(tmp_store / "a")
    out_a = ds.get_broadcast("ns/bcast1", a_dest, fanout=1)
    assert (tmp_store / "a" / "w.bin").read_bytes() == b"x" * 4096
    st = httpx.get(http_store + "/fsbcast/status",
                   params={"key": "ns/bcast1"}).json()
    assert len(st["sources"]) == 1  # A registered as a source
    a_url = st["sources"][0]

    # joiner B: must be assigned peer A (peers preferred over the store)
    r = httpx.post(http_store + "/fsbcast/join",
                   json={"key": "ns/bcast1", "fanout": 1})
    assert r.json()["source"] == a_url
    # while B is mid-download, A is saturated (fanout=1) and the store has
    # capacity 1 -> C gets the store; D must wait
    r2 = httpx.post(http_store + "/fsbcast/join",
                    json={"key": "ns/bcast1", "fanout": 1})
    assert r2.json()["source"] == "store"
    r3 = httpx.post(http_store + "/fsbcast/join",
                    json={"key": "ns/bcast1", "fanout": 1})
    assert r3.json() == {"wait": True}
    # B completes -> A has a free slot again
    httpx.post(http_store + "/fsbcast/complete",
               json={"key": "ns/bcast1", "parent": a_url})
    r4 = httpx.post(http_store + "/fsbcast/join",
                    json={"key": "ns/bcast1", "fanout": 1})
    assert r4.json()["source"] == a_url

    # the peer actually serves the bytes (fetch from A's BcastFileServer)
    b_dest = str(tmp_store / "b")
    ds._fetch_from_peer(a_url, "ns/bcast1", b_dest, 30)
    assert (tmp_store / "b" / "w.bin").read_bytes() == b"x" * 4096
    assert (tmp_store / "b" / "cfg.json").read_text() == "{}"
    httpx.delete(http_store + "/fsbcast/ns/bcast1")
    srv = fileserve._server
    if srv is not None:
        srv.close()
        fileserve._server = None


def test_fs_broadcast_concurrent(http_store, tmp_store):
    """8 concurrent getters, fanout=2: everyone converges with correct
    bytes and the store served at most 2 downloads directly."""
    import threading

    import httpx

    from kubetorch_amd.data_store import commands as ds
    from kubetorch_amd.data_store import fileserve

    src = tmp_store / "csrc"
    src.mkdir()
    (src / "w.bin").write_bytes(b"y" * 10000)
    ds.put("ns/bcast2", src=str(src))

    store_hits = []
    orig_get = ds.get

    def counting_get(key, dest=None, **kw):
        store_hits.append(dest)
        return orig_get(key, dest, **kw)

    ds.get = counting_get
    try:
        errs = []

        def worker(i):
            try:
                d = str(tmp_store / f"w{i}")
                ds.get_broadcast("ns/bcast2", d, fanout=2)
                assert (tmp_store / f"w{i}" / "w.bin").read_bytes() == b"y" * 10000
            except Exception as e:  # pragma: no cover
                errs.append(e)

        threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(60)
        assert not errs, errs
        # store handled at most its fanout's worth of direct downloads
        # (every later joiner was fed by a completed peer)
        assert len(store_hits) <= 4, store_hits
    finally:
        ds.get = orig_get
        httpx.delete(http_store + "/fsbcast/ns/bcast2")
        srv = fileserve._server
        if srv is not None:
            srv.close()
            fileserve._server = None


def test_put_batch_keys(tmp_path):
    """Batch mode: put(key=[...], src=[...]) stores pairwise (reference:
    data_store_cmds.put list keys)."""
    import kubetorch_amd as kt

    a = tmp_path / "a.txt"; a.write_text("A")
    b = tmp_path / "b.txt"; b.write_text("B")
    out = kt.put(["batch/ka", "batch/kb"], [str(a), str(b)])
    assert len(out) == 2
    da = tmp_path / "outa"; db = tmp_path / "outb"
    kt.get("batch/ka", str(da)); kt.get("batch/kb", str(db))
    ra = da / "a.txt" if (da / "a.txt").exists() else da
    rb = db / "b.txt" if (db / "b.txt").exists() else db
    assert ra.read_text() == "A" and rb.read_text() == "B"
    kt.rm("batch/ka"); kt.rm("batch/kb")
    import pytest as _pytest

    with _pytest.raises(ValueError, match="keys but"):
        kt.put(["k1", "k2"], [str(a)])


def test_get_batch_and_contents(tmp_path):
    import kubetorch_amd as kt

    a = tmp_path / "x.txt"; a.write_text("XYZ")
    kt.put("gb/k1", str(a)); kt.put("gb/k2", str(a))
    outs = kt.get(["gb/k1", "gb/k2"],
                  [str(tmp_path / "o1"), str(tmp_path / "o2")])
    assert len(outs) == 2
    assert kt.get("gb/k1", contents=True) == b"XYZ"
    kt.rm("gb/k1"); kt.rm("gb/k2")


def test_rm_prefix_bulk(tmp_path):
    import kubetorch_amd as kt

    a = tmp_path / "x.txt"; a.write_text("1")
    kt.put("bulk/a", str(a)); kt.put("bulk/b", str(a)); kt.put("keepme/c", str(a))
    kt.rm("bulk", prefix=True)
    assert not [e for e in kt.ls("") if e["key"].startswith("bulk")]
    assert [e for e in kt.ls("keepme")]
    kt.rm("keepme", prefix=True)
