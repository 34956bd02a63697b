"""Unit tests for the distribution layer's per-framework env contracts and
worker-selection logic (reference parity: test_distributed.py asserts the
RANK/WORLD_SIZE/... env on every rank and the worker-filter arg)."""
import json
import os

from kubetorch_amd.serving.supervisors import (
    PROCESS_CLASSES,
    JaxProcess,
    TensorflowProcess,
    TorchProcess,
)

HOSTS = ["10.0.0.1:32300", "10.0.0.2:32300", "10.0.0.3:32300"]
ASSETS = os.path.join(os.path.dirname(__file__), "assets", "summer")


def test_torch_env_contract():
    env = TorchProcess.env_vars(HOSTS, node_rank=1, local_rank=3, num_proc=8)
    assert env["WORLD_SIZE"] == "24"
    assert env["RANK"] == "11"          # node_rank*num_proc + local_rank
    assert env["LOCAL_RANK"] == "3"
    assert env["NODE_RANK"] == "1"
    assert env["LOCAL_WORLD_SIZE"] == "8"
    assert env["MASTER_ADDR"] == "10.0.0.1"
    assert env["POD_IPS"] == "10.0.0.1,10.0.0.2,10.0.0.3"
    # RCCL/xGMI defaults ride along (incl. the dmabuf IPC requirement)
    assert env.get("HSA_ENABLE_IPC_MODE_LEGACY") == "0"


def test_jax_env_contract():
    env = JaxProcess.env_vars(HOSTS, node_rank=2, local_rank=0, num_proc=1)
    assert env["JAX_COORDINATOR_ADDRESS"] == "10.0.0.1:1234"
    assert env["JAX_PROCESS_ID"] == "2"
    assert env["JAX_NUM_PROCESSES"] == "3"
    assert env["JAX_LOCAL_DEVICE_IDS"] == "0"


def test_tf_env_contract():
    env = TensorflowProcess.env_vars(HOSTS, node_rank=1, local_rank=0,
                                     num_proc=1)
    cfg = json.loads(env["TF_CONFIG"])
    assert cfg["cluster"]["worker"] == [
        "10.0.0.1:2222", "10.0.0.2:2222", "10.0.0.3:2222"]
    assert cfg["task"] == {"type": "worker", "index": 1}


def test_process_class_registry():
    for name in ("pytorch", "jax", "tensorflow", "spmd"):
        assert name in PROCESS_CLASSES


def test_k8s_driver_event_parsing(monkeypatch):
    """K8sDriver.get_events: kubectl JSON -> filtered, since-windowed event
    dicts (the cluster path that CPU CI can't run live)."""
    import json
    import subprocess
    from types import SimpleNamespace

    from kubetorch_amd.controller.drivers import K8sDriver

    payload = {"items": [
        {"involvedObject": {"name": "svc-a-0"}, "type": "Normal",
         "reason": "Scheduled", "message": "assigned node1",
         "lastTimestamp": "2026-09-11T10:00:00Z"},
        {"involvedObject": {"name": "svc-a-0"}, "type": "Warning",
         "reason": "BackOff", "message": "crash loop",
         "lastTimestamp": "2026-09-11T10:05:00Z"},
        {"involvedObject": {"name": "other-svc-0"}, "type": "Normal",
         "reason": "Scheduled", "message": "not ours",
         "lastTimestamp": "2026-09-11T10:01:00Z"},
    ]}

    def fake_run(args, **kw):
        return SimpleNamespace(returncode=0,
                               stdout=json.dumps(payload).encode())

    monkeypatch.setattr(subprocess, "run", fake_run)
    evs = K8sDriver().get_events("svc-a", "ns")
    assert [e["reason"] for e in evs] == ["Scheduled", "BackOff"]
    assert all(e["pod"].startswith("svc-a") for e in evs)
    # since-window: only events after the first timestamp
    later = K8sDriver().get_events("svc-a", "ns", since=evs[0]["ts"])
    assert [e["reason"] for e in later] == ["BackOff"]


def test_pod_ips_local_ips_env(monkeypatch):
    """pod_ips honors the KT_LOCAL_IPS fake-cluster contract and quorum."""
    monkeypatch.setenv("KT_LOCAL_IPS", "127.0.0.1:1,127.0.0.1:2,127.0.0.1:3")
    from kubetorch_amd.serving.discovery import pod_ips

    peers = pod_ips(num_workers=3, timeout=5)
    assert len(peers) == 3 and peers[0].startswith("127.0.0.1")


def test_service_url_resolution(monkeypatch):
    from kubetorch_amd.globals import service_url

    assert service_url("svc", "ns", ["1.2.3.4:32300"]) == "http://1.2.3.4:32300"
    monkeypatch.setenv("KUBERNETES_SERVICE_HOST", "10.0.0.1")
    assert "svc.ns.svc.cluster.local" in service_url("svc", "ns")
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST")
    assert service_url("svc", "ns").startswith("http://svc.ns:")


def test_endpoint_and_stock_images():
    from kubetorch_amd.resources import images
    from kubetorch_amd.resources.endpoint import Endpoint

    e = Endpoint(url="http://my-router:9000")
    assert e.resolve(default_url="http://x") == "http://my-router:9000"
    e2 = Endpoint(selector={"role": "head"}, port=8265)
    assert e2.resolve(default_url="http://x") == "http://x"
    assert e2.selector == {"role": "head"}

    img = images.pytorch()
    assert "rocm" in img.image_id.lower()
    assert images.ray().image_id.startswith("rayproject/")


def test_inactivity_ttl_env_parse(monkeypatch):
    from kubetorch_amd.serving.metrics import inactivity_ttl_seconds

    for raw, want in (("120s", 120), ("5m", 300), ("2h", 7200),
                      ("1d", 86400), ("90", 90), ("bogus", None)):
        monkeypatch.setenv("KT_INACTIVITY_TTL", raw)
        assert inactivity_ttl_seconds() == want, raw
    monkeypatch.delenv("KT_INACTIVITY_TTL")
    assert inactivity_ttl_seconds() is None


def test_ringlog_ring_and_filters():
    from kubetorch_amd.serving.log_capture import RingLog

    r = RingLog(size=5)
    for i in range(8):
        r.append(f"line{i}", request_id="r1" if i % 2 else None)
    entries = r.tail()
    assert len(entries) == 5  # ring keeps the last `size`
    assert entries[0]["line"] == "line3"
    # seq windowing + request-id filtering
    only_r1 = r.tail(request_id="r1")
    assert all(e["request_id"] == "r1" for e in only_r1)
    last_seq = entries[-1]["seq"]
    assert r.tail(since=last_seq)[0]["line"] == "line7"
    # wait_for returns immediately when new entries already exist
    assert r.wait_for(since=0, timeout=0.1)


def test_extract_pointers_and_exception_roundtrip():
    from tests.assets.summer import summer as summer_mod

    from kubetorch_amd.client.pointers import extract_pointers
    from kubetorch_amd.exceptions import (package_exception,
                                          reconstruct_exception)

    ptr = extract_pointers(summer_mod.summer)
    assert ptr["name"] == "summer"
    assert ptr["file_path"].endswith("summer.py")
    assert ptr["project_root"]

    # exception packaging -> reconstruction keeps the real class + traceback
    try:
        raise ValueError("kaboom 42")
    except ValueError as e:
        payload = package_exception(e)
    exc = reconstruct_exception(payload)
    assert isinstance(exc, ValueError) and "kaboom 42" in str(exc)
    assert "ValueError" in exc.remote_traceback
    # chained hop: re-packaging preserves the original remote traceback
    payload2 = package_exception(exc)
    exc2 = reconstruct_exception(payload2)
    assert "kaboom 42" in exc2.remote_traceback


def test_secret_provider_presets_complete():
    """All 14 reference provider presets exist and render env keys."""
    from kubetorch_amd.resources.secret import PROVIDERS, secret_factory

    expected = {"anthropic", "aws", "azure", "cohere", "gcp", "github",
                "huggingface", "kubeconfig", "lambda", "langchain",
                "openai", "pinecone", "ssh", "wandb"}
    assert expected <= set(PROVIDERS), expected - set(PROVIDERS)
    s = secret_factory("openai", values={"OPENAI_API_KEY": "sk-x"})
    m = s.to_manifest("ns")
    assert m["kind"] == "Secret"
    import pytest as _pytest

    with _pytest.raises(ValueError, match="known"):
        secret_factory("not-a-provider")


def test_config_set_persist(tmp_path):
    from kubetorch_amd.config import KTConfig

    path = tmp_path / "config"
    cfg = KTConfig(path=str(path))
    cfg.set("namespace", "teamspace", persist=True)
    assert path.exists()
    cfg2 = KTConfig(path=str(path))
    assert cfg2.namespace == "teamspace"


def test_worker_crash_fails_inflight_request(monkeypatch):
    """A worker killed mid-request fails the caller promptly with
    PodTerminatedError instead of hanging until the HTTP timeout."""
    import time

    from kubetorch_amd.serving.process_pool import ProcessPool
    from kubetorch_amd.serving.supervisors import _encode_call

    monkeypatch.setenv("KT_FILE_PATH", os.path.join(ASSETS, "summer.py"))
    monkeypatch.setenv("KT_PROJECT_ROOT", ASSETS)
    monkeypatch.setenv("KT_CLS_OR_FN_NAME", "slow_echo")
    monkeypatch.setenv("KT_MODULE_TYPE", "fn")
    pool = ProcessPool(num_proc=1)
    try:
        body = _encode_call((1,), {"delay": 60})
        fut = pool.submit(0, body)
        time.sleep(2.0)  # request in flight inside the worker
        pool.workers[0].proc.kill()
        t0 = time.time()
        resp = fut.result(timeout=30)
        assert time.time() - t0 < 15, "crash not detected promptly"
        assert resp["ok"] is False
        assert resp["error"]["error_type"] == "PodTerminatedError"
    finally:
        pool.terminate()


def test_idle_worker_death_self_heals(monkeypatch):
    """A worker that dies while idle is respawned on the next submit (the
    reference restarts workers on reload; death between requests must not
    strand the queue)."""
    from kubetorch_amd.serving.process_pool import ProcessPool
    from kubetorch_amd.serving.supervisors import _decode_resp, _encode_call

    monkeypatch.setenv("KT_FILE_PATH", os.path.join(ASSETS, "summer.py"))
    monkeypatch.setenv("KT_PROJECT_ROOT", ASSETS)
    monkeypatch.setenv("KT_CLS_OR_FN_NAME", "summer")
    monkeypatch.setenv("KT_MODULE_TYPE", "fn")
    pool = ProcessPool(num_proc=1)
    try:
        r = pool.submit(0, _encode_call((1, 2), {})).result(60)
        assert _decode_resp(r) == 3
        pool.workers[0].proc.kill()
        pool.workers[0].proc.join(10)
        r = pool.submit(0, _encode_call((5, 6), {})).result(60)
        assert _decode_resp(r) == 11
    finally:
        pool.terminate()


def test_pool_concurrent_routing_integrity(monkeypatch):
    """Many interleaved in-flight requests across 2 workers: every future
    must get ITS OWN answer (the response router matches request ids under
    concurrency — advisor-flagged lock path)."""
    from kubetorch_amd.serving.process_pool import ProcessPool
    from kubetorch_amd.serving.supervisors import _decode_resp, _encode_call

    monkeypatch.setenv("KT_FILE_PATH", os.path.join(ASSETS, "summer.py"))
    monkeypatch.setenv("KT_PROJECT_ROOT", ASSETS)
    monkeypatch.setenv("KT_CLS_OR_FN_NAME", "summer")
    monkeypatch.setenv("KT_MODULE_TYPE", "fn")
    pool = ProcessPool(num_proc=2)
    try:
        futs = []
        for i in range(40):
            body = _encode_call((i, 1000 * i), {})
            futs.append((i, pool.submit(i % 2, body)))
        for i, fut in futs:
            resp = fut.result(timeout=120)
            assert resp["ok"], resp
            assert _decode_resp(resp) == i + 1000 * i
    finally:
        pool.terminate()
