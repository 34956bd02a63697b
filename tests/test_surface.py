"""Surface-debt behaviors: endpoint selector routing
actually routes, Compute setters mutate BYO manifests in place, ssh() runs
commands through the exec route, App.wait() follows app exit."""
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "summer"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "surften"

import kubetorch_amd as kt  # noqa: E402
from tests.assets.summer import summer as summer_mod  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]


@pytest.mark.timeout(240)
def test_endpoint_selector_routes_to_one_pod():
    """2-replica service + Endpoint(selector={'pod-index': 1}): every call
    lands on pod 1 (reference: Endpoint sub-selector routing, e.g. Ray
    head only)."""
    comp = kt.Compute(cpus=1, endpoint=kt.Endpoint(selector={"pod-index": 1}))
    comp.replicas = 2
    f = kt.fn(summer_mod.pod_name).to(comp)
    try:
        seen = {f(kt_timeout=60) for _ in range(4)}
        assert len(seen) == 1
        assert seen.pop().endswith("-1")
    finally:
        f.teardown()


@pytest.mark.timeout(240)
def test_endpoint_head_role_routes_to_pod0():
    comp = kt.Compute(cpus=1, endpoint=kt.Endpoint(selector={"role": "head"}))
    comp.replicas = 2
    f = kt.fn(summer_mod.pod_name).to(comp)
    try:
        assert f(kt_timeout=60).endswith("-0")
    finally:
        f.teardown()


def test_endpoint_url_and_validation():
    ep = kt.Endpoint(url="http://my-router:9000")
    assert ep.resolve(default_url="http://x") == "http://my-router:9000"
    with pytest.raises(ValueError):
        kt.Endpoint()
    with pytest.raises(ValueError):
        kt.Endpoint(url="http://x", selector={"a": "b"})
    assert kt.Endpoint(selector={"role": "head"}).to_service_config() == \
        {"type": "selector", "selector": {"role": "head"}}


def test_service_manifest_honors_selector():
    from kubetorch_amd.provisioning.manifests import build_service_manifests

    svc, headless = build_service_manifests(
        "svc-x", "default", selector={"app": "ray", "role": "head"})
    assert svc["spec"]["selector"] == {"app": "ray", "role": "head"}
    # discovery service keeps the full pod set
    assert headless["spec"]["selector"] == {"kubetorch.amd.com/service": "svc-x"}


def test_compute_setters_mutate_raw_manifest_in_place():
    manifest = kt.Compute(gpus=1, cpus=2).to_manifest("svc-y")
    comp = kt.Compute.from_manifest(manifest)
    comp.set_gpus(4).set_memory("64Gi").set_gpu_type("MI355X").set_replicas(3)
    pod = comp._raw_manifest["spec"]["template"]["spec"]
    res = pod["containers"][0]["resources"]
    assert res["requests"]["amd.com/gpu"] == "4"
    assert res["limits"]["amd.com/gpu"] == "4"
    assert res["limits"]["memory"] == "64Gi"
    assert pod["nodeSelector"]["amd.com/gpu.product-name"] == "MI355X"
    assert comp._raw_manifest["spec"]["replicas"] == 3
    # the rendered manifest is the mutated one
    assert comp.to_manifest("svc-y")["spec"]["replicas"] == 3


@pytest.mark.timeout(240)
def test_ssh_command_mode():
    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        out = f.ssh("echo from-pod-$((2+3))")
        assert out["returncode"] == 0
        assert "from-pod-5" in out["stdout"]
    finally:
        f.teardown()


def test_ssh_interactive_refused_in_local_mode():
    f = kt.fn(summer_mod.summer)
    f.compute = kt.Compute(cpus=1)
    with pytest.raises(RuntimeError, match="local-driver"):
        f.ssh()


@pytest.mark.timeout(240)
def test_volume_persists_across_services():
    """PVC behavior on the local driver: the controller creates the
    volume at deploy, two different services mount the same claim
    (KT_VOLUME_MOUNT_<NAME> stand-in) and share state; existing-PV binds
    render volumeName in the PVC manifest."""
    vol = kt.Volume("shared-cache", size="1Gi")
    writer = kt.fn(summer_mod.write_volume).to(
        kt.Compute(cpus=1, volumes=[vol]))
    reader = kt.fn(summer_mod.read_volume).to(
        kt.Compute(cpus=1, volumes=[vol]))
    try:
        d = writer("KT_VOLUME_MOUNT_SHARED_CACHE", "state.txt", "v42",
                   kt_timeout=60)
        assert d, "volume mount env missing in pod"
        assert reader("KT_VOLUME_MOUNT_SHARED_CACHE", "state.txt",
                      kt_timeout=60) == "v42"
        from kubetorch_amd.globals import controller_client

        names = {v["name"] for v in controller_client().list_volumes("default")}
        assert "shared-cache" in names
        controller_client().delete_volume("shared-cache", "default")
    finally:
        writer.teardown()
        reader.teardown()


def test_volume_existing_pv_bind():
    m = kt.Volume("bindit", existing_pv="pv-123").to_pvc_manifest("default")
    assert m["spec"]["volumeName"] == "pv-123"
    assert m["spec"]["storageClassName"] == ""
    # existing claim -> no creation needed
    assert not kt.Volume("x", existing_claim="already").needs_create
    assert kt.Volume("y").needs_create


@pytest.mark.timeout(300)
def test_byo_pytorchjob_manifest_runs_distributed():
    """BYO: a prebuilt PyTorchJob manifest deploys through
    Compute.from_manifest and the local driver spawns the job's full
    replica count; the auto-derived SPMD config fans the call out to
    every rank (reference: test_byo_manifest.py with PyTorchJob CRDs)."""
    base = kt.Compute(cpus=1)
    base.kind = "pytorchjob"
    base.distributed_config = {"type": "pytorch", "workers": 2, "num_proc": 1}
    manifest = base.to_manifest("byojob")
    comp = kt.Compute.from_manifest(manifest)
    assert comp.kind == "pytorchjob"
    assert comp.distributed_config["workers"] == 2
    f = kt.fn(summer_mod.rank_env).to(comp)
    try:
        results = f(kt_timeout=180)
        assert len(results) == 2, results
        assert sorted(r["rank"] for r in results) == [0, 1]
    finally:
        f.teardown()


@pytest.mark.timeout(240)
def test_raycluster_local_pod_count():
    """distribute('ray', workers=3) renders a RayCluster (head + 2 worker
    groups); the local driver spawns all 3 pods (head included)."""
    f = kt.fn(summer_mod.summer).to(
        kt.Compute(cpus=1).distribute("ray", workers=3))
    from kubetorch_amd.controller.app import HUB

    try:
        assert len(HUB.driver.pods(f.name, "default")) == 3
        # no ray wheel on this image: the head-only supervisor degrades to
        # a single-worker call path, which still serves
        assert f(20, 22, kt_timeout=120) == 42
    finally:
        f.teardown()


def test_autoscale_distribute_mutually_exclusive():
    with pytest.raises(ValueError, match="mutually exclusive"):
        kt.Compute(cpus=1).distribute("pytorch", workers=2).autoscale(target=1)
    with pytest.raises(ValueError, match="mutually exclusive"):
        kt.Compute(cpus=1).autoscale(target=1).distribute("pytorch")


def test_workload_configs_expand_and_flat_override():
    """kt.LoggingConfig/MetricsConfig/DebugConfig bundle the per-call
    options; flat kt_* kwargs win over a bundle (reference parity:
    workload_configs public API)."""
    import kubetorch_amd as kt
    from kubetorch_amd.workload_configs import expand_config

    assert expand_config(None) == {}
    assert expand_config(kt.LoggingConfig()) == {"stream_logs": True}
    assert expand_config(kt.LoggingConfig(stream_logs=False)) == \
        {"stream_logs": False}
    merged = expand_config([kt.MetricsConfig(), kt.DebugConfig()])
    assert merged == {"stream_metrics": True, "debug": True}


def test_workload_config_flows_through_call(monkeypatch):
    import kubetorch_amd as kt
    from kubetorch_amd.client.fn import Fn

    seen = {}

    def fake_call(self, args, kwargs, method=None, **opts):
        seen.update(opts)
        return "ok"

    monkeypatch.setattr(Fn, "_call", fake_call)
    f = Fn({"name": "x", "file_path": "", "rel_path": "",
            "project_root": "."})
    f(1, kt_config=[kt.LoggingConfig(stream_logs=False),
                    kt.MetricsConfig()],
      kt_stream_metrics=False)  # flat kwarg beats the bundle
    assert seen["stream_logs"] is False
    assert seen["stream_metrics"] is False


def test_fn_remote_dir_mode():
    """kt.fn(remote_dir=..., remote_import_path=...) dispatches to code
    already present on the pod image — no client sync (reference:
    fn(remote_dir/remote_import_path)). Simulated by pointing remote_dir
    at a dir that exists on the 'image' (this host)."""
    import os

    import kubetorch_amd as kt

    assets = os.path.join(os.path.dirname(__file__), "assets", "summer")
    f = kt.fn(remote_dir=assets, remote_import_path="summer:summer",
              name="baked")
    assert f.pointers["file_path"] == os.path.join(assets, "summer.py")
    f.to(kt.Compute(cpus=1))
    try:
        assert f(20, 22) == 42
    finally:
        f.teardown()


def test_fn_remote_dir_requires_import_path():
    import pytest as _pytest

    import kubetorch_amd as kt

    with _pytest.raises(ValueError, match="remote_import_path"):
        kt.fn(remote_dir="/app")
    with _pytest.raises(TypeError):
        kt.fn()


def test_cls_remote_dir_mode():
    import os

    import kubetorch_amd as kt

    assets = os.path.join(os.path.dirname(__file__), "assets", "summer")
    c = kt.cls(remote_dir=assets, remote_import_path="summer:Counter",
               init_args={"start": 5}, name="baked-counter")
    c.to(kt.Compute(cpus=1))
    try:
        assert c.add(3) == 8
    finally:
        c.teardown()


def test_fn_sync_dir_override():
    """sync_dir narrows the synced project root to a chosen directory
    (reference: Module sync_dir); the fn's file must live under it."""
    import os

    import kubetorch_amd as kt
    from tests.assets.summer import summer as summer_mod

    assets = os.path.join(os.path.dirname(__file__), "assets", "summer")
    f = kt.fn(summer_mod.summer, sync_dir=assets)
    assert f.pointers["project_root"] == os.path.abspath(assets)
    assert f.pointers["rel_path"] == "summer.py"
    with pytest.raises(ValueError, match="not under sync_dir"):
        kt.fn(summer_mod.summer, sync_dir="/nonexistent-root")
    with pytest.raises(ValueError, match="mutually exclusive"):
        kt.fn(summer_mod.summer, sync_dir=assets, remote_dir="/app")
