"""Manifest/unit tests without a cluster (reference parity: test_compute.py,
test_autodown.py, test_kueue.py style — assert the generated YAML)."""
import os

import pytest

import kubetorch_amd as kt
from kubetorch_amd import constants as C
from kubetorch_amd.resources.autoscaling import AutoscalingConfig
from kubetorch_amd.resources.compute import Compute
from kubetorch_amd.resources.image import Image
from kubetorch_amd.resources.secret import Secret
from kubetorch_amd.resources.volume import Volume
from kubetorch_amd.provisioning import manifests as M


def test_gpu_deployment_manifest_amd():
    comp = Compute(gpus=8, gpu_type="MI355X", memory="64Gi", cpus=16,
                   inactivity_ttl="2h", queue="gpu-queue")
    m = comp.to_manifest("svc-a", username="alice", module="train")
    assert m["kind"] == "Deployment"
    cont = m["spec"]["template"]["spec"]["containers"][0]
    assert cont["resources"]["limits"]["amd.com/gpu"] == "8"
    assert "nvidia" not in str(m)
    sel = m["spec"]["template"]["spec"]["nodeSelector"]
    assert sel[C.GPU_PRODUCT_LABEL] == "MI355X"
    assert m["metadata"]["labels"][C.KUEUE_QUEUE_LABEL] == "gpu-queue"
    assert m["metadata"]["annotations"][C.INACTIVITY_TTL_ANNOTATION] == "2h"
    assert m["metadata"]["labels"][C.USERNAME_LABEL] == "alice"


def test_cpu_pod_gets_gpu_anti_affinity():
    m = Compute(cpus=2).to_manifest("svc-b")
    aff = m["spec"]["template"]["spec"]["affinity"]["nodeAffinity"]
    terms = aff["requiredDuringSchedulingIgnoredDuringExecution"]["nodeSelectorTerms"]
    keys = {e["key"] for t in terms for e in t["matchExpressions"]}
    assert any(k.startswith("amd.com/") for k in keys)


def test_knative_autoscale_manifest():
    comp = Compute(cpus=1).autoscale(target=10, metric="rps", min_scale=1,
                                     max_scale=8)
    m = comp.to_manifest("svc-c")
    assert m["apiVersion"].startswith("serving.knative.dev")
    ann = m["spec"]["template"]["metadata"]["annotations"]
    assert ann["autoscaling.knative.dev/metric"] == "rps"
    assert ann["autoscaling.knative.dev/max-scale"] == "8"


def test_autoscaling_hpa_class_for_cpu_metric():
    ann = AutoscalingConfig(metric="cpu").to_annotations()
    assert "hpa" in ann["autoscaling.knative.dev/class"]


def test_pytorchjob_manifest():
    comp = Compute(gpus=8)
    comp.kind = "pytorchjob"
    comp.distributed_config = {"type": "pytorch", "workers": 4, "num_proc": 8}
    m = comp.to_manifest("svc-d")
    assert m["kind"] == "PyTorchJob"
    specs = m["spec"]["pytorchReplicaSpecs"]
    assert specs["Master"]["replicas"] == 1
    assert specs["Worker"]["replicas"] == 3
    assert m["spec"]["nprocPerNode"] == "8"


def test_byo_manifest_roundtrip():
    m = M.build_pytorchjob_manifest("byo", "ns1", "img", workers=2, num_proc=4)
    comp = Compute.from_manifest(m)
    assert comp.kind == "pytorchjob"
    assert comp.distributed_config["workers"] == 2  # master + 1 worker
    assert comp.to_manifest("byo") == m


def test_service_manifests():
    svc, headless = M.build_service_manifests("svc-e", "default")
    assert headless["spec"]["clusterIP"] == "None"
    assert headless["metadata"]["name"] == "svc-e-headless"


def test_volume_and_secret_mounts():
    v = Volume("cache", size="50Gi", mount_path="/cache")
    s = Secret("hf", values={"HF_TOKEN": "x"})
    m = Compute(cpus=1, volumes=[v], secrets=[s]).to_manifest("svc-f")
    spec = m["spec"]["template"]["spec"]
    mounts = {mt["mountPath"] for mt in spec["containers"][0]["volumeMounts"]}
    assert "/cache" in mounts
    assert {"secretRef": {"name": "kt-secret-hf"}} in spec["containers"][0]["envFrom"]
    pvc = v.to_pvc_manifest("default")
    assert pvc["spec"]["resources"]["requests"]["storage"] == "50Gi"
    sm = s.to_manifest("default")
    assert sm["kind"] == "Secret" and "HF_TOKEN" in sm["data"]


def test_image_dockerfile_roundtrip():
    img = (Image("rocm/pytorch:latest").pip_install(["einops", "rich"])
           .set_env_vars({"A": "1"}).run_bash("echo hi"))
    text = img.contents()
    assert text.startswith("FROM rocm/pytorch:latest")
    img2 = Image.from_dockerfile(text)
    assert img2.image_id == "rocm/pytorch:latest"
    assert img2.steps == img.steps


def test_image_setup_interpreter_caching(tmp_path, monkeypatch):
    from kubetorch_amd.serving import image_setup

    monkeypatch.setattr(image_setup, "_CACHED_STEPS", [])
    marker = tmp_path / "m.txt"
    c1 = f"RUN echo one >> {marker}\nENV KT_TEST_XYZ=42"
    ran = image_setup.cached_image_setup(c1)
    assert ran == 2
    import os

    assert os.environ["KT_TEST_XYZ"] == "42"
    assert marker.read_text().count("one") == 1
    # unchanged contents -> nothing re-runs
    assert image_setup.cached_image_setup(c1) == 0
    assert marker.read_text().count("one") == 1
    # appended step -> only the new suffix runs
    c2 = c1 + f"\nRUN echo two >> {marker}"
    assert image_setup.cached_image_setup(c2) == 1
    assert marker.read_text().count("one") == 1


def test_pip_freeze_diff_skips_satisfied_installs():
    """Reference parity: pip-freeze diff — a pip install whose exact pins
    are already present is skipped; ranges/urls/options always run."""
    from importlib import metadata

    from kubetorch_amd.serving.image_setup import _pip_requirements_satisfied

    have = metadata.version("pytest")
    assert _pip_requirements_satisfied(f"pip install pytest=={have}")
    assert _pip_requirements_satisfied("python -m pip install pytest numpy")
    assert not _pip_requirements_satisfied("pip install not-a-real-pkg-xyz")
    assert not _pip_requirements_satisfied("pip install pytest==0.0.1")
    assert not _pip_requirements_satisfied("pip install pytest>=1.0")  # range
    assert not _pip_requirements_satisfied("pip install -r reqs.txt")
    assert not _pip_requirements_satisfied("pip install git+https://x/y.git")
    assert not _pip_requirements_satisfied("echo not pip at all")


def test_decorators_build_module():
    @kt.compute(cpus=1)
    @kt.distribute("pytorch", workers=2, num_proc=1)
    def train():
        return 1

    assert train() == 1  # still locally callable
    mod = train.build_module()
    assert mod.compute.distributed_config["workers"] == 2
    assert mod.pointers["name"] == "train"


def test_config_layering(tmp_path, monkeypatch):
    import yaml

    from kubetorch_amd.config import KTConfig

    monkeypatch.delenv("KT_USERNAME", raising=False)
    monkeypatch.delenv("KT_NAMESPACE", raising=False)
    p = tmp_path / "cfg.yaml"
    p.write_text(yaml.safe_dump({"namespace": "from-file", "username": "bob"}))
    cfg = KTConfig(path=str(p))
    assert cfg.namespace == "from-file"
    monkeypatch.setenv("KT_NAMESPACE", "from-env")
    assert cfg.namespace == "from-env"
    assert cfg.username == "bob"


def test_gpu_memory_and_disk_size():
    """gpu_memory: whole amd.com/gpu requested + gpu-memory annotation;
    disk_size -> ephemeral-storage resources (reference parity)."""
    c = Compute(gpu_memory="64Gi", disk_size="50Gi")
    assert c.gpus == 1  # whole GPU still requested
    m = c.to_manifest("svc", username="u")
    pod = m["spec"]["template"]["spec"]
    res = pod["containers"][0]["resources"]
    assert res["limits"]["amd.com/gpu"] == "1"
    assert res["requests"]["ephemeral-storage"] == "50Gi"
    assert m["metadata"]["annotations"]["gpu-memory"] == "64Gi"


def test_distribute_sets_replicas_and_spmd_env():
    c = Compute(cpus=1).distribute("pytorch", workers=3, num_proc=2)
    assert c.replicas == 3
    assert c.distributed_config["type"] == "pytorch"
    m = c.to_manifest("svc", username="u")
    assert m["spec"]["replicas"] == 3


def test_autoscale_bounds_in_manifest():
    c = Compute(cpus=1).autoscale(min_scale=1, max_scale=6, target=5,
                                  metric="concurrency")
    m = c.to_manifest("svc", username="u")
    ann = m["spec"]["template"]["metadata"]["annotations"]
    assert ann["autoscaling.knative.dev/min-scale"] == "1"
    assert ann["autoscaling.knative.dev/max-scale"] == "6"


def test_cluster_config_layer(tmp_path, monkeypatch):
    """Cluster-wide defaults from the controller sit BELOW env and the
    user's file (reference: kubetorch-config ConfigMap fetch)."""
    import json

    from kubetorch_amd.config import KTConfig

    # controller side: env-injected JSON beats the mounted file
    monkeypatch.setenv("KT_CLUSTER_CONFIG", json.dumps({"image": "rocm/x:1"}))
    from kubetorch_amd.controller.app import cluster_config

    assert cluster_config()["config"] == {"image": "rocm/x:1"}
    monkeypatch.delenv("KT_CLUSTER_CONFIG")
    cfgfile = tmp_path / "cc.yaml"
    cfgfile.write_text("image: rocm/y:2\nstream_logs: false\n")
    monkeypatch.setenv("KT_CLUSTER_CONFIG_PATH", str(cfgfile))
    assert cluster_config()["config"]["image"] == "rocm/y:2"

    # client side layering: cluster < file < env
    c = KTConfig(path=str(tmp_path / "none.yaml"))
    c._cluster = {"image": "cluster-img", "namespace": "team-ns"}
    assert c.get("image") == "cluster-img"
    assert c.get("namespace") == "team-ns"
    c._file["image"] = "file-img"
    assert c.get("image") == "file-img"
    monkeypatch.setenv("KT_IMAGE", "env-img")
    assert c.get("image") == "env-img"


def test_image_setup_env_and_run_and_cmd(monkeypatch, tmp_path):
    """run_step semantics: ENV expands $vars into os.environ, RUN executes
    with expansion and raises on failure, CMD manages the app process
    (old process terminated on re-CMD)."""
    import time

    from kubetorch_amd.serving import image_setup

    monkeypatch.setenv("KT_BASE", str(tmp_path))
    image_setup.run_step("ENV", "KT_SUB=$KT_BASE/sub")
    assert os.environ["KT_SUB"] == f"{tmp_path}/sub"

    image_setup.run_step("RUN", "mkdir -p $KT_SUB && echo hi > $KT_SUB/f")
    assert (tmp_path / "sub" / "f").read_text().strip() == "hi"

    with pytest.raises(RuntimeError, match="image step failed"):
        image_setup.run_step("RUN", "exit 3")

    state = {}
    image_setup.run_step("CMD", "sleep 30", app_state=state)
    first = state["app_proc"]
    assert first.poll() is None
    image_setup.run_step("CMD", "sleep 30", app_state=state)
    second = state["app_proc"]
    assert second is not first
    deadline = time.time() + 10
    while first.poll() is None and time.time() < deadline:
        time.sleep(0.1)
    assert first.poll() is not None, "old CMD process not terminated"
    second.terminate()
    second.wait(10)


def test_image_setup_copy_missing_target_is_nonfatal(capsys):
    from kubetorch_amd.serving import image_setup

    image_setup.run_step("COPY", "src /definitely/not/present")
    out = capsys.readouterr().out
    assert "not present" in out
