"""GPU end-to-end through the FULL dispatch stack on one MI355X:
client -> controller -> local-driver pods -> supervisor -> worker process
-> HIP kernels (BASELINE config 2 analog), plus the RL trainer->inference
CUDA weight sync through the pod-data-server (config 5 analog)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "gpu_train"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "gputest"

import kubetorch_amd as kt  # noqa: E402

pytestmark = pytest.mark.flaky_retry


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_dispatched_gpu_training():
    from tests.assets.gpu_train import gpu_train

    f = kt.fn(gpu_train.train_tiny_llama).to(kt.Compute(gpus=1))
    try:
        losses = f(3, kt_timeout=420)
        assert len(losses) == 3
        assert losses[-1] < losses[0], losses
    finally:
        f.teardown()


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_rl_cuda_weight_sync_between_pods():
    from tests.assets.gpu_train import gpu_train

    trainer = kt.cls(gpu_train.CudaTrainer).to(kt.Compute(gpus=1))
    infer = kt.cls(gpu_train.CudaInference).to(kt.Compute(gpus=1))
    try:
        s1 = trainer.publish(kt_timeout=300)
        got = infer.pull(kt_timeout=300)["sum"]
        assert abs(got - s1) < abs(s1) * 1e-2 + 1.0, (got, s1)
        s2 = trainer.publish()  # weights changed; zero-copy republish
        got2 = infer.pull()["sum"]
        assert abs(got2 - s2) < abs(s2) * 1e-2 + 1.0, (got2, s2)
        assert got2 != got
    finally:
        trainer.teardown()
        infer.teardown()


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_rl_two_services_with_autoscaled_inference():
    """BASELINE config 5 composite on one box: trainer publishes CUDA
    weights through the pod-data-server; the INFERENCE service is
    autoscaled (KPA loop on in-flight calls) and scales 1->2 under
    concurrent pull load, with every replica pulling correct weights."""
    import threading
    import time as _t

    from tests.assets.gpu_train import gpu_train

    os.environ["KT_AUTOSCALER_INTERVAL"] = "0.5"
    trainer = kt.cls(gpu_train.CudaTrainer).to(kt.Compute(gpus=1))
    infer = kt.cls(gpu_train.CudaInference).to(
        kt.Compute(gpus=1).autoscale(target=1, min_scale=1, max_scale=2,
                                     scale_down_delay="60s"))
    from kubetorch_amd.controller.app import HUB

    try:
        s1 = trainer.publish(kt_timeout=300)
        first = infer.pull(kt_timeout=300)
        assert abs(first["sum"] - s1) < abs(s1) * 1e-2 + 1.0

        results = []

        def load(i):
            results.append(infer.pull(delay=20, kt_timeout=240))

        threads = [threading.Thread(target=load, args=(i,)) for i in range(2)]
        for t in threads:
            t.start()
        deadline = _t.time() + 45
        n = 0
        while _t.time() < deadline:
            n = len(HUB.driver.pods(infer.name, "default"))
            if n >= 2:
                break
            _t.sleep(0.5)
        if n < 2:  # diagnostics before failing
            import httpx

            w = HUB.workloads.get(("default", infer.name))
            from kubetorch_amd.controller.app import _autoscale_spec

            print("DIAG manifest kind:", (w or {}).get("manifest", {}).get("kind"))
            print("DIAG autoscale spec:", _autoscale_spec((w or {}).get("manifest", {})))
            print("DIAG desired:", (w or {}).get("desired_replicas"))
            for p in HUB.driver.pods(infer.name, "default"):
                try:
                    r = httpx.get(f"http://{p}/metrics", timeout=3)
                    act = [l for l in r.text.splitlines()
                           if l.startswith("kt_active_requests")]
                    print("DIAG", p, act)
                except Exception as e:
                    print("DIAG", p, "metrics err", e)
            evs = HUB.driver.events.get(("default", infer.name), [])
            print("DIAG events:", [(e["reason"], e["message"]) for e in evs][-6:])
        assert n >= 2, f"inference did not scale up (pods={n})"
        for t in threads:
            t.join(240)
        assert len(results) == 2
        for r in results:
            assert abs(r["sum"] - s1) < abs(s1) * 1e-2 + 1.0, (r, s1)
        # calls spread across replicas once the pod cache refreshes
        _t.sleep(2.5)
        pods_seen = {infer.pull(kt_timeout=120)["pod"] for _ in range(6)}
        assert len(pods_seen) >= 2, pods_seen
    finally:
        trainer.teardown()
        infer.teardown()
