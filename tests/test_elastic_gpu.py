"""GPU elastic drill (BASELINE config 4 on a 1-GPU box): kill 1 of 2
worker pods mid-distributed-call on MI355X hardware, controller auto
re-provisions, next call re-forms the group; plus the RCCL group
destroy/reinit lifecycle with device tensors (the 8-GPU multi-rank RCCL
variant needs the driver's multi-GPU node — collectives here run gloo
because two ranks cannot share one GPU under RCCL)."""
import os
import sys
import threading
import time

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "gpu_train"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "gpufault"

import kubetorch_amd as kt  # noqa: E402
from kubetorch_amd.exceptions import WorkerMembershipChanged  # noqa: E402

pytestmark = pytest.mark.flaky_retry


@pytest.mark.gpu
@pytest.mark.timeout(420)
def test_gpu_kill_pod_mid_step_auto_rejoin():
    from tests.assets.gpu_train import gpu_train

    f = kt.fn(gpu_train.elastic_gpu_step).to(
        kt.Compute(gpus=1).distribute("pytorch", workers=2, num_proc=1,
                                      quorum_timeout=60))
    try:
        healthy = f(1.0, kt_timeout=180)
        assert [r["world"] for r in healthy] == [2, 2]
        assert all(abs(r["sum"] - 2.0) < 1e-3 for r in healthy), healthy

        result = {}

        def call():
            try:
                result["value"] = f(2.0, delay=30, kt_timeout=120)
            except BaseException as e:  # noqa: BLE001
                result["error"] = e

        t = threading.Thread(target=call)
        t.start()
        time.sleep(3.0)
        from kubetorch_amd.controller.app import HUB

        pods = HUB.driver.services[("default", f.name)]
        killed = pods[1]
        killed.kill()
        t.join(90)
        assert not t.is_alive(), "call did not abort after pod death"
        assert isinstance(result.get("error"), WorkerMembershipChanged), result

        deadline = time.time() + 90
        while time.time() < deadline:
            alive = HUB.driver.pods(f.name, "default")
            if len(alive) == 2 and killed.host not in alive:
                break
            time.sleep(0.5)
        else:
            raise AssertionError(f"no auto-respawn: {alive}")
        rejoined = f(3.0, kt_restart_procs=True, kt_timeout=180)
        assert [r["world"] for r in rejoined] == [2, 2]
        assert all(abs(r["sum"] - 6.0) < 1e-3 for r in rejoined), rejoined
    finally:
        f.teardown()


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_rccl_group_destroy_reinit_cycle():
    """RCCL (nccl backend) group lifecycle on gfx950 with device tensors:
    init -> all_reduce -> barrier -> destroy, three times in one process —
    the primitive the per-call elastic model and the data plane's
    per-transfer groups depend on."""
    from tests.assets.gpu_train import gpu_train

    f = kt.fn(gpu_train.rccl_group_lifecycle).to(kt.Compute(gpus=1))
    try:
        res = f(3, kt_timeout=240)
        assert res == [1024.0, 2048.0, 3072.0], res
    finally:
        f.teardown()
