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
        got = infer.pull(kt_timeout=300)
        assert abs(got - s1) < abs(s1) * 1e-2 + 1.0, (got, s1)
        s2 = trainer.publish()  # weights changed; zero-copy republish
        got2 = infer.pull()
        assert abs(got2 - s2) < abs(s2) * 1e-2 + 1.0, (got2, s2)
        assert got2 != got
    finally:
        trainer.teardown()
        infer.teardown()
