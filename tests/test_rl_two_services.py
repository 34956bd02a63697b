"""RL-style two-service job (BASELINE config 5 analog, local driver):
trainer service publishes policy weights into the tensor store; a separate
inference service pulls them (the trainer->inference weight-sync path)."""
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "rl_services"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "rltest"

import kubetorch_amd as kt  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]
from tests.assets.rl_services import rl_services  # noqa: E402


@pytest.mark.timeout(300)
def test_trainer_to_inference_weight_sync(tmp_path):
    os.environ["KT_STORE_ROOT"] = str(tmp_path / "store")
    trainer = kt.cls(rl_services.Trainer).to(kt.Compute(cpus=1))
    infer = kt.cls(rl_services.InferenceServer).to(kt.Compute(cpus=1))
    try:
        before = infer.weight_sum()
        out = trainer.train_step(4)
        assert out["version"] == 1
        infer.sync_weights()
        after = infer.weight_sum()
        assert after != before
        assert abs(after - trainer.weight_sum()) < 1e-4
        action = infer.act([0.0] * 8)
        assert len(action) == 4
    finally:
        trainer.teardown()
        infer.teardown()
