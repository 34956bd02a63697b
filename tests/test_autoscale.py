"""Behavioral autoscaling on the local driver (VERDICT r1 item 7 /
BASELINE config 5 analog): a KPA-style controller loop scales replicas on
summed in-flight requests; load 1->N, idle N->min. Reference model:
python_client/tests/test_autoscale.py (Knative in-cluster)."""
import os
import sys
import threading
import time

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "summer"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "scaletest"
os.environ["KT_AUTOSCALER_INTERVAL"] = "0.5"

import kubetorch_amd as kt  # noqa: E402
from tests.assets.summer import summer as summer_mod  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]


@pytest.mark.timeout(300)
def test_scale_up_under_load_then_down():
    f = kt.fn(summer_mod.slow_echo).to(
        kt.Compute(cpus=1).autoscale(target=1, min_scale=1, max_scale=3,
                                     window="10s", scale_down_delay="3s"))
    from kubetorch_amd.controller.app import HUB

    try:
        assert f(0, delay=0, kt_timeout=60) == 0  # warm, 1 pod
        assert len(HUB.driver.pods(f.name, "default")) == 1

        # 3 concurrent long calls -> in_flight 3 / target 1 -> 3 replicas
        threads = [threading.Thread(
            target=lambda i=i: f(i, delay=15, kt_timeout=120))
            for i in range(3)]
        for t in threads:
            t.start()
        deadline = time.time() + 30
        n = 0
        while time.time() < deadline:
            n = len(HUB.driver.pods(f.name, "default"))
            if n >= 3:
                break
            time.sleep(0.5)
        assert n >= 3, f"no scale-up observed (pods={n})"
        evs = HUB.driver.events[("default", f.name)]
        assert any(e["reason"] == "Autoscaled" for e in evs)
        for t in threads:
            t.join(120)

        # idle -> scale back down to min after the scale-down delay
        deadline = time.time() + 45
        while time.time() < deadline:
            n = len(HUB.driver.pods(f.name, "default"))
            if n == 1:
                break
            time.sleep(0.5)
        assert n == 1, f"no scale-down observed (pods={n})"
    finally:
        f.teardown()


@pytest.mark.timeout(240)
def test_load_balanced_calls_spread_over_pods():
    """Calls to an autoscaled service round-robin over live pods (the
    client-side stand-in for the K8s Service VIP)."""
    f = kt.fn(summer_mod.pod_name).to(
        kt.Compute(cpus=1).autoscale(target=1, min_scale=2, max_scale=2))
    from kubetorch_amd.controller.app import HUB

    try:
        deadline = time.time() + 30
        while time.time() < deadline:
            if len(HUB.driver.pods(f.name, "default")) >= 2:
                break
            time.sleep(0.5)
        time.sleep(2.5)  # let the client's pod cache expire
        seen = {f(kt_timeout=60) for _ in range(6)}
        assert len(seen) >= 2, f"calls pinned to one pod: {seen}"
    finally:
        f.teardown()


@pytest.mark.timeout(300)
def test_autoscaled_service_heals_dead_pod():
    """Pod death in an autoscaled service: the reconciliation loops
    (autoscaler owns desired replicas, monitor heals to them) cooperate —
    the pod comes back and calls keep succeeding."""
    f = kt.fn(summer_mod.pod_name).to(
        kt.Compute(cpus=1).autoscale(target=1, min_scale=2, max_scale=2,
                                     scale_down_delay="60s"))
    from kubetorch_amd.controller.app import HUB

    try:
        deadline = time.time() + 30
        while time.time() < deadline:
            if len(HUB.driver.pods(f.name, "default")) >= 2:
                break
            time.sleep(0.5)
        pods = HUB.driver.services[("default", f.name)]
        victim = pods[1]
        victim.kill()
        deadline = time.time() + 60
        healed = []
        while time.time() < deadline:
            healed = HUB.driver.pods(f.name, "default")
            if len(healed) == 2 and victim.host not in healed:
                break
            time.sleep(0.5)
        assert len(healed) == 2 and victim.host not in healed, healed
        time.sleep(2.5)  # lb cache refresh
        seen = {f(kt_timeout=60) for _ in range(6)}
        assert len(seen) >= 2, seen
    finally:
        f.teardown()
