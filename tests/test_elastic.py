"""Elastic fault handling (BASELINE config 4 analog, CPU/local driver):
kill 1 of 2 worker pods mid-call -> WorkerMembershipChanged aborts the
in-flight distributed call; re-deploy re-provisions the pod and the next
call re-forms the process group (per-call rendezvous = elastic re-join)."""
import os
import sys
import threading
import time

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "summer"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "faulttest"

import kubetorch_amd as kt  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]
from kubetorch_amd.exceptions import WorkerMembershipChanged  # noqa: E402
from tests.assets.summer import summer as summer_mod  # noqa: E402


@pytest.mark.timeout(300)
def test_kill_pod_mid_call_then_rejoin():
    f = kt.fn(summer_mod.slow_echo).to(
        kt.Compute(cpus=1).distribute("pytorch", workers=2, num_proc=1,
                                      quorum_timeout=30))
    try:
        # healthy distributed call
        assert f(1, delay=0) == [1, 1]

        # start a long call, then kill the non-coordinator pod
        result = {}

        def call():
            try:
                result["value"] = f(2, delay=30, kt_timeout=120)
            except BaseException as e:  # noqa: BLE001
                result["error"] = e

        t = threading.Thread(target=call)
        t.start()
        time.sleep(2.0)
        from kubetorch_amd.controller.app import HUB

        pods = HUB.driver.services[("default", f.name)]
        killed = pods[1]
        killed.kill()
        t.join(90)
        assert not t.is_alive(), "call did not abort after pod death"
        assert "error" in result, f"expected abort, got {result}"
        err = result["error"]
        assert isinstance(err, WorkerMembershipChanged), repr(err)

        # AUTOMATIC re-provision: the controller's pod monitor respawns the
        # dead pod (no client re-deploy) and the next call's rendezvous
        # re-forms the group with the replacement (elastic re-join)
        deadline = time.time() + 60
        while time.time() < deadline:
            alive = HUB.driver.pods(f.name, "default")
            if len(alive) == 2 and killed.host not in alive:
                break
            time.sleep(0.5)
        else:
            raise AssertionError(
                f"pod monitor did not respawn: {HUB.driver.pods(f.name, 'default')}")
        # respawned pod must be ready before it can serve the fan-out
        assert f(3, delay=0, kt_restart_procs=True, kt_timeout=120) == [3, 3]
        evs = HUB.driver.events[("default", f.name)]
        assert any(e["reason"] == "Respawned" for e in evs)
    finally:
        f.teardown()


def test_peers_rendezvous_rank_order_and_generation(monkeypatch):
    """PeersRendezvous concludes rank/world from the controller's live peer
    list: sorted-host rank order, master = rank 0's host, fresh port per
    generation, and a clear error when self is missing (we were replaced)."""
    from kubetorch_amd.parallel.elastic import PeersRendezvous
    from kubetorch_amd.serving import discovery, supervisors

    peers = ["10.0.0.3:32300", "10.0.0.1:32300", "10.0.0.2:32300"]
    monkeypatch.setattr(discovery, "current_peers",
                        lambda *a, **k: list(peers))
    monkeypatch.setattr(supervisors, "_self_host", lambda: "10.0.0.2:32300")

    rdv = PeersRendezvous(service_name="svc", settle=0.3, quorum_timeout=10)
    rank, world, addr, port = rdv()
    assert (rank, world) == (1, 3)        # sorted order, me second
    assert addr == "10.0.0.1"             # master = rank 0's host
    rank2, world2, addr2, port2 = rdv()
    assert port2 == port + 1              # fresh port per generation

    # a shrunk group re-ranks the survivors
    peers[:] = ["10.0.0.3:32300", "10.0.0.2:32300"]
    rank3, world3, addr3, _ = rdv()
    assert (rank3, world3) == (0, 2)
    assert addr3 == "10.0.0.2"

    # self evicted from the peer list -> loud error, not a silent hang
    monkeypatch.setattr(supervisors, "_self_host", lambda: "10.9.9.9:32300")
    with pytest.raises(RuntimeError, match="not in peer list"):
        rdv()


def test_peers_rendezvous_waits_for_settle(monkeypatch):
    """A flapping peer list (pod still being respawned) delays conclusion
    until the membership is stable for `settle` seconds."""
    import itertools

    from kubetorch_amd.parallel.elastic import PeersRendezvous
    from kubetorch_amd.serving import discovery, supervisors

    t0 = time.time()
    flap_until = t0 + 1.0

    def peers(*a, **k):
        if time.time() < flap_until:  # membership still churning
            return ["10.0.0.1:1", f"10.0.0.{int(time.time()*10)%5+2}:1"]
        return ["10.0.0.1:1", "10.0.0.2:1"]

    monkeypatch.setattr(discovery, "current_peers", peers)
    monkeypatch.setattr(supervisors, "_self_host", lambda: "10.0.0.1:1")
    rdv = PeersRendezvous(service_name="svc", settle=0.5, quorum_timeout=20)
    rank, world, _, _ = rdv()
    assert (rank, world) == (0, 2)
    assert time.time() - t0 >= 1.4  # flap window + settle, not instant
