"""In-step elastic re-join (parallel/elastic.py): 3 ranks train with
FlatDDP over gloo, rank 2 dies mid-run, the survivors detect the
collective failure, re-form a world-2 group via FileRendezvous and keep
stepping — no process restart, no checkpoint reload. Beyond-reference
capability (SURVEY §7 stage 4)."""
import json
import os
import socket
import sys

import pytest
import torch

pytestmark = pytest.mark.flaky_retry

STEPS_BEFORE_DEATH = 2
TOTAL_STEPS = 6


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, world, port, rdv_root, out_dir, use_cuda=False,
            die_rank=2):
    import datetime

    import torch.distributed as dist

    from kubetorch_amd.parallel import ElasticStepper, FileRendezvous, FlatDDP

    torch.manual_seed(7)  # identical init on every rank
    store = dist.TCPStore("127.0.0.1", port, world, rank == 0,
                          timeout=datetime.timedelta(seconds=60))
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=10))
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)

    # gloo cannot all-reduce HIP tensors, so the engine/PG stays on CPU;
    # in GPU mode each step still runs fwd+bwd matmuls on cuda:0 through a
    # fixed device-resident stage (autograd crosses devices)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    gpu_stage = (torch.randn(16, 16, device="cuda") / 4.0
                 if use_cuda else None)
    engine = FlatDDP(model, lr=1e-2, bucket_mb=1, overlap_optimizer=False)
    engine.broadcast_params(src=0)
    rdv = FileRendezvous(rdv_root, uid=f"u{rank}", port_base=port + 1000,
                         settle=1.0, timeout=45)
    stepper = ElasticStepper(engine, rdv, pg_timeout_s=10)

    gen = torch.Generator().manual_seed(99)  # same batch everywhere
    x = torch.randn(8, 16, generator=gen)
    y = torch.randn(8, 4, generator=gen)

    def fb():
        h = (x.to("cuda") @ gpu_stage).cpu() if gpu_stage is not None else x
        loss = torch.nn.functional.mse_loss(model(h), y)
        loss.backward()
        return loss

    losses = []
    for step in range(TOTAL_STEPS):
        if rank == die_rank and step == STEPS_BEFORE_DEATH:
            os._exit(17)  # simulated pod death, no cleanup
        losses.append(float(stepper.step(fb)))

    out = {
        "rank": rank,
        "losses": losses,
        "reforms": stepper.reforms,
        "final_world": int(os.environ["WORLD_SIZE"]),
        "params": [p.detach().float().cpu().numpy().tolist()
                   for p in model.parameters()],
        "steps": engine.step_count,
    }
    with open(os.path.join(out_dir, f"rank{rank}.json"), "w") as f:
        json.dump(out, f)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_instep_rejoin_after_rank_death(tmp_path):
    import torch.multiprocessing as mp

    world = 3
    port = _free_port()
    rdv_root = str(tmp_path / "rdv")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker,
                         args=(r, world, port, rdv_root, out_dir))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        if p.is_alive():
            p.terminate()
            raise AssertionError("worker hung")
    assert procs[2].exitcode == 17  # the killed rank
    assert procs[0].exitcode == 0 and procs[1].exitcode == 0, \
        [p.exitcode for p in procs]

    r0 = json.load(open(os.path.join(out_dir, "rank0.json")))
    r1 = json.load(open(os.path.join(out_dir, "rank1.json")))
    assert r0["reforms"] >= 1 and r1["reforms"] >= 1
    assert r0["final_world"] == 2 and r1["final_world"] == 2
    assert len(r0["losses"]) == TOTAL_STEPS
    # bitwise param agreement between survivors (post-reform broadcast +
    # identical reduced grads afterwards)
    for a, b in zip(r0["params"], r1["params"]):
        assert a == b
    # training progressed through the fault
    assert r0["losses"][-1] < r0["losses"][0]


def test_elastic_refuses_zero_engine():
    from kubetorch_amd.parallel import ElasticStepper

    class FakeEngine:
        zero = True

    with pytest.raises(ValueError, match="ZeRO"):
        ElasticStepper(FakeEngine(), lambda: None)


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_instep_rejoin_on_gpu(tmp_path):
    """Same drill on MI355X hardware: 2 ranks share cuda:0 (gloo
    collectives — one GPU cannot host two RCCL ranks), rank 1 dies, the
    survivor re-forms world=1 in-step and finishes training."""
    import torch.multiprocessing as mp

    world = 2
    port = _free_port()
    rdv_root = str(tmp_path / "rdv")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker,
                         args=(r, world, port, rdv_root, out_dir, True, 1))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        if p.is_alive():
            p.terminate()
            raise AssertionError("worker hung")
    assert procs[1].exitcode == 17
    assert procs[0].exitcode == 0, [p.exitcode for p in procs]
    r0 = json.load(open(os.path.join(out_dir, "rank0.json")))
    assert r0["reforms"] >= 1 and r0["final_world"] == 1
    assert len(r0["losses"]) == TOTAL_STEPS
    assert r0["losses"][-1] < r0["losses"][0]


@pytest.mark.timeout(300)
def test_two_sequential_rank_deaths(tmp_path):
    """4 ranks; rank 3 dies at step 2 and rank 2 at step 4 — survivors
    re-form twice (4 -> 3 -> 2) and finish."""
    import torch.multiprocessing as mp

    world = 4
    port = _free_port()
    rdv_root = str(tmp_path / "rdv")
    out_dir = str(tmp_path / "out")
    os.makedirs(out_dir)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker2deaths,
                         args=(r, world, port, rdv_root, out_dir))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        if p.is_alive():
            p.terminate()
            raise AssertionError("worker hung")
    assert procs[3].exitcode == 17 and procs[2].exitcode == 17
    assert procs[0].exitcode == 0 and procs[1].exitcode == 0, \
        [p.exitcode for p in procs]
    r0 = json.load(open(os.path.join(out_dir, "rank0.json")))
    r1 = json.load(open(os.path.join(out_dir, "rank1.json")))
    assert r0["reforms"] >= 2 and r0["final_world"] == 2
    assert len(r0["losses"]) == TOTAL_STEPS
    for a, b in zip(r0["params"], r1["params"]):
        assert a == b


def _worker2deaths(rank, world, port, rdv_root, out_dir):
    import datetime

    import torch.distributed as dist

    from kubetorch_amd.parallel import ElasticStepper, FileRendezvous, FlatDDP

    torch.manual_seed(7)
    store = dist.TCPStore("127.0.0.1", port, world, rank == 0,
                          timeout=datetime.timedelta(seconds=60))
    dist.init_process_group("gloo", store=store, rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=8))
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    model = torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))
    engine = FlatDDP(model, lr=1e-2, bucket_mb=1, overlap_optimizer=False)
    engine.broadcast_params(src=0)
    rdv = FileRendezvous(rdv_root, uid=f"u{rank}", port_base=port + 1000,
                         settle=1.0, timeout=45)
    stepper = ElasticStepper(engine, rdv, pg_timeout_s=8)
    gen = torch.Generator().manual_seed(99)
    x = torch.randn(8, 16, generator=gen)
    y = torch.randn(8, 4, generator=gen)

    def fb():
        loss = torch.nn.functional.mse_loss(model(x), y)
        loss.backward()
        return loss

    losses = []
    for step in range(TOTAL_STEPS):
        if rank == 3 and step == 2:
            os._exit(17)
        if rank == 2 and step == 4:
            os._exit(17)
        losses.append(float(stepper.step(fb)))
    out = {"rank": rank, "losses": losses, "reforms": stepper.reforms,
           "final_world": int(os.environ["WORLD_SIZE"]),
           "params": [p.detach().float().cpu().numpy().tolist()
                      for p in model.parameters()]}
    with open(os.path.join(out_dir, f"rank{rank}.json"), "w") as f:
        json.dump(out, f)
