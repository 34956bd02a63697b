"""Multi-process FlatDDP test on gloo, world_size=2 (CPU).

Verifies the distributed path is correct by construction: two ranks with
different data shards must produce identical post-step params, equal to a
single-process run on the combined batch (grad averaging).
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

# spawn-based gloo rendezvous is sensitive to CI host load: retry once
pytestmark = pytest.mark.flaky_retry

from kubetorch_amd.models import Llama, llama_tiny
from kubetorch_amd.parallel import FlatDDP

def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _make(seed=0):
    torch.manual_seed(seed)
    cfg = llama_tiny(n_layers=1, dim=128, intermediate=256, vocab_size=256,
                     n_heads=4, n_kv_heads=2)
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    try:
        model = Llama(cfg)
    finally:
        torch.set_default_dtype(prev)
    return model, cfg


def _data(cfg):
    g = torch.Generator().manual_seed(7)
    x = torch.randint(0, cfg.vocab_size, (4, 32), generator=g)
    y = torch.randint(0, cfg.vocab_size, (4, 32), generator=g)
    return x, y


def _worker(rank, world, q, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model, cfg = _make()
        eng = FlatDDP(model, lr=1e-2, bucket_mb=1)
        eng.broadcast_params(src=0)
        x, y = _data(cfg)
        # shard the batch across ranks
        xs = x[rank * 2:(rank + 1) * 2]
        ys = y[rank * 2:(rank + 1) * 2]
        for _ in range(2):
            loss = model.loss(xs, ys)
            loss.backward()
            eng.step()
        flat = torch.cat([b.flat_param.float() for b in eng.buckets])
        q.put((rank, flat))
    finally:
        dist.destroy_process_group()


def test_flatddp_two_ranks_match_single_process():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_worker, args=(r, 2, q, port)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, flat = q.get()
        results[rank] = flat
    for p in procs:
        p.join(120)
        assert p.exitcode == 0

    # ranks must agree exactly (same reduced grads, same update)
    torch.testing.assert_close(results[0], results[1], rtol=0, atol=0)

    # and match a single-process run over the full batch
    model, cfg = _make()
    eng = FlatDDP(model, lr=1e-2, bucket_mb=1)
    x, y = _data(cfg)
    for _ in range(2):
        # average of shard losses == mean over full batch here since shards
        # are equal-sized; emulate grad averaging by running both shards
        # and averaging grads manually
        l0 = model.loss(x[:2], y[:2])
        l0.backward()
        g0 = [b.flat_grad.clone() for b in eng.buckets]
        eng.zero_grad()
        l1 = model.loss(x[2:], y[2:])
        l1.backward()
        for b, g in zip(eng.buckets, g0):
            b.flat_grad.add_(g)
        # emulate SUM all-reduce then 1/world scale inside step
        eng._grad_scale = 0.5
        eng.step()
        eng._grad_scale = 1.0
    single = torch.cat([b.flat_param.float() for b in eng.buckets])
    torch.testing.assert_close(results[0], single, rtol=2e-2, atol=2e-2)


def test_grad_clipping_scales_update():
    """clip_norm: the engine must apply AdamW to clipped grads; with a huge
    clip it must match the unclipped engine exactly."""
    from kubetorch_amd.models import Llama, llama_tiny
    from kubetorch_amd.parallel import FlatDDP

    def build(clip):
        torch.manual_seed(0)
        m = Llama(llama_tiny())
        return m, FlatDDP(m, lr=1e-3, bucket_mb=4, clip_norm=clip)

    torch.manual_seed(1)
    x = torch.randint(0, 512, (2, 32))
    y = torch.randint(0, 512, (2, 32))

    m1, e1 = build(clip=1e9)       # effectively unclipped
    m2, e2 = build(clip=None)
    m3, e3 = build(clip=1e-3)      # aggressive clip
    for m, e in ((m1, e1), (m2, e2), (m3, e3)):
        loss = m.loss(x, y)
        loss.backward()
        e.step()
    # huge clip == no clip
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2, rtol=0, atol=0)
    # aggressive clip must actually change the update
    diff = sum((p1 - p3).abs().sum().item()
               for p1, p3 in zip(m1.parameters(), m3.parameters()))
    assert diff > 0
    # reported norm matches autograd's own global grad norm
    torch.manual_seed(0)
    m4 = Llama(llama_tiny())
    loss = m4.loss(x, y)
    loss.backward()
    ref_norm = torch.norm(torch.stack(
        [p.grad.float().norm() for p in m4.parameters()])).item()
    assert abs(e3.last_grad_norm - ref_norm) / ref_norm < 5e-2, (
        e3.last_grad_norm, ref_norm)


def test_lr_schedules():
    from kubetorch_amd.parallel import (constant_with_warmup, warmup_cosine,
                                        warmup_linear)

    # warmup ramps linearly, peak at base, decays to min
    assert warmup_cosine(0, 1.0, 10, 100) == pytest.approx(0.1)
    assert warmup_cosine(9, 1.0, 10, 100) == pytest.approx(1.0)
    assert warmup_cosine(10, 1.0, 10, 100) == pytest.approx(1.0)
    assert warmup_cosine(55, 1.0, 10, 100, min_lr=0.1) == pytest.approx(0.55)
    assert warmup_cosine(100, 1.0, 10, 100, min_lr=0.1) == pytest.approx(0.1)
    assert warmup_cosine(500, 1.0, 10, 100, min_lr=0.1) == pytest.approx(0.1)
    assert warmup_linear(55, 1.0, 10, 100) == pytest.approx(0.5)
    assert constant_with_warmup(999, 3e-4, 10) == pytest.approx(3e-4)
    # monotone decay after warmup
    vals = [warmup_cosine(s, 1.0, 10, 100) for s in range(10, 101)]
    assert all(a >= b for a, b in zip(vals, vals[1:]))


def test_grad_accum_mean_semantics():
    """grad_accum_steps=A over the same micro-batch repeated A times must
    produce EXACTLY the single-step update (mean over micro-batches is
    folded into the fused optimizer's grad scale — no hidden LR*A)."""
    from kubetorch_amd.models import Llama, llama_tiny
    from kubetorch_amd.parallel import FlatDDP

    torch.manual_seed(3)
    x = torch.randint(0, 512, (2, 32))
    y = torch.randint(0, 512, (2, 32))

    torch.manual_seed(0)
    m1 = Llama(llama_tiny())
    e1 = FlatDDP(m1, lr=1e-3, bucket_mb=4)
    m1.loss(x, y).backward()
    e1.step()

    torch.manual_seed(0)
    m2 = Llama(llama_tiny())
    e2 = FlatDDP(m2, lr=1e-3, bucket_mb=4, grad_accum_steps=2)
    m2.loss(x, y).backward()
    m2.loss(x, y).backward()   # accumulates: bucket holds 2x the grad
    e2.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2, rtol=0, atol=0)


def _zero_worker(rank, world, q, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        model, cfg = _make()
        eng = FlatDDP(model, lr=1e-2, bucket_mb=1, zero=True)
        assert eng.zero
        # optimizer state is shard-sized
        for b in eng.buckets:
            assert b.m.numel() == b.numel // world
        eng.broadcast_params(src=0)
        x, y = _data(cfg)
        xs = x[rank * 2:(rank + 1) * 2]
        ys = y[rank * 2:(rank + 1) * 2]
        for _ in range(2):
            loss = model.loss(xs, ys)
            loss.backward()
            eng.step()
        flat = torch.cat([b.flat_param.float() for b in eng.buckets])
        q.put((rank, flat))
    finally:
        dist.destroy_process_group()


def test_zero1_two_ranks_match_full_optimizer():
    """ZeRO-1 (sharded AdamW + param all-gather) must produce the same
    params as plain FlatDDP on every rank."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = _free_port()
    procs = [ctx.Process(target=_zero_worker, args=(r, 2, q, port))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, flat = q.get()
        results[rank] = flat
    for p in procs:
        p.join(120)
        assert p.exitcode == 0
    torch.testing.assert_close(results[0], results[1], rtol=0, atol=0)

    # same update as the non-zero single-process emulation
    model, cfg = _make()
    eng = FlatDDP(model, lr=1e-2, bucket_mb=1)
    x, y = _data(cfg)
    for _ in range(2):
        l0 = model.loss(x[:2], y[:2])
        l0.backward()
        g0 = [b.flat_grad.clone() for b in eng.buckets]
        eng.zero_grad()
        l1 = model.loss(x[2:], y[2:])
        l1.backward()
        for b, g in zip(eng.buckets, g0):
            b.flat_grad.add_(g)
        eng._grad_scale = 0.5
        eng.step()
        eng._grad_scale = 1.0
    single = torch.cat([b.flat_param.float() for b in eng.buckets])
    # bucket padding differs (zero pads to world*8): compare per-param
    m2, _ = _make()
    e2 = FlatDDP(m2, lr=1e-2, bucket_mb=1, zero=True)  # world=1: zero off
    assert not e2.zero
    n = min(results[0].numel(), single.numel())
    torch.testing.assert_close(results[0][:n], single[:n], rtol=2e-2,
                               atol=2e-2)


def test_ema_tracking():
    """EMA buffers follow d*ema + (1-d)*param after every step; the state
    dict maps back to model parameter names/shapes."""
    from kubetorch_amd.models import Llama, llama_tiny
    from kubetorch_amd.parallel import FlatDDP

    torch.manual_seed(4)
    m = Llama(llama_tiny())
    e = FlatDDP(m, lr=1e-2, bucket_mb=4, ema_decay=0.5)
    x = torch.randint(0, 512, (2, 32))
    y = torch.randint(0, 512, (2, 32))

    m.loss(x, y).backward()
    e.step()
    p1 = {n: p.float().clone() for n, p in m.named_parameters()}
    ema1 = e.ema_state_dict()
    for n in p1:  # first step: ema == params
        torch.testing.assert_close(ema1[n], p1[n])

    m.loss(x, y).backward()
    e.step()
    p2 = {n: p.float().clone() for n, p in m.named_parameters()}
    ema2 = e.ema_state_dict()
    for n in p1:
        torch.testing.assert_close(ema2[n], 0.5 * p1[n] + 0.5 * p2[n],
                                   rtol=1e-5, atol=1e-6)
    assert ema2[n].shape == p2[n].shape


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_clip_and_accum_on_gpu_match_reference():
    """clip_norm + grad accumulation numerics on gfx950 (the fused HIP
    AdamW path) vs a plain torch fp32 AdamW reference with manual global
    clipping — world=1, the kernel-side semantics the gloo tests cover on
    CPU."""
    import torch

    torch.manual_seed(3)

    def build(dev, dtype):
        torch.manual_seed(3)
        return torch.nn.Sequential(
            torch.nn.Linear(64, 128), torch.nn.Tanh(),
            torch.nn.Linear(128, 16)).to(dev, dtype)

    xc = torch.randn(32, 64) * 7  # big grads -> clipping engages
    yc = torch.randn(32, 16)
    x, y = xc.to("cuda", torch.bfloat16), yc.to("cuda", torch.bfloat16)

    # engine: bf16 params + fused HIP AdamW (fp32 m/v) on gfx950
    m1 = build("cuda", torch.bfloat16)
    eng = FlatDDP(m1, lr=1e-2, betas=(0.9, 0.95), eps=1e-8, weight_decay=0.0,
                  bucket_mb=1, clip_norm=0.5, grad_accum_steps=2)
    # reference: fp32 autograd + torch AdamW + manual global clipping,
    # fed the engine's OWN bf16 mean grads each step so the comparison
    # isolates clip + fused-AdamW semantics from bf16 forward noise
    m2 = build("cpu", torch.float32)
    opt = torch.optim.AdamW(m2.parameters(), lr=1e-2, betas=(0.9, 0.95),
                            eps=1e-8, weight_decay=0.0)
    norms = []
    for step in range(4):
        for micro in range(2):  # backward twice; buckets accumulate
            torch.nn.functional.mse_loss(m1(x), y).backward()
        # engine grads (pre-step SUM over micro-batches) -> reference
        grads = {p: eng._param_view[p].float().cpu() / 2.0
                 for p in m1.parameters()}
        eng.step()
        norms.append(eng.last_grad_norm)
        opt.zero_grad()
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            p2.grad = grads[p1].clone()
        torch.nn.utils.clip_grad_norm_(m2.parameters(), 0.5)
        opt.step()
        # keep the fp32 reference walking the engine's bf16 trajectory
        for p1, p2 in zip(m1.parameters(), m2.parameters()):
            torch.testing.assert_close(p1.float().cpu(), p2,
                                       rtol=2e-2, atol=2e-2)
            p2.data.copy_(p1.detach().float().cpu())
    # clipping actually engaged at least once during the run
    assert norms and max(norms) > 0.5, norms
