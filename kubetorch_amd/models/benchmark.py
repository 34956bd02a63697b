"""The flagship training benchmark as an importable function, so the SAME
loop runs both from `bench.py` (driver contract) and from a worker process
launched through the product API (`kt.fn(run_training_benchmark).to(
Compute(gpus=N).distribute("pytorch", ...))` — BASELINE's "launched through
the API" wording; tests/bench_launcher.py).

Rank/world comes from the standard env contract either way; the SPMD
supervisor sets it for launched workers, torch.distributed.run for direct
bench.py runs.
"""
import os
import sys
import time

import torch


def _log(msg):
    print(msg, file=sys.stderr, flush=True)


def run_training_benchmark(steps=8, warmup=3, batch=4, seq=4096,
                           model="llama3-8b", bucket_mb=256, zero=False,
                           lr=1e-4, sdpa="efficient", ckpt=False,
                           profile=False, expected_gpus=None):
    """One full benchmark run on this rank. Returns the result dict on
    rank 0, None on other ranks. Weak scaling: per-GPU batch is fixed."""
    from kubetorch_amd.models import Llama, llama3_8b, llama_tiny
    from kubetorch_amd.parallel import FlatDDP, init_distributed

    rank, world, local_rank = init_distributed()
    if expected_gpus is not None and world != expected_gpus:
        _log(f"[bench] note: WORLD_SIZE={world} != --gpus {expected_gpus}; "
             f"reporting actual world size {world}")
    if torch.cuda.is_available():
        dev = torch.device("cuda", local_rank % torch.cuda.device_count())
    else:
        dev = torch.device("cpu")

    if model == "llama3-8b":
        cfg = llama3_8b(max_seq_len=seq)
    else:
        cfg = llama_tiny(max_seq_len=max(256, seq))

    torch.manual_seed(1234)
    _log(f"[bench rank{rank}/{world}] building {model} on {dev} ...")
    t_build = time.time()
    prev_dtype = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(dev):
            net = Llama(cfg)
    finally:
        torch.set_default_dtype(prev_dtype)
    if ckpt:
        net.gradient_checkpointing_enable()
    engine = FlatDDP(net, lr=lr, bucket_mb=bucket_mb, zero=zero)
    engine.broadcast_params(src=0)
    _log(f"[bench rank{rank}] model+engine ready in {time.time()-t_build:.1f}s "
         f"({sum(p.numel() for p in net.parameters())/1e9:.2f}B params)")

    B, S = batch, seq
    gen = torch.Generator(device="cpu").manual_seed(4321 + rank)
    tokens = torch.randint(0, cfg.vocab_size, (B, S + 1), generator=gen).to(dev)
    x, y = tokens[:, :-1].contiguous(), tokens[:, 1:].contiguous()

    from contextlib import nullcontext

    if dev.type == "cuda":
        from torch.nn.attention import SDPBackend, sdpa_kernel

        backend = {"flash": SDPBackend.FLASH_ATTENTION,
                   "efficient": SDPBackend.EFFICIENT_ATTENTION,
                   "math": SDPBackend.MATH}[sdpa]

        def sdpa_ctx():
            return sdpa_kernel(backend)
    else:
        def sdpa_ctx():
            return nullcontext()

    def one_step():
        with sdpa_ctx():
            loss = net.loss(x, y)
        loss.backward()
        engine.step()
        return loss

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    for i in range(warmup):
        loss = one_step()
        _log(f"[bench rank{rank}] warmup {i}: loss={loss.item():.4f}")

    if profile:
        # every rank takes the profiled step (collectives stay matched);
        # only rank 0 prints the table
        from kubetorch_amd.utils.profiling import profile_step

        printer = _log if rank == 0 else (lambda *_: None)
        with profile_step("bench_step", top=20, printer=printer):
            one_step()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        one_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=dev if dev.type == "cuda" else "cpu")
    if world > 1:
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(t.item())

    if dev.type == "cuda":
        _log(f"[bench rank{rank}] peak HBM: "
             f"{torch.cuda.max_memory_allocated(dev)/2**30:.1f} GiB")

    tokens_per_step = B * S * world
    toks_per_sec = tokens_per_step * steps / elapsed
    ms_per_step = elapsed / steps * 1000.0

    if rank != 0:
        return None

    # achieved model FLOP/s (6*N per token fwd+bwd + causal attention)
    n_params = sum(p.numel() for p in net.parameters())
    att = 12 * cfg.n_layers * cfg.dim * S * 0.5  # per token, causal
    flops_per_tok = 6 * n_params + 3 * att  # bwd ~2x fwd attention
    tf = toks_per_sec * flops_per_tok / 1e12
    _log(f"[bench] ~{tf:.0f} TFLOP/s model FLOPs "
         f"({100 * tf / 2500:.0f}% of 2.5 PF dense bf16 peak)")

    return {
        "metric": "llama3_8b_ddp_bf16_tokens_per_sec" if model == "llama3-8b"
                  else "tiny_ddp_tokens_per_sec",
        "value": round(toks_per_sec, 2),
        "unit": "tokens/s",
        "n_gpus": world,
        "steps": steps,
        "warmup": warmup,
        "ms_per_step": round(ms_per_step, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": model,
            "global_batch": B * world,
            "seq_len": S,
            "parallelism": f"dp{world}",
        },
    }


def bench_entry(steps=8, warmup=3, batch=4, seq=4096, model="llama3-8b",
                **kw):
    """Launcher-deployable entrypoint: runs the benchmark on every rank of
    the SPMD fan-out; rank 0's result dict comes back through the call
    aggregation (other ranks return None)."""
    res = run_training_benchmark(steps=steps, warmup=warmup, batch=batch,
                                 seq=seq, model=model, **kw)
    # leave the process group up for subsequent calls; the supervisor's
    # framework_cleanup destroys it on reload/teardown
    return res
