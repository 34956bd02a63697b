"""Flagship benchmark: Llama-3-8B DDP bf16 tokens/sec on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
N>1 is launched via torch.distributed.run (one rank per GPU, RCCL over
xGMI); rank/world info comes from the env. Rank 0 prints ONE JSON line.

Measures the BASELINE.json north-star metric: Llama-3-8B DDP bf16
tokens/sec through kubetorch_amd's training engine (FlatDDP flat-bucket
all-reduce + fused HIP AdamW + gfx950 fused ops), synthetic data,
random-init weights, weak scaling (fixed per-GPU batch).
"""
import argparse
import json
import sys
import time

import torch


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=4, help="per-GPU micro batch")
    ap.add_argument("--seq", type=int, default=4096)
    ap.add_argument("--model", type=str, default="llama3-8b",
                    choices=["llama3-8b", "tiny"])
    ap.add_argument("--bucket-mb", type=int, default=256)
    ap.add_argument("--zero", action="store_true",
                    help="ZeRO-1 optimizer-state sharding")
    ap.add_argument("--lr", type=float, default=1e-4)
    # AOTriton's "efficient" kernels beat its flash kernels on gfx950 by
    # ~14% end-to-end (profiles/r01_step_profile.md) -> default efficient.
    ap.add_argument("--sdpa", type=str, default="efficient",
                    choices=["flash", "efficient", "math"])
    ap.add_argument("--ckpt", action="store_true",
                    help="per-layer activation checkpointing (memory for "
                         "compute; off for the headline number)")
    ap.add_argument("--profile", action="store_true",
                    help="profile one post-warmup step (top-kernel table "
                         "to stderr; excluded from the timed region)")
    args = ap.parse_args()

    from kubetorch_amd.models import Llama, llama3_8b, llama_tiny
    from kubetorch_amd.parallel import FlatDDP, init_distributed

    rank, world, local_rank = init_distributed()
    if world != args.gpus:
        log(f"[bench] note: WORLD_SIZE={world} != --gpus {args.gpus}; "
            f"reporting actual world size {world}")
    if torch.cuda.is_available():
        dev = torch.device("cuda", local_rank % torch.cuda.device_count())
    else:
        dev = torch.device("cpu")

    if args.model == "llama3-8b":
        cfg = llama3_8b(max_seq_len=args.seq)
    else:
        cfg = llama_tiny(max_seq_len=max(256, args.seq))

    torch.manual_seed(1234)
    log(f"[bench rank{rank}/{world}] building {args.model} on {dev} ...")
    t_build = time.time()
    prev_dtype = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(dev):
            model = Llama(cfg)
    finally:
        torch.set_default_dtype(prev_dtype)
    if args.ckpt:
        model.gradient_checkpointing_enable()
    engine = FlatDDP(model, lr=args.lr, bucket_mb=args.bucket_mb,
                     zero=args.zero)
    engine.broadcast_params(src=0)
    log(f"[bench rank{rank}] model+engine ready in {time.time()-t_build:.1f}s "
        f"({sum(p.numel() for p in model.parameters())/1e9:.2f}B params)")

    B, S = args.batch, args.seq
    gen = torch.Generator(device="cpu").manual_seed(4321 + rank)
    tokens = torch.randint(0, cfg.vocab_size, (B, S + 1), generator=gen).to(dev)
    x, y = tokens[:, :-1].contiguous(), tokens[:, 1:].contiguous()

    from contextlib import nullcontext

    if dev.type == "cuda":
        from torch.nn.attention import SDPBackend, sdpa_kernel

        backend = {"flash": SDPBackend.FLASH_ATTENTION,
                   "efficient": SDPBackend.EFFICIENT_ATTENTION,
                   "math": SDPBackend.MATH}[args.sdpa]

        def sdpa_ctx():
            return sdpa_kernel(backend)
    else:
        def sdpa_ctx():
            return nullcontext()

    def one_step():
        with sdpa_ctx():
            loss = model.loss(x, y)
        loss.backward()
        engine.step()
        return loss

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    for i in range(args.warmup):
        loss = one_step()
        log(f"[bench rank{rank}] warmup {i}: loss={loss.item():.4f}")

    if args.profile:
        # every rank takes the profiled step (collectives stay matched);
        # only rank 0 prints the table
        from kubetorch_amd.utils.profiling import profile_step

        printer = log if rank == 0 else (lambda *_: None)
        with profile_step("bench_step", top=20, printer=printer):
            one_step()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
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
        log(f"[bench rank{rank}] peak HBM: "
            f"{torch.cuda.max_memory_allocated(dev)/2**30:.1f} GiB")

    tokens_per_step = B * S * world
    toks_per_sec = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        # achieved model FLOP/s (6*N per token fwd+bwd + causal attention)
        n_params = sum(p.numel() for p in model.parameters())
        att = 12 * cfg.n_layers * cfg.dim * S * 0.5  # per token, causal
        flops_per_tok = 6 * n_params + 3 * att  # bwd ~2x fwd attention
        tf = toks_per_sec * flops_per_tok / 1e12
        log(f"[bench] ~{tf:.0f} TFLOP/s model FLOPs "
            f"({100 * tf / 2500:.0f}% of 2.5 PF dense bf16 peak)")

    if rank == 0:
        result = {
            "metric": "llama3_8b_ddp_bf16_tokens_per_sec" if args.model == "llama3-8b"
                      else "tiny_ddp_tokens_per_sec",
            "value": round(toks_per_sec, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": B * world,
                "seq_len": S,
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(result), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
