"""Flagship benchmark: Llama-3-8B DDP bf16 tokens/sec on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
N>1 is launched via torch.distributed.run (one rank per GPU, RCCL over
xGMI); rank/world info comes from the env. Rank 0 prints ONE JSON line.

Measures the BASELINE.json north-star metric: Llama-3-8B DDP bf16
tokens/sec through kubetorch_amd's training engine (FlatDDP flat-bucket
all-reduce + fused HIP AdamW + gfx950 fused ops), synthetic data,
random-init weights, weak scaling (fixed per-GPU batch). The same loop
is importable (kubetorch_amd/models/benchmark.py) and deployable through
the product API (tests/bench_launcher.py).
"""
import argparse
import json


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

    from kubetorch_amd.models.benchmark import run_training_benchmark

    result = run_training_benchmark(
        steps=args.steps, warmup=args.warmup, batch=args.batch, seq=args.seq,
        model=args.model, bucket_mb=args.bucket_mb, zero=args.zero,
        lr=args.lr, sdpa=args.sdpa, ckpt=args.ckpt, profile=args.profile,
        expected_gpus=args.gpus)
    if result is not None:
        print(json.dumps(result), flush=True)

    import torch.distributed as dist

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
