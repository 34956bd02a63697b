"""Serving benchmark: continuous-batching decode throughput (tokens/s).

Not the driver's headline bench (that is bench.py's training step) — this
measures the serving half: N concurrent requests through BatchedGenerator
on one GPU. Usage:

    python bench_decode.py --model llama3-8b --batch 8 --prompt 128 --new 128
"""
import argparse
import json
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b",
                    choices=["llama3-8b", "tiny"])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--prompt", type=int, default=128)
    ap.add_argument("--new", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=1)
    args = ap.parse_args()

    from kubetorch_amd.models import (BatchedGenerator, Llama, llama3_8b,
                                      llama_tiny)

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    cfg = (llama3_8b() if args.model == "llama3-8b" else llama_tiny())
    torch.manual_seed(0)
    prev = torch.get_default_dtype()
    if dev.type == "cuda":
        torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(dev):
            model = Llama(cfg).eval()
    finally:
        torch.set_default_dtype(prev)

    def run_once():
        eng = BatchedGenerator(model, max_batch=args.batch,
                               max_len=args.prompt + args.new + 8)
        for i in range(args.batch):
            prompt = torch.randint(0, cfg.vocab_size, (args.prompt,))
            eng.submit(prompt.tolist(), max_new_tokens=args.new)
        out = eng.run()
        return sum(len(v) - args.prompt for v in out.values())

    for _ in range(args.warmup):
        run_once()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    new_toks = run_once()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    print(json.dumps({
        "metric": "decode_tokens_per_sec",
        "value": round(new_toks / dt, 2),
        "unit": "tokens/s",
        "batch": args.batch,
        "prompt_len": args.prompt,
        "new_tokens": args.new,
        "model": args.model,
        "device": dev.type,
    }), flush=True)
    print(f"[decode] {new_toks} tokens in {dt:.2f}s", file=sys.stderr)


if __name__ == "__main__":
    main()
