"""GEMM-shape microbench for the Llama-3-8B step (profiles/ROUND2.md
lever 2): times every model GEMM (fwd + both backward shapes) at several
token counts M and prints achieved TF/s vs the 2.5 PF bf16 dense peak.
Run on the box: PYTHONPATH=. python tests/bench_gemm.py [--tokens 16384]
"""
import argparse
import time

import torch

PEAK_TF = 2500.0

# (name, K, N) for y[M,N] = x[M,K] @ W[K,N]
LAYERS = [
    ("qkv", 4096, 6144),
    ("wo", 4096, 4096),
    ("w13", 4096, 28672),
    ("w2", 14336, 4096),
    ("lm_head", 4096, 128256),
]


def t(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_shape(M, K, N, tag):
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(K, N, dtype=torch.bfloat16, device="cuda")
    res = torch.randn(M, N, dtype=torch.bfloat16, device="cuda")
    go = torch.randn(M, N, dtype=torch.bfloat16, device="cuda")
    fl = 2.0 * M * K * N / 1e12

    r = {}
    r["fwd_nn"] = fl / t(lambda: x @ w)
    # fused residual: D = 1*C + A@B through the GEMM epilogue (one kernel)
    r["fwd_addmm"] = fl / t(lambda: torch.addmm(res, x, w, beta=1.0))
    wt = w.t().contiguous()
    r["fwd_nt"] = fl / t(lambda: torch.nn.functional.linear(x, wt))
    r["dgrad"] = fl / t(lambda: go @ w.t())            # [M,N]@[N,K]
    r["wgrad"] = fl / t(lambda: x.t() @ go)            # [K,M]@[M,N]
    line = " ".join(f"{k}={v:.0f}TF({100*v/PEAK_TF:.0f}%)"
                    for k, v in r.items())
    print(f"[{tag}] M={M} K={K} N={N}: {line}", flush=True)
    del x, w, res, go, wt
    torch.cuda.empty_cache()
    return r


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, nargs="*",
                    default=[8192, 16384, 24576, 32768])
    args = ap.parse_args()
    for M in args.tokens:
        for name, K, N in LAYERS:
            bench_shape(M, K, N, name)


if __name__ == "__main__":
    main()
