"""GPU debug for the custom attention fwd: numerics vs torch SDPA (fwd+bwd),
LSE check vs aten efficient attention, and perf timing. Run on a GPU box:
PYTHONPATH=. python tests/debug_attn.py"""
import time

import torch
import torch.nn.functional as F

from kubetorch_amd import ops

torch.manual_seed(0)
dev = "cuda"


def ref_sdpa(q, k, v):
    from torch.nn.attention import SDPBackend, sdpa_kernel

    with sdpa_kernel(SDPBackend.EFFICIENT_ATTENTION):
        Hq, Hkv = q.shape[1], k.shape[1]
        if Hq != Hkv:
            return F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                                  enable_gqa=True)
        return F.scaled_dot_product_attention(q, k, v, is_causal=True)


def check(B, Hq, Hkv, S, tag):
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    scale = 128 ** -0.5

    o, lse = ops._ext().attn_fwd(q, k, v, scale)
    # fp32 reference on a slice of heads (memory)
    with torch.no_grad():
        ref = F.scaled_dot_product_attention(
            q.float(), k.repeat_interleave(Hq // Hkv, 1).float(),
            v.repeat_interleave(Hq // Hkv, 1).float(), is_causal=True)
    d = (o.float() - ref).abs()
    print(f"[{tag}] fwd max_abs={d.max().item():.4e} "
          f"mean={d.mean().item():.2e} nan={o.isnan().any().item()}")

    # lse vs manual
    with torch.no_grad():
        scores = torch.einsum(
            "bhsd,bhtd->bhst", q.float(),
            k.repeat_interleave(Hq // Hkv, 1).float()) * scale
        mask = torch.ones(S, S, device=dev, dtype=torch.bool).tril()
        scores = scores.masked_fill(~mask, float("-inf"))
        lse_ref = torch.logsumexp(scores, dim=-1)
    dl = (lse - lse_ref).abs()
    print(f"[{tag}] lse max_abs={dl.max().item():.4e}")
    del scores, lse_ref

    # full autograd path vs torch SDPA bf16
    qa = q.clone().requires_grad_(True)
    ka = k.clone().requires_grad_(True)
    va = v.clone().requires_grad_(True)
    out = ops.flash_attention(qa, ka, va)
    gout = torch.randn_like(out)
    out.backward(gout)

    qb = q.clone().requires_grad_(True)
    kb = k.clone().requires_grad_(True)
    vb = v.clone().requires_grad_(True)
    outb = ref_sdpa(qb, kb, vb)
    outb.backward(gout)
    for name, a, b in (("dq", qa.grad, qb.grad), ("dk", ka.grad, kb.grad),
                       ("dv", va.grad, vb.grad)):
        dd = (a.float() - b.float()).abs()
        rel = dd.max() / b.float().abs().max().clamp(min=1e-6)
        print(f"[{tag}] {name} max_abs={dd.max().item():.4e} rel={rel.item():.3e}")


def perf(B, Hq, Hkv, S, iters=20):
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    scale = 128 ** -0.5

    def timeit(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    t_mine = timeit(lambda: ops._ext().attn_fwd(q, k, v, scale))
    with torch.no_grad():
        t_torch = timeit(lambda: ref_sdpa(q, k, v))
    flops = 4 * B * Hq * S * S * 128 * 0.5
    print(f"perf B{B} H{Hq}/{Hkv} S{S}: mine {t_mine*1e3:.2f} ms "
          f"({flops/t_mine/1e12:.0f} TF)  torch {t_torch*1e3:.2f} ms "
          f"({flops/t_torch/1e12:.0f} TF)")


if __name__ == "__main__":
    check(1, 2, 2, 128, "tiny MHA")
    check(1, 4, 2, 256, "small GQA")
    check(2, 8, 2, 1024, "mid GQA")
    perf(4, 32, 8, 4096)
    perf(1, 32, 8, 2048)
    print("done")
