"""GPU debug for the custom attention fwd: numerics vs torch (fwd+bwd via
the aten efficient-attention kernels), LSE check, perf timing.
PYTHONPATH=. python tests/debug_attn.py"""
import sys
import time

import torch
import torch.nn.functional as F

from kubetorch_amd import ops

torch.manual_seed(0)
dev = "cuda"
aten_fwd = torch.ops.aten._scaled_dot_product_efficient_attention
aten_bwd = torch.ops.aten._scaled_dot_product_efficient_attention_backward


def check(B, Hq, Hkv, S, tag, strided=False):
    g = Hq // Hkv
    if strided:
        # [B,S,H,D] storage, permuted views (the model layout, no copies)
        q = torch.randn(B, S, Hq, 128, dtype=torch.bfloat16,
                        device=dev).permute(0, 2, 1, 3)
        k = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16,
                        device=dev).permute(0, 2, 1, 3)
        v = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16,
                        device=dev).permute(0, 2, 1, 3)
    else:
        q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
        k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
        v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    k_exp = k.repeat_interleave(g, 1).contiguous()
    v_exp = v.repeat_interleave(g, 1).contiguous()
    scale = 128 ** -0.5

    o, lse = ops._ext().attn_fwd_ck(q, k, v, scale)
    o_tr, lse_tr = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    o_v3, lse_v3 = ops._ext().attn_fwd_v3(q, k, v, scale)
    out_ref, lse_ref, seed, offset = aten_fwd(q, k_exp, v_exp, None, True,
                                              0.0, True, scale=scale)
    d = (o.float() - out_ref.float()).abs()
    dl = (lse - lse_ref).abs()
    dv3 = (o_v3.float() - out_ref.float()).abs()
    print(f"[{tag}] v3 fwd max={dv3.max().item():.4e} "
          f"v3 lse max={(lse_v3 - lse_ref).abs().max().item():.4e}")
    dtr = (o_tr.float() - out_ref.float()).abs()
    print(f"[{tag}] tr fwd max={dtr.max().item():.4e} "
          f"tr lse max={(lse_tr - lse_ref).abs().max().item():.4e}")
    print(f"[{tag}] fwd max={d.max().item():.4e} lse max={dl.max().item():.4e} "
          f"lse_ref sample={lse_ref.flatten()[:2].tolist()} "
          f"mine={lse.flatten()[:2].tolist()}")
    print(f"[{tag}] seed={seed} offset={offset} lse shape ref={tuple(lse_ref.shape)}")

    gout = torch.randn_like(o)
    # bwd with aten's own fwd outputs (sanity)
    dq0, dk0, dv0, _ = aten_bwd(gout, q, k_exp, v_exp, None, out_ref,
                                lse_ref, seed, offset, 0.0,
                                [True, True, True, False], True, scale=scale)
    # bwd with MY fwd outputs
    dq1, dk1, dv1, _ = aten_bwd(gout, q, k_exp, v_exp, None, o,
                                lse, seed, offset, 0.0,
                                [True, True, True, False], True, scale=scale)
    for name, a, b in (("dq", dq1, dq0), ("dk", dk1, dk0), ("dv", dv1, dv0)):
        dd = (a.float() - b.float()).abs()
        print(f"[{tag}] {name} vs aten-own max={dd.max().item():.4e} "
              f"nan_mine={a.isnan().any().item()} nan_aten={b.isnan().any().item()}")

    # full wrapper path
    qa = q.clone().requires_grad_(True)
    ka = k.clone().requires_grad_(True)
    va = v.clone().requires_grad_(True)
    out = ops.flash_attention(qa, ka, va)
    out.backward(gout)
    # reference grads in fp32 autograd
    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    outr = F.scaled_dot_product_attention(
        qr, kr.repeat_interleave(g, 1), vr.repeat_interleave(g, 1),
        is_causal=True)
    outr.backward(gout.float())
    for name, a, b in (("dq", qa.grad, qr.grad), ("dk", ka.grad, kr.grad),
                       ("dv", va.grad, vr.grad)):
        dd = (a.float() - b).abs()
        rel = dd.max() / b.abs().max().clamp(min=1e-6)
        print(f"[{tag}] wrapper {name} max={dd.max().item():.4e} rel={rel.item():.3e}")


def perf(B, Hq, Hkv, S, iters=20):
    g = Hq // Hkv
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    k_exp = k.repeat_interleave(g, 1).contiguous()
    v_exp = v.repeat_interleave(g, 1).contiguous()
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

    t_mine = timeit(lambda: ops._ext().attn_fwd_ck(q, k, v, scale))
    t_tr = timeit(lambda: ops._ext().attn_fwd_ck_tr(q, k, v, scale))
    t_v3 = timeit(lambda: ops._ext().attn_fwd_v3(q, k, v, scale))
    t_wmma = timeit(lambda: ops._ext().attn_fwd(q, k, v, scale))
    t_torch = timeit(lambda: aten_fwd(q, k_exp, v_exp, None, True, 0.0, True,
                                      scale=scale))
    flops = 4 * B * Hq * S * S * 128 * 0.5
    print(f"perf B{B} H{Hq}/{Hkv} S{S}: ck {flops/t_mine/1e12:.0f} TF  "
          f"ck_tr {flops/t_tr/1e12:.0f} TF  v3 {flops/t_v3/1e12:.0f} TF  "
          f"wmma {flops/t_wmma/1e12:.0f} TF  "
          f"aten {flops/t_torch/1e12:.0f} TF")


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "bwd":
    pass  # skip the fwd suite; bwd_mask_ab runs below
elif __name__ == "__main__":
    check(1, 2, 2, 256, "tiny MHA")
    check(1, 4, 2, 384, "small GQA")
    check(2, 8, 2, 1024, "mid GQA")
    check(2, 8, 2, 1024, "strided GQA", strided=True)
    perf(4, 32, 8, 4096)
    perf(1, 32, 8, 2048)
    print("done")


def bwd_mask_ab():
    """Round-2 kickoff: sweep the CK bwd mask conventions in ONE gpu call.
    Run: python tests/debug_attn.py bwd"""
    import os

    import torch.nn.functional as F

    torch.manual_seed(9)
    B, Hq, Hkv, S, D = 2, 8, 2, 256, 128
    scale = D ** -0.5
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    gout = torch.randn_like(o)
    g = Hq // Hkv
    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    outr = F.scaled_dot_product_attention(
        qr, kr.repeat_interleave(g, 1), vr.repeat_interleave(g, 1),
        is_causal=True, scale=scale)
    outr.backward(gout.float())
    for mode in ("0", "1", "2"):
        os.environ["KT_CKBWD_MASK"] = mode
        dq, dk_e, dv_e = ops._ext().attn_bwd_ck(
            gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
        dk = dk_e.view(B, Hkv, g, S, D).sum(2)
        dv = dv_e.view(B, Hkv, g, S, D).sum(2)
        print(f"mask_mode={mode}: "
              f"dq {(dq.float()-qr.grad).abs().max():.3e} "
              f"dk {(dk.float()-kr.grad).abs().max():.3e} "
              f"dv {(dv.float()-vr.grad).abs().max():.3e} "
              f"(dq row0 {dq[0,0,0].abs().max():.3e} "
              f"rowN {dq[0,0,-1].abs().max():.3e})")
    os.environ.pop("KT_CKBWD_MASK", None)


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "bwd":
    bwd_mask_ab()
