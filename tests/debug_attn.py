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


if __name__ == "__main__" and len(sys.argv) > 1:
    pass  # skip the fwd suite; the named probe runs below
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
    print(f"ref scale: dq {qr.grad.abs().max():.3e} dk {kr.grad.abs().max():.3e} "
          f"dv {vr.grad.abs().max():.3e} "
          f"(ref dq row0 {qr.grad[0,0,0].abs().max():.3e} "
          f"rowN {qr.grad[0,0,-1].abs().max():.3e})")
    for pipe in ("0", "1"):
        os.environ["KT_CKBWD_PIPE"] = pipe
        for mode in ("0", "2"):
            os.environ["KT_CKBWD_MASK"] = mode
            dq, dk_e, dv_e = ops._ext().attn_bwd_ck(
                gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
            dk = dk_e.view(B, Hkv, g, S, D).sum(2)
            dv = dv_e.view(B, Hkv, g, S, D).sum(2)
            # per-row error norms reveal the error's structure (triangular
            # pattern = mask problem; uniform = layout/lse problem)
            dq_err = (dq.float() - qr.grad).abs()
            rows = dq_err[0, 0].max(dim=-1).values
            q4 = [f"{rows[i*S//8].item():.2e}" for i in range(8)]
            print(f"pipe={pipe} mask={mode}: "
                  f"dq {dq_err.max():.3e} "
                  f"dk {(dk.float()-kr.grad).abs().max():.3e} "
                  f"dv {(dv.float()-vr.grad).abs().max():.3e} "
                  f"| dq row-err profile {q4}")
    os.environ.pop("KT_CKBWD_MASK", None)
    os.environ.pop("KT_CKBWD_PIPE", None)


def bwd_probe():
    """Decompose the CK bwd into its math and find which quantity diverges.
    MHA B=1 H=1 S=128 single tile. Run: python tests/debug_attn.py probe"""
    import os

    torch.manual_seed(5)
    B, H, S, D = 1, 1, 128, 128
    scale = D ** -0.5
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    gout = torch.randn_like(o)

    # model the kernel's math in fp32
    qf, kf, vf, of, gof = (t.float() for t in (q, k, v, o, gout))
    s = qf @ kf.transpose(-1, -2)                      # raw logits
    mask = torch.ones(S, S, device="cuda", dtype=torch.bool).tril()
    P = torch.exp(scale * s - lse.unsqueeze(-1)).where(mask, torch.zeros(()).cuda())
    dv_model = P.transpose(-1, -2) @ gof
    Dt = (gof * of).sum(-1)                            # [B,H,S]
    dP = gof @ vf.transpose(-1, -2)
    dS = P * (dP - Dt.unsqueeze(-1))
    dk_model = scale * dS.transpose(-1, -2) @ qf
    dq_model = scale * dS @ kf
    # autograd reference for sanity
    qr = q.float().requires_grad_(True)
    kr = k.float().requires_grad_(True)
    vr = v.float().requires_grad_(True)
    outr = F.scaled_dot_product_attention(qr, kr, vr, is_causal=True, scale=scale)
    outr.backward(gout.float())
    print(f"model-vs-autograd: dq {(dq_model-qr.grad).abs().max():.2e} "
          f"dk {(dk_model-kr.grad).abs().max():.2e} "
          f"dv {(dv_model-vr.grad).abs().max():.2e}")

    for pipe in ("0", "1"):
        os.environ["KT_CKBWD_PIPE"] = pipe
        dq, dk, dv = ops._ext().attn_bwd_ck(
            gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
        print(f"pipe={pipe}: kernel-vs-model dq {(dq.float()-dq_model).abs().max():.2e} "
              f"dk {(dk.float()-dk_model).abs().max():.2e} "
              f"dv {(dv.float()-dv_model).abs().max():.2e}")
        # candidate wrong-P models to fingerprint the failure
        P_nomask = torch.exp(scale * s - lse.unsqueeze(-1))
        P_nolse = torch.exp(scale * s).where(mask, torch.zeros(()).cuda())
        P_sm = torch.softmax(scale * s + torch.where(mask, 0.0, -torch.inf), -1)
        for name, Pc in (("nomask", P_nomask), ("nolse", P_nolse), ("sm", P_sm)):
            dvc = Pc.transpose(-1, -2) @ gof
            print(f"  pipe={pipe} dv vs {name}-P model: "
                  f"{(dv.float()-dvc).abs().max():.2e}")
        # structural transforms of kernel dv vs model
        dvk = dv.float()
        print(f"  dv transposed-SD {(dvk.transpose(-1,-2)-dv_model).abs().max():.2e} "
              f"rev-rows {(dvk.flip(2)-dv_model).abs().max():.2e} "
              f"corr {torch.corrcoef(torch.stack([dvk.flatten(), dv_model.flatten()]))[0,1]:.3f}")
        print(f"  dv[0,0,:2,:4] kernel {dvk[0,0,:2,:4].tolist()}")
        print(f"  dv[0,0,:2,:4] model  {dv_model[0,0,:2,:4].tolist()}")
    os.environ.pop("KT_CKBWD_PIPE", None)


def bwd_probe2():
    """Expose the kernel's reconstructed P: with dO = e_m (single nonzero
    query row of ones), dv[j,:] = P[m,j]. Run: debug_attn.py probe2"""
    torch.manual_seed(5)
    B, H, S, D = 1, 1, 128, 128
    scale = D ** -0.5
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    qf, kf = q.float(), k.float()
    s = qf @ kf.transpose(-1, -2)
    mask = torch.ones(S, S, device="cuda", dtype=torch.bool).tril()
    P = torch.exp(scale * s - lse.unsqueeze(-1)).where(
        mask, torch.zeros((), device="cuda"))
    for m in (0, 1, 63, 64, 127):
        gout = torch.zeros_like(o)
        gout[0, 0, m, :] = 1.0
        dq, dk, dv = ops._ext().attn_bwd_ck(
            gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
        p_kernel = dv.float()[0, 0, :, 0]          # P[m, j] for all j
        p_model = P[0, 0, m, :]
        diff = (p_kernel - p_model).abs()
        nz_k = p_kernel.abs().gt(1e-6).nonzero().flatten()
        bad = diff.gt(1e-2).nonzero().flatten()
        print(f"m={m}: P-row err max={diff.max():.2e} "
              f"kernel nz range=[{nz_k.min().item() if nz_k.numel() else -1},"
              f"{nz_k.max().item() if nz_k.numel() else -1}] "
              f"rowsum kernel={p_kernel.sum():.4f} model={p_model.sum():.4f} "
              f"bad_cols={bad[:8].tolist()}")
        # check dv column consistency: every head-dim col should be equal
        spread = (dv.float()[0, 0] - dv.float()[0, 0, :, :1]).abs().max()
        print(f"   dv col-spread (0 if clean rank-1) = {spread:.2e} "
              f"P[m,:6] kernel={p_kernel[:6].tolist()}")
        print(f"                               model={p_model[:6].tolist()}")


def bwd_probe3():
    """Full-P extraction: with S=D=128 and dO=I, dv = P^T. Fit the kernel's
    P against exp(scale*s[perm] - lse) to identify operand permutations.
    Run: debug_attn.py probe3"""
    torch.manual_seed(5)
    B, H, S, D = 1, 1, 128, 128
    scale = D ** -0.5
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    gout = torch.zeros_like(o)
    gout[0, 0] = torch.eye(S, dtype=torch.bfloat16, device="cuda")
    dq, dk, dv = ops._ext().attn_bwd_ck(
        gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
    Pk = dv.float()[0, 0].transpose(0, 1)              # [S_q, S_k] kernel P
    M = (scale * (q.float() @ k.float().transpose(-1, -2))
         - lse.unsqueeze(-1))[0, 0]                    # unmasked log-P model
    Pm = torch.exp(M).tril()
    print(f"P err max={(Pk - Pm).abs().max():.3e} frob={(Pk - Pm).norm():.3f} "
          f"Pk rowsums[:4]={Pk.sum(1)[:4].tolist()}")
    # permutation fit on key axis: for each row m, for each nonzero col j,
    # best j' with M[m, j'] ~ ln Pk[m, j]
    lnPk = Pk.clamp(min=1e-30).log()
    for m in (1, 5, 64, 127):
        cols = Pk[m].abs().gt(1e-5).nonzero().flatten()[:10]
        best = [(int(j), int((M[m] - lnPk[m, j]).abs().argmin()),
                 float((M[m] - lnPk[m, j]).abs().min()))
                for j in cols.tolist()]
        print(f"m={m}: (j -> best j', resid) {best}")
    # q-axis permutation fit: ln Pk[m, j] vs M[:, j]
    for m in (1, 5, 64):
        cols = Pk[m].abs().gt(1e-5).nonzero().flatten()[:6]
        best = [(int(j), int((M[:, j] - lnPk[m, j]).abs().argmin()),
                 float((M[:, j] - lnPk[m, j]).abs().min()))
                for j in cols.tolist()]
        print(f"m={m} (q-axis): (j -> best m', resid) {best}")


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "bwd":
    bwd_mask_ab()
elif __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "probe":
    bwd_probe()
elif __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "probe2":
    bwd_probe2()
elif __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "probe3":
    bwd_probe3()


def bwd_probe4():
    """Fingerprint the kernel's S-accumulation: regress ln(P_kernel)+lse
    against head-dim chunk contractions of q@k^T. Run: debug_attn.py probe4"""
    torch.manual_seed(5)
    B, H, S, D = 1, 1, 128, 128
    scale = D ** -0.5
    q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    gout = torch.zeros_like(o)
    gout[0, 0] = torch.eye(S, dtype=torch.bfloat16, device="cuda")
    dq, dk, dv = ops._ext().attn_bwd_ck(
        gout, q, k, v, o.contiguous(), lse.contiguous(), scale)
    Pk = dv.float()[0, 0].transpose(0, 1).contiguous()   # [S_q, S_k]
    E = Pk.clamp(min=1e-30).log() + lse[0, 0].unsqueeze(-1)  # scale*s'
    valid = Pk.gt(1e-4) & torch.ones(S, S, device="cuda").tril().bool()
    # chunk contractions at 8-dim granularity
    nch = 16
    qf, kf = q.float()[0, 0], k.float()[0, 0]
    C = torch.stack([scale * (qf[:, i*8:(i+1)*8] @ kf[:, i*8:(i+1)*8].T)
                     for i in range(nch)], -1)            # [S,S,16]
    A = C[valid]                                          # [n,16]
    y = E[valid]
    sol = torch.linalg.lstsq(A, y.unsqueeze(-1)).solution.flatten()
    resid = (A @ sol - y).abs()
    print(f"chunk coeffs (1.0 = correct): {[round(float(x),3) for x in sol]}")
    print(f"lstsq resid max={resid.max():.3e} mean={resid.mean():.3e} n={y.numel()}")
    # also try cross-chunk terms: q chunk i against k chunk j (misaligned frag)
    if resid.max() > 0.05:
        CC = torch.stack([scale * (qf[:, i*8:(i+1)*8] @ kf[:, j*8:(j+1)*8].T)
                          for i in range(nch) for j in range(nch)], -1)
        A2 = CC[valid]
        sol2 = torch.linalg.lstsq(A2, y.unsqueeze(-1)).solution.flatten()
        resid2 = (A2 @ sol2 - y).abs()
        big = [(i // nch, i % nch, round(float(c), 3))
               for i, c in enumerate(sol2) if abs(float(c)) > 0.1]
        print(f"cross-chunk resid max={resid2.max():.3e}; "
              f"coeffs>0.1 (qchunk,kchunk,coef): {big}")


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "probe4":
    bwd_probe4()


def bwd_perf():
    """Time the CK GQA-native bwd (both pipelines) vs the aten path at the
    Llama-3-8B shape. Run: debug_attn.py bwdperf"""
    import os

    B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
    g = Hq // Hkv
    scale = D ** -0.5
    q = torch.randn(B, Hq, S, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(B, Hkv, S, D, dtype=torch.bfloat16, device="cuda")
    o, lse = ops._ext().attn_fwd_ck_tr(q, k, v, scale)
    o = o.contiguous()
    lse = lse.contiguous()
    gout = torch.randn_like(o)

    def timeit(fn, iters=10):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    def ck_path():
        dq, dk_e, dv_e = ops._ext().attn_bwd_ck(gout, q, k, v, o, lse, scale)
        dk = dk_e.view(B, Hkv, g, S, D).sum(2)
        dv = dv_e.view(B, Hkv, g, S, D).sum(2)
        return dq, dk, dv

    def aten_path():
        k_exp = k.repeat_interleave(g, dim=1)
        v_exp = v.repeat_interleave(g, dim=1)
        seed = torch.zeros((), dtype=torch.long, device=q.device)
        offset = torch.zeros((), dtype=torch.long, device=q.device)
        dq, dk, dv, _ = aten_bwd(gout, q, k_exp, v_exp, None, o, lse, seed,
                                 offset, 0.0, [True, True, True, False], True,
                                 scale=scale)
        dk = dk.view(B, Hkv, g, S, D).sum(2)
        dv = dv.view(B, Hkv, g, S, D).sum(2)
        return dq, dk, dv

    flops = 4 * B * Hq * S * S * D * 0.5 * 2.5  # bwd ~2.5x fwd matmul work
    ra = aten_path()
    for pipe in ("0", "1", "2", "3"):
        os.environ["KT_CKBWD_PIPE"] = pipe
        t = timeit(ck_path)
        rc = ck_path()
        errs = " ".join(f"{n}={(a.float()-b.float()).abs().max():.1e}"
                        for n, a, b in zip("dq dk dv".split(), rc, ra))
        print(f"ck bwd pipe={pipe}: {t*1e3:.2f} ms ({flops/t/1e12:.0f} TF) "
              f"vs-aten {errs}")
    os.environ.pop("KT_CKBWD_PIPE", None)
    t = timeit(aten_path)
    print(f"aten bwd (expand+sum): {t*1e3:.2f} ms ({flops/t/1e12:.0f} TF)")


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "bwdperf":
    bwd_perf()


def v3_sweep():
    """Sweep KT_V3_REMAP (XCD tile remap) on the Llama shape.
    Run: debug_attn.py v3sweep"""
    import os

    B, Hq, Hkv, S = 4, 32, 8, 4096
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    scale = 128 ** -0.5
    flops = 4 * B * Hq * S * S * 128 * 0.5

    def timeit(fn, iters=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    ref = None
    for remap in ("0", "1", "2"):
        os.environ["KT_V3_REMAP"] = remap
        t = timeit(lambda: ops._ext().attn_fwd_v3(q, k, v, scale))
        o, lse = ops._ext().attn_fwd_v3(q, k, v, scale)
        if ref is None:
            ref = o.float()
        err = (o.float() - ref).abs().max().item()
        print(f"remap={remap}: {flops/t/1e12:.0f} TF ({t*1e3:.3f} ms) err={err:.1e}")
    os.environ.pop("KT_V3_REMAP", None)


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "v3sweep":
    v3_sweep()


def flash_fwd_probe():
    """Does aten's FLASH forward (AITER asm candidate) beat our CK v3 on
    the Llama shape? Run: debug_attn.py flashfwd"""
    import torch.nn.functional as F
    from torch.nn.attention import SDPBackend, sdpa_kernel

    B, Hq, Hkv, S = 4, 32, 8, 4096
    q = torch.randn(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    scale = 128 ** -0.5
    flops = 4 * B * Hq * S * S * 128 * 0.5

    def timeit(fn, iters=20):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    for name, backend in (("flash", SDPBackend.FLASH_ATTENTION),
                          ("efficient", SDPBackend.EFFICIENT_ATTENTION)):
        try:
            with sdpa_kernel(backend):
                t = timeit(lambda: F.scaled_dot_product_attention(
                    q, k, v, is_causal=True, scale=scale, enable_gqa=True))
            print(f"sdpa {name} (gqa-native): {flops/t/1e12:.0f} TF")
        except RuntimeError as e:
            print(f"sdpa {name}: unavailable ({str(e)[:80]})")
    t = timeit(lambda: ops._ext().attn_fwd_v3(q, k, v, scale))
    print(f"our v3 (w/ lse):  {flops/t/1e12:.0f} TF")
    # training needs LSE: time the aten flash fwd WITH logsumexp
    try:
        t = timeit(lambda: torch.ops.aten._flash_attention_forward(
            q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
            None, None, S, S, 0.0, True, True, scale=scale))
        print(f"aten _flash_attention_forward (lse): {flops/t/1e12:.0f} TF")
    except Exception as e:
        print(f"_flash_attention_forward: {str(e)[:100]}")


if __name__ == "__main__" and len(sys.argv) > 1 and sys.argv[1] == "flashfwd":
    flash_fwd_probe()
