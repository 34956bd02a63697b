"""kubetorch_amd.ops — MI355X-native fused ops with autograd.

Dispatch policy:
  * GPU tensors run the hand-written gfx950 HIP kernels (_hip_ops.so,
    built in-tree by kubetorch_amd.ops.build). If the extension is missing
    on a GPU box the ops raise — there is no silent eager fallback.
  * CPU tensors run a plain fp32 PyTorch reference of the same op; this is
    the numerics reference the GPU tests compare against, and what CPU-only
    CI exercises.
"""
import importlib.util
import os

import torch

_EXT = None
_EXT_ERR = None


def _ext():
    """Load the in-tree HIP extension, raising loudly if unavailable."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    if _EXT_ERR is not None:
        raise _EXT_ERR
    so = os.path.join(os.path.dirname(__file__), "_hip_ops.so")
    if not os.path.exists(so):
        _EXT_ERR = RuntimeError(
            f"kubetorch_amd HIP extension not built: {so} missing. "
            "Run `python -m kubetorch_amd.ops.build` (gfx950)."
        )
        raise _EXT_ERR
    spec = importlib.util.spec_from_file_location("kubetorch_amd.ops._hip_ops", so)
    mod = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(mod)
    except Exception as e:  # pragma: no cover
        _EXT_ERR = RuntimeError(f"failed to load {so}: {e}")
        raise _EXT_ERR
    _EXT = mod
    return _EXT


def hip_available():
    try:
        _ext()
        return True
    except Exception:
        # On a GPU machine a missing extension must FAIL, not silently
        # run the eager fallback at a fraction of the speed (the gfx950
        # .so ships in-tree; its absence means a broken build/snapshot).
        # KT_ALLOW_EAGER_FALLBACK=1 overrides for debugging.
        import torch

        if torch.cuda.is_available() and \
                os.environ.get("KT_ALLOW_EAGER_FALLBACK") != "1":
            raise
        return False


_SPILL = None


def _spill_ext():
    """The native checkpoint spill engine (_hip_spill.so)."""
    global _SPILL
    if _SPILL is not None:
        return _SPILL
    so = os.path.join(os.path.dirname(__file__), "_hip_spill.so")
    if not os.path.exists(so):
        raise RuntimeError(
            f"spill extension not built: {so} missing "
            "(python -m kubetorch_amd.ops.build)")
    spec = importlib.util.spec_from_file_location(
        "kubetorch_amd.ops._hip_spill", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _SPILL = mod
    return _SPILL


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------
def _rmsnorm_ref_fwd(x, w, eps):
    xf = x.float()
    invrms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    y = (xf * invrms * w.float()).to(x.dtype)
    return y, invrms.squeeze(-1).reshape(-1)


def _rmsnorm_ref_bwd(dy, ds, s, w, invrms):
    H = s.shape[-1]
    sf = s.float().reshape(-1, H)
    dyf = dy.float().reshape(-1, H)
    wf = w.float()
    ir = invrms.reshape(-1, 1)
    S = (dyf * wf * sf).sum(-1, keepdim=True)
    g = ir * dyf * wf - sf * (ir ** 3) * S / H
    if ds is not None:
        g = g + ds.float().reshape(-1, H)
    dw = (dyf * sf * ir).sum(0).to(w.dtype)
    return g.to(s.dtype).reshape(s.shape), dw


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        if x.is_cuda:
            y, invrms = _ext().rmsnorm_fwd(x.contiguous(), None,
                                           w.contiguous(), eps)
        else:
            y, invrms = _rmsnorm_ref_fwd(x, w, eps)
        ctx.save_for_backward(x, w, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, invrms = ctx.saved_tensors
        if x.is_cuda:
            dx, dw = _ext().rmsnorm_bwd(dy.contiguous(), None, x.contiguous(),
                                        w.contiguous(), invrms)
        else:
            dx, dw = _rmsnorm_ref_bwd(dy, None, x, w, invrms)
        return dx, dw, None


def rmsnorm(x, w, eps=1e-5):
    """y = x * rsqrt(mean(x^2, -1) + eps) * w (bf16 in/out, fp32 accum)."""
    return _RMSNorm.apply(x, w, eps)


class _AddRMSNorm(torch.autograd.Function):
    """Fused pre-norm residual add: (y, s) = (rmsnorm(x + res) * w, x + res).
    Backward folds the residual fan-in add: dx = dres = ds + d(norm)·dy."""

    @staticmethod
    def forward(ctx, x, res, w, eps):
        ctx.set_materialize_grads(False)  # unused s output -> ds is None
        if x.is_cuda:
            y, invrms, s = _ext().rmsnorm_fwd(x.contiguous(), res.contiguous(),
                                              w.contiguous(), eps)
        else:
            s = (x.float() + res.float()).to(x.dtype)
            y, invrms = _rmsnorm_ref_fwd(s, w, eps)
        ctx.save_for_backward(s, w, invrms)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, w, invrms = ctx.saved_tensors
        if dy is None:
            dy = torch.zeros_like(s)
        if ds is not None:
            ds = ds.contiguous()
        if s.is_cuda:
            g, dw = _ext().rmsnorm_bwd(dy.contiguous(), ds, s, w.contiguous(),
                                       invrms)
        else:
            g, dw = _rmsnorm_ref_bwd(dy, ds, s, w, invrms)
        return g, g, dw, None


def add_rmsnorm(x, res, w, eps=1e-5):
    """Fused residual add + RMSNorm: returns (normed, x + res)."""
    return _AddRMSNorm.apply(x, res, w, eps)


# ---------------------------------------------------------------------------
# RoPE (Llama rotate-half)
# ---------------------------------------------------------------------------
def precompute_rope(seq_len, head_dim, base=500000.0, device=None):
    """cos/sin tables [S, D/2] fp32 (host-side trig per CDNA guide App. B)."""
    inv_freq = 1.0 / (
        base ** (torch.arange(0, head_dim, 2, dtype=torch.float32, device=device) / head_dim)
    )
    t = torch.arange(seq_len, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    return freqs.cos().contiguous(), freqs.sin().contiguous()


def _rope_ref(x, cos, sin, sign, oscale=1.0):
    # x: [B, S, Hh, D]
    B, S, Hh, D = x.shape
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2 :]
    c = cos[:S].view(1, S, 1, D // 2)
    s = sin[:S].view(1, S, 1, D // 2) * sign
    o1 = (x1 * c - x2 * s) * oscale
    o2 = (x2 * c + x1 * s) * oscale
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, oscale):
        ctx.S = x.shape[1]
        ctx.oscale = oscale
        ctx.save_for_backward(cos, sin)
        if x.is_cuda:
            B, S, Hh, D = x.shape
            xv = x.reshape(B * S, Hh, D)  # strided views pass through
            return _ext().rope(xv, cos, sin, S, 1.0, oscale).view(x.shape)
        return _rope_ref(x, cos, sin, 1.0, oscale)

    @staticmethod
    def backward(ctx, dy):
        # RoPE is linear: y = oscale * R(x)  =>  dx = oscale * R^T(dy), so
        # the same oscale rides along for free in the backward kernel.
        cos, sin = ctx.saved_tensors
        if dy.is_cuda:
            B, S, Hh, D = dy.shape
            dyv = dy.reshape(B * S, Hh, D)
            dx = _ext().rope(dyv, cos, sin, S, -1.0, ctx.oscale).view(dy.shape)
        else:
            dx = _rope_ref(dy, cos, sin, -1.0, ctx.oscale)
        return dx, None, None, None


def rope(x, cos, sin, oscale=1.0):
    """Apply rotate-half RoPE to x: [B, S, Hh, D] using [S, D/2] tables.

    oscale multiplies the output in fp32 before the bf16 round (free in the
    memory-bound kernel). Used to fold the attention softmax scale * log2e
    into Q for the v3 FMHA path (flash_attention impl="v3")."""
    return _Rope.apply(x, cos, sin, oscale)


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------
def _swiglu_ref_fwd(gu):
    I = gu.shape[-1] // 2
    g, u = gu[..., :I].float(), gu[..., I:].float()
    return (torch.nn.functional.silu(g) * u).to(gu.dtype)


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gu):
        ctx.save_for_backward(gu)
        if gu.is_cuda:
            return _ext().swiglu_fwd(gu.contiguous())
        return _swiglu_ref_fwd(gu)

    @staticmethod
    def backward(ctx, dout):
        (gu,) = ctx.saved_tensors
        if gu.is_cuda:
            return _ext().swiglu_bwd(dout.contiguous(), gu.contiguous())
        I = gu.shape[-1] // 2
        g, u = gu[..., :I].float(), gu[..., I:].float()
        do = dout.float()
        sig = torch.sigmoid(g)
        dg = do * u * sig * (1 + g * (1 - sig))
        du = do * g * sig
        return torch.cat([dg, du], dim=-1).to(gu.dtype)


def swiglu(gate_up):
    """out = silu(gate) * up with gate_up = [..., 2I] packed from one GEMM."""
    return _SwiGLU.apply(gate_up)


# ---------------------------------------------------------------------------
# Fused cross entropy (destroys logits: overwritten with the gradient)
# ---------------------------------------------------------------------------
class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        V = logits.shape[-1]
        flat = logits.reshape(-1, V)
        t = targets.reshape(-1)
        n_valid = (t != ignore_index).sum().clamp(min=1)
        if logits.is_cuda:
            loss = _ext().cross_entropy_fwd_(flat, t.contiguous(), 1.0, ignore_index)
            dlogits = flat  # overwritten in place by the kernel
        else:
            lf = flat.float()
            logp = torch.log_softmax(lf, dim=-1)
            loss = torch.nn.functional.nll_loss(
                logp, t, reduction="none", ignore_index=ignore_index
            )
            dlogits = torch.softmax(lf, dim=-1)
            valid = (t != ignore_index).unsqueeze(-1)
            dlogits.scatter_add_(
                1, t.clamp(min=0).unsqueeze(1), -torch.ones_like(t, dtype=torch.float32).unsqueeze(1)
            )
            dlogits = torch.where(valid, dlogits, torch.zeros_like(dlogits)).to(logits.dtype)
        ctx.save_for_backward(dlogits, n_valid)
        ctx.shape = logits.shape
        return loss.sum() / n_valid.to(loss.dtype)

    @staticmethod
    def backward(ctx, gout):
        dlogits, n_valid = ctx.saved_tensors
        scale = (gout / n_valid).to(torch.float32)
        return (dlogits * scale).to(dlogits.dtype).reshape(ctx.shape), None, None


def fused_cross_entropy(logits, targets, ignore_index=-100):
    """Mean CE over valid tokens. WARNING: logits buffer is overwritten with
    the (unscaled) gradient on GPU — do not reuse logits after this call."""
    return _FusedCE.apply(logits, targets, ignore_index)


# ---------------------------------------------------------------------------
# Flash attention: custom gfx950 forward + torch (AITER asm) backward
# ---------------------------------------------------------------------------
_LN2 = 0.6931471805599453
LOG2E = 1.4426950408889634


class _FlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, impl):
        if impl == "v3":
            # AITER-schedule CK v3 kernel (fastest fwd, profiles/). Contract:
            # q arrives PRE-SCALED by scale*log2e (folded into the RoPE
            # kernel's oscale), so softmax(scale*q0@kT) == softmax(ln2*q@kT)
            # and the effective scale for fwd LSE and backward is ln2.
            o, lse = _ext().attn_fwd_v3(q, k, v, scale, prescaled=True)
            scale = _LN2
        elif impl == "ck":
            # CK path is stride-aware: permuted [B,S,H,D] views go in as-is
            o, lse = _ext().attn_fwd_ck_tr(q, k, v, scale)
        else:
            q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
            o, lse = _ext().attn_fwd(q, k, v, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, grad_out):
        import os

        q, k, v, o, lse = ctx.saved_tensors
        B, Hq, S, D = q.shape
        Hkv = k.shape[1]
        g = Hq // Hkv
        if os.environ.get("KT_ATTN_BWD") == "ck" and S % 128 == 0:
            # Opt-in: CK-tile GQA-native bwd (no KV expansion; dk/dv come
            # back Hq-expanded and are group-summed). Numerically validated
            # on gfx950 (tests/test_ops.py::test_attn_bwd_ck_gqa_native);
            # measured 4.42 ms vs 2.99 ms for the aten/AITER-asm path at the
            # Llama-3-8B shape (B4 H32/8 S4096), so aten stays the default
            # hot path — see profiles/ROUND2.md lever #1.
            dq, dk_e, dv_e = _ext().attn_bwd_ck(
                grad_out.contiguous(), q.contiguous(), k.contiguous(),
                v.contiguous(), o.contiguous(), lse.contiguous(), ctx.scale)
            if g > 1:
                dk = dk_e.view(B, Hkv, g, S, D).sum(2)
                dv = dv_e.view(B, Hkv, g, S, D).sum(2)
            else:
                dk, dv = dk_e, dv_e
            return dq, dk, dv, None, None
        if g > 1:  # expand KV for the dense backward, then reduce over groups
            k_exp = k.repeat_interleave(g, dim=1)
            v_exp = v.repeat_interleave(g, dim=1)
        else:
            k_exp, v_exp = k, v
        seed = torch.zeros((), dtype=torch.long, device=q.device)
        offset = torch.zeros((), dtype=torch.long, device=q.device)
        dq, dk, dv, _ = torch.ops.aten._scaled_dot_product_efficient_attention_backward(
            grad_out.contiguous(), q, k_exp, v_exp, None, o, lse, seed, offset,
            0.0, [True, True, True, False], True, scale=ctx.scale,
        )
        if g > 1:
            dk = dk.view(B, Hkv, g, S, D).sum(2)
            dv = dv.view(B, Hkv, g, S, D).sum(2)
        return dq, dk, dv, None, None


def flash_attention(q, k, v, scale=None, impl="ck"):
    """Causal GQA attention fwd, bf16, D=128, layout [B, H, S, D], paired
    with torch's AITER asm backward. impl="ck" (default) runs the CK-tile
    FMHA tr-load instantiation (1.8x AOTriton on the Llama shape —
    profiles/); impl="v3" runs the AITER-schedule v3 kernel (2.0x) and
    REQUIRES q pre-scaled by scale*log2e (fold it into ops.rope's oscale);
    impl="wmma" runs the in-tree rocWMMA kernel (ops/hip/attention.hip,
    requires S % 128 == 0)."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    return _FlashAttention.apply(q, k, v, scale, impl)


def flash_attention_supported(q, k, v, is_causal):
    return (is_causal and q.is_cuda and q.dtype == torch.bfloat16
            and q.shape[-1] == 128 and q.shape[2] % 128 == 0
            and q.shape[2] >= 256 and hip_available())


def flash_attention_v3_supported(S, head_dim, device, dtype):
    """Shape gate for the v3 (prescaled-Q) path, checkable BEFORE RoPE so
    the softmax scale can be folded into the RoPE output."""
    return (head_dim == 128 and S % 256 == 0 and S >= 256
            and device.type == "cuda" and dtype == torch.bfloat16
            and hip_available())


# ---------------------------------------------------------------------------
# Multi-tensor pack/unpack (GPU data plane: packed state-dict transfers)
# ---------------------------------------------------------------------------
PACK_ALIGN = 16  # bytes; segment starts in the flat buffer are 16B-aligned


def aligned_offsets(numels, elem_size):
    """Element offsets into the packed flat buffer with every segment start
    16B-aligned (so the pack kernel runs pure uint4 copies). Returns
    (offsets, total_elems)."""
    step = PACK_ALIGN // elem_size
    offs, cur = [], 0
    for n in numels:
        offs.append(cur)
        cur += -(-n // step) * step  # round numel up to the alignment step
    return offs, cur


def pack_tensors(tensors, flat=None):
    """Pack same-dtype contiguous GPU tensors into one flat buffer with a
    single kernel (vs torch.cat's per-tensor copies). Returns (flat,
    offsets) where offsets are element positions (aligned)."""
    dt = tensors[0].dtype
    es = dt.itemsize
    numels = [t.numel() for t in tensors]
    offs, total = aligned_offsets(numels, es)
    dev = tensors[0].device
    if flat is None:
        flat = torch.zeros(total, dtype=dt, device=dev)
    if dev.type == "cuda" and hip_available():
        ptrs = torch.tensor([t.data_ptr() for t in tensors],
                            dtype=torch.int64).to(dev, non_blocking=True)
        nb = torch.tensor([n * es for n in numels],
                          dtype=torch.int64).to(dev, non_blocking=True)
        ob = torch.tensor([o * es for o in offs],
                          dtype=torch.int64).to(dev, non_blocking=True)
        _ext().pack_segments(flat, ptrs, nb, ob, True, max(numels) * es)
    else:
        for t, o, n in zip(tensors, offs, numels):
            flat[o:o + n].copy_(t.reshape(-1))
    return flat, offs


def unpack_tensors(flat, tensors, offsets=None):
    """Scatter a packed flat buffer back into pre-allocated tensors (the
    inverse of pack_tensors) with one kernel on GPU."""
    es = flat.dtype.itemsize
    numels = [t.numel() for t in tensors]
    if offsets is None:
        offsets, _ = aligned_offsets(numels, es)
    if (flat.is_cuda and hip_available()
            and all(t.is_contiguous() for t in tensors)):
        dev = flat.device
        ptrs = torch.tensor([t.data_ptr() for t in tensors],
                            dtype=torch.int64).to(dev, non_blocking=True)
        nb = torch.tensor([n * es for n in numels],
                          dtype=torch.int64).to(dev, non_blocking=True)
        ob = torch.tensor([o * es for o in offsets],
                          dtype=torch.int64).to(dev, non_blocking=True)
        _ext().pack_segments(flat, ptrs, nb, ob, False, max(numels) * es)
    else:
        for t, o in zip(tensors, offsets):
            t.copy_(flat[o:o + t.numel()].view(t.shape))
    return tensors


# ---------------------------------------------------------------------------
# Fused AdamW on flat bf16 buckets (used by kubetorch_amd.parallel)
# ---------------------------------------------------------------------------
def adamw_(p, g, m, v, lr, beta1, beta2, eps, wd, step, grad_scale=1.0):
    if p.is_cuda:
        _ext().adamw_(p, g, m, v, lr, beta1, beta2, eps, wd, step, grad_scale)
        return
    # CPU reference (fp32 math, bf16 params)
    gf = g.float() * grad_scale
    pf = p.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    mhat = m / bc1
    vhat = v / bc2
    pf = pf - lr * (mhat / (vhat.sqrt() + eps) + wd * pf)
    p.copy_(pf.to(p.dtype))
