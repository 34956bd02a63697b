"""Llama-3 family, MI355X-first.

The flagship model for bench.py (Llama-3-8B DDP bf16 tokens/sec — the
BASELINE.json north-star config). Hot memory-bound ops run the gfx950 HIP
kernels from kubetorch_amd.ops (RMSNorm, RoPE, SwiGLU, fused CE); GEMMs go
to hipBLASLt via torch.nn.Linear; attention uses torch SDPA (flash) on ROCm.

Reference parity note: the reference (run-house/kubetorch) ships no model
code at all — it launches user training scripts. This
module is the MI355X-native flagship workload those launchers run.
"""
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from kubetorch_amd import ops


@dataclass
class LlamaConfig:
    dim: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    intermediate: int = 14336
    vocab_size: int = 128256
    max_seq_len: int = 8192
    rope_base: float = 500000.0
    norm_eps: float = 1e-5
    init_std: float = 0.02

    @property
    def head_dim(self):
        return self.dim // self.n_heads


def llama3_8b(**overrides) -> LlamaConfig:
    return LlamaConfig(**overrides)


def llama3_70b(**overrides) -> LlamaConfig:
    d = dict(dim=8192, n_layers=80, n_heads=64, n_kv_heads=8,
             intermediate=28672, vocab_size=128256, max_seq_len=8192,
             rope_base=500000.0)
    d.update(overrides)
    return LlamaConfig(**d)


def llama2_7b(**overrides) -> LlamaConfig:
    d = dict(dim=4096, n_layers=32, n_heads=32, n_kv_heads=32,
             intermediate=11008, vocab_size=32000, max_seq_len=4096,
             rope_base=10000.0)
    d.update(overrides)
    return LlamaConfig(**d)


def llama_tiny(**overrides) -> LlamaConfig:
    """Small config for tests / smoke (runs on CPU)."""
    d = dict(dim=256, n_layers=2, n_heads=4, n_kv_heads=2, intermediate=512,
             vocab_size=512, max_seq_len=256)
    d.update(overrides)
    return LlamaConfig(**d)


class KVCache:
    """Per-layer KV cache for autoregressive decode: preallocated
    [B, n_kv, max_len, head_dim] buffers appended in place (no reallocation
    per step, sized for 288 GB HBM3E budgets at batch)."""

    def __init__(self, cfg: LlamaConfig, batch, max_len, device, dtype):
        shape = (batch, cfg.n_kv_heads, max_len, cfg.head_dim)
        self.k = [torch.zeros(shape, device=device, dtype=dtype)
                  for _ in range(cfg.n_layers)]
        self.v = [torch.zeros(shape, device=device, dtype=dtype)
                  for _ in range(cfg.n_layers)]
        self.pos = 0
        self.max_len = max_len

    def append(self, layer, k, v):
        # k, v: [B, n_kv, S, hd] for the current chunk
        S = k.shape[2]
        self.k[layer][:, :, self.pos:self.pos + S] = k
        self.v[layer][:, :, self.pos:self.pos + S] = v
        return (self.k[layer][:, :, :self.pos + S],
                self.v[layer][:, :, :self.pos + S])

    def advance(self, S):
        self.pos += S
        if self.pos > self.max_len:
            raise RuntimeError("KV cache overflow")


class RMSNorm(nn.Module):
    def __init__(self, dim, eps):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return ops.rmsnorm(x, self.weight, self.eps)


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        d, hd = cfg.dim, cfg.head_dim
        self.n_heads, self.n_kv = cfg.n_heads, cfg.n_kv_heads
        self.wqkv = nn.Linear(d, (cfg.n_heads + 2 * cfg.n_kv_heads) * hd, bias=False)
        self.wo = nn.Linear(cfg.n_heads * hd, d, bias=False)

    def _sdpa(self, q, k, v, causal):
        if self.n_kv != self.n_heads:
            try:
                return F.scaled_dot_product_attention(
                    q, k, v, is_causal=causal, enable_gqa=True)
            except (TypeError, RuntimeError):
                rep = self.n_heads // self.n_kv
                k = k.repeat_interleave(rep, dim=1)
                v = v.repeat_interleave(rep, dim=1)
        return F.scaled_dot_product_attention(q, k, v, is_causal=causal)

    def _sdpa_masked(self, q, k, v, mask):
        """Boolean-masked SDPA (True = attend) for ragged decode batches
        (models/serving.py: per-slot cache lengths)."""
        if self.n_kv != self.n_heads:
            try:
                return F.scaled_dot_product_attention(
                    q, k, v, attn_mask=mask, enable_gqa=True)
            except (TypeError, RuntimeError):
                rep = self.n_heads // self.n_kv
                k = k.repeat_interleave(rep, dim=1)
                v = v.repeat_interleave(rep, dim=1)
        return F.scaled_dot_product_attention(q, k, v, attn_mask=mask)

    def forward(self, x, cos, sin, cache=None, layer=0):
        import os

        B, S, _ = x.shape
        hd = self.cfg.head_dim
        qkv = self.wqkv(x)
        q, k, v = qkv.split(
            [self.n_heads * hd, self.n_kv * hd, self.n_kv * hd], dim=-1
        )
        if cache is not None:
            # Decode path: cos/sin come pre-sliced at the cache position;
            # prefill (pos==0) is causal over the chunk, decode steps (S==1)
            # attend over the whole cache.
            q = ops.rope(q.view(B, S, self.n_heads, hd), cos, sin)
            k = ops.rope(k.view(B, S, self.n_kv, hd), cos, sin)
            v = v.view(B, S, self.n_kv, hd)
            q, k, v = (t.transpose(1, 2) for t in (q, k, v))
            kf, vf = cache.append(layer, k, v)
            if cache.pos == 0:
                o = self._sdpa(q, kf, vf, causal=True)
            elif S == 1:
                o = self._sdpa(q, kf, vf, causal=False)
            else:
                # chunked prefill: query row i (position pos+i) attends
                # keys [0, pos+i] — causal with a column offset
                P = cache.pos
                L = P + S
                ar = torch.arange(L, device=x.device)
                mask = (ar.view(1, L) <= (P + torch.arange(
                    S, device=x.device)).view(S, 1)).view(1, 1, S, L)
                o = self._sdpa_masked(q, kf, vf, mask)
            return self.wo(o.transpose(1, 2).reshape(B, S, -1))
        attn_impl = os.environ.get("KT_ATTN", "ck")
        # The v3 FMHA kernel wants Q pre-scaled by softmax_scale*log2e (its
        # LSE contract, ops/hip/bindings.cpp) — fold that into the RoPE
        # kernel's fp32 output scaling so it costs nothing. Decide BEFORE
        # RoPE runs.
        use_v3 = (attn_impl == "ck"
                  and ops.flash_attention_v3_supported(S, hd, x.device, x.dtype))
        q_oscale = (hd ** -0.5) * ops.LOG2E if use_v3 else 1.0
        q = ops.rope(q.view(B, S, self.n_heads, hd), cos, sin, oscale=q_oscale)
        k = ops.rope(k.view(B, S, self.n_kv, hd), cos, sin)
        v = v.view(B, S, self.n_kv, hd)
        q, k, v = (t.transpose(1, 2) for t in (q, k, v))
        if use_v3:
            # Fastest path: AITER-schedule CK v3 fwd (723 TF on the Llama
            # shape) + AITER asm bwd, zero transpose copies, zero extra
            # elementwise passes (scale folded into RoPE above).
            o = ops.flash_attention(q, k, v, impl="v3")
            return self.wo(o.transpose(1, 2).reshape(B, S, -1))
        if (attn_impl in ("ck", "custom")
                and ops.flash_attention_supported(q, k, v, True)):
            # CK-tile FMHA fwd + AITER asm bwd (KT_ATTN=torch for
            # the SDPA path, KT_ATTN=custom for the in-tree rocWMMA kernel).
            # The CK path is stride-aware: the permuted [B,S,H,D] views go
            # in directly and O comes back in the same layout, so the whole
            # attention block runs without a single transpose copy.
            o = ops.flash_attention(
                q, k, v, impl="wmma" if attn_impl == "custom" else "ck")
            return self.wo(o.transpose(1, 2).reshape(B, S, -1))
        o = self._sdpa(q, k, v, causal=True)
        o = o.transpose(1, 2).reshape(B, S, -1)
        return self.wo(o)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.w_gate_up = nn.Linear(cfg.dim, 2 * cfg.intermediate, bias=False)
        self.w_down = nn.Linear(cfg.intermediate, cfg.dim, bias=False)

    def forward(self, x):
        return self.w_down(ops.swiglu(self.w_gate_up(x)))


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.attn = Attention(cfg)
        self.mlp_norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.mlp = MLP(cfg)

    def forward(self, x, cos, sin):
        x = x + self.attn(self.attn_norm(x), cos, sin)
        x = x + self.mlp(self.mlp_norm(x))
        return x


class Llama(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.dim)
        self.layers = nn.ModuleList(Block(cfg) for _ in range(cfg.n_layers))
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps)
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
        cos, sin = ops.precompute_rope(cfg.max_seq_len, cfg.head_dim, cfg.rope_base)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.gradient_checkpointing = False
        self.reset_parameters()

    def gradient_checkpointing_enable(self, enabled=True):
        """Per-layer activation checkpointing: store only the residual
        stream between layers and recompute each block in backward (frees
        the attn/MLP activations — the lever for long-context or
        large-batch runs past 288 GB)."""
        self.gradient_checkpointing = enabled
        return self

    def _apply(self, fn, recurse=True):
        # Keep the RoPE tables fp32 across model.to(bf16): recompute rather
        # than round-trip through bf16 (precision).
        ret = super()._apply(fn, recurse)
        if self.rope_cos.dtype != torch.float32:
            cos, sin = ops.precompute_rope(
                self.cfg.max_seq_len, self.cfg.head_dim, self.cfg.rope_base,
                device=self.rope_cos.device,
            )
            self.rope_cos, self.rope_sin = cos, sin
        return ret

    def reset_parameters(self):
        std = self.cfg.init_std
        for m in self.modules():
            if isinstance(m, (nn.Linear, nn.Embedding)):
                nn.init.normal_(m.weight, mean=0.0, std=std)
            elif isinstance(m, RMSNorm):
                nn.init.ones_(m.weight)

    def forward(self, tokens):
        """Pre-norm transformer with the residual adds fused into the norm
        kernels: the residual stream is carried as (res, delta) where delta
        is the last branch output not yet added; ops.add_rmsnorm performs
        `res += delta` and the next norm in one pass."""
        S = tokens.shape[1]
        cos = self.rope_cos[:S]
        sin = self.rope_sin[:S]
        res = self.embed(tokens)
        delta = None
        eps = self.cfg.norm_eps
        if self.gradient_checkpointing and torch.is_grad_enabled():
            from torch.utils.checkpoint import checkpoint

            def block_step(layer, d, r):
                n1, r = ops.add_rmsnorm(d, r, layer.attn_norm.weight, eps)
                a = layer.attn(n1, cos, sin)
                n2, r = ops.add_rmsnorm(a, r, layer.mlp_norm.weight, eps)
                return layer.mlp(n2), r

            delta = torch.zeros_like(res)  # uniform (delta, res) carry
            for layer in self.layers:
                delta, res = checkpoint(block_step, layer, delta, res,
                                        use_reentrant=False)
            h, _ = ops.add_rmsnorm(delta, res, self.norm.weight, eps)
            return self.lm_head(h)
        for layer in self.layers:
            if delta is None:
                n1 = ops.rmsnorm(res, layer.attn_norm.weight, eps)
            else:
                n1, res = ops.add_rmsnorm(delta, res, layer.attn_norm.weight, eps)
            a = layer.attn(n1, cos, sin)
            n2, res = ops.add_rmsnorm(a, res, layer.mlp_norm.weight, eps)
            delta = layer.mlp(n2)
        if delta is None:
            h = ops.rmsnorm(res, self.norm.weight, eps)
        else:
            h, _ = ops.add_rmsnorm(delta, res, self.norm.weight, eps)
        return self.lm_head(h)

    def loss(self, tokens, targets):
        """Forward + fused CE (logits buffer is consumed by the fused op)."""
        logits = self.forward(tokens)
        return ops.fused_cross_entropy(logits, targets)

    def _forward_cached(self, tokens, cache):
        """One forward chunk through the KV cache (prefill or decode step);
        returns logits for the LAST position only."""
        S = tokens.shape[1]
        p = cache.pos
        cos = self.rope_cos[p:p + S]
        sin = self.rope_sin[p:p + S]
        h = self.embed(tokens)
        eps = self.cfg.norm_eps
        for i, layer in enumerate(self.layers):
            h = h + layer.attn(ops.rmsnorm(h, layer.attn_norm.weight, eps),
                               cos, sin, cache=cache, layer=i)
            h = h + layer.mlp(ops.rmsnorm(h, layer.mlp_norm.weight, eps))
        cache.advance(S)
        h = ops.rmsnorm(h[:, -1:].contiguous(), self.norm.weight, eps)
        return self.lm_head(h)[:, -1]

    @torch.no_grad()
    def generate(self, tokens, max_new_tokens, temperature=0.0, top_k=None,
                 stop_token=None, max_len=None, prefill_chunk=None):
        """Autoregressive decode with a preallocated KV cache: prefill
        (optionally in prefill_chunk-token pieces to bound latency spikes
        on long prompts), then single-token steps. temperature=0 is greedy
        argmax. Returns [B, prompt+new] token ids. (The reference ships no
        model code — this is the serving half of the flagship workload,
        the training half being bench.py's DDP step.)"""
        B, S0 = tokens.shape
        dev = tokens.device
        dtype = self.embed.weight.dtype
        cache = KVCache(self.cfg, B, max_len or min(self.cfg.max_seq_len,
                                                    S0 + max_new_tokens),
                        dev, dtype)
        out = [tokens]
        if prefill_chunk and prefill_chunk < S0:
            for s in range(0, S0, prefill_chunk):
                logits = self._forward_cached(
                    tokens[:, s:s + prefill_chunk], cache)
        else:
            logits = self._forward_cached(tokens, cache)
        for _ in range(max_new_tokens):
            if temperature > 0:
                lg = logits.float() / temperature
                if top_k:
                    kth = lg.topk(top_k, dim=-1).values[:, -1:]
                    lg = lg.masked_fill(lg < kth, float("-inf"))
                nxt = torch.multinomial(torch.softmax(lg, -1), 1)
            else:
                nxt = logits.argmax(-1, keepdim=True)
            out.append(nxt)
            if stop_token is not None and (nxt == stop_token).all():
                break
            logits = self._forward_cached(nxt, cache)
        return torch.cat(out, dim=1)
