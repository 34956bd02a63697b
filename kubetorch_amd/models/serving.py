"""Continuous-batching generation engine for the Llama family.

`Llama.generate` decodes one fixed batch to completion; a serving pod needs
requests to join and leave the batch as they arrive/finish (continuous
batching). This engine keeps a slot-per-sequence KV cache
([max_batch, n_kv, max_len, hd] per layer, plus a per-slot length vector)
and runs one batched decode step for every active slot per `step()`;
prefills fill free slots as requests arrive.

Per-slot positions make the stock forward unusable (RoPE offset and the
causal horizon differ per row), so the decode step is written against the
same ops the model uses (rmsnorm/swiglu kernels, SDPA with a length mask),
keeping numerics aligned with `Llama.generate`.
"""
from dataclasses import dataclass, field

import torch

from kubetorch_amd import ops


@dataclass
class _Request:
    rid: int
    prompt: torch.Tensor          # [S0] int64
    max_new_tokens: int
    temperature: float = 0.0
    stop_token: int = None
    out: list = field(default_factory=list)
    slot: int = -1


class BatchedGenerator:
    def __init__(self, model, max_batch=8, max_len=None, device=None,
                 prefill_chunk=None, graph=None):
        self.model = model
        cfg = model.cfg
        self.cfg = cfg
        self.max_batch = max_batch
        self.prefill_chunk = prefill_chunk  # bound prefill latency spikes
        self.max_len = max_len or cfg.max_seq_len
        p = next(model.parameters())
        self.device = device or p.device
        self.dtype = p.dtype
        shape = (max_batch, cfg.n_kv_heads, self.max_len, cfg.head_dim)
        self.k = [torch.zeros(shape, device=self.device, dtype=self.dtype)
                  for _ in range(cfg.n_layers)]
        self.v = [torch.zeros(shape, device=self.device, dtype=self.dtype)
                  for _ in range(cfg.n_layers)]
        self.lens = torch.zeros(max_batch, dtype=torch.long, device=self.device)
        self.slots = [None] * max_batch   # slot -> _Request
        self.pending = []
        self.finished = {}
        self._next_rid = 0
        # hipGraph-captured decode step (the single-token step is
        # launch-bound: ~10 kernels x n_layers per step; one graph replay
        # removes the per-launch host cost — profiles/ROUND2.md lever 5).
        # Shape-static by construction: full max_batch every step, fixed
        # max_len horizon, static in/out buffers. KT_DECODE_GRAPH=0 or
        # graph=False forces the eager path.
        if graph is None:
            import os

            graph = (self.device.type == "cuda"
                     and os.environ.get("KT_DECODE_GRAPH", "1") != "0")
        if graph and any(hasattr(m, "ep_world") for m in model.modules()):
            graph = False  # MoE top-k dispatch is shape-dynamic: the
            # variable-size expert index_selects cannot be graph-captured
        self._use_graph = bool(graph)
        self._graphs = {}      # L bucket -> (CUDAGraph, logits buffer)
        self._g_pool = None    # shared capture mempool across buckets
        self._g_toks = None
        self._g_act = None
        self._g_rows = torch.arange(max_batch, device=self.device)
        # host mirror of per-slot lengths: picks the smallest captured
        # horizon bucket without a device sync
        self._host_lens = [0] * max_batch

    # -- client API ----------------------------------------------------------
    def submit(self, prompt_ids, max_new_tokens=32, temperature=0.0,
               stop_token=None):
        rid = self._next_rid
        self._next_rid += 1
        prompt = torch.as_tensor(prompt_ids, dtype=torch.long,
                                 device=self.device).reshape(-1)
        if prompt.numel() + max_new_tokens > self.max_len:
            raise ValueError("prompt + max_new_tokens exceeds max_len")
        self.pending.append(_Request(rid, prompt, max_new_tokens,
                                     temperature, stop_token))
        return rid

    @property
    def has_work(self):
        return bool(self.pending) or any(s is not None for s in self.slots)

    def run(self):
        """Drive to completion; returns {rid: token list (prompt + new)}."""
        while self.has_work:
            self.step()
        out, self.finished = self.finished, {}
        return out

    # -- engine --------------------------------------------------------------
    @torch.no_grad()
    def step(self):
        """One engine iteration: admit pending prompts into free slots
        (prefill — same-length prompts share one batched forward), then
        one batched decode step for every active slot."""
        admitted = []
        for slot in range(self.max_batch):
            if self.slots[slot] is None and self.pending:
                req = self.pending.pop(0)
                req.slot = slot
                self.slots[slot] = req
                admitted.append(req)
        if admitted:
            by_len = {}
            for req in admitted:
                by_len.setdefault(req.prompt.numel(), []).append(req)
            for group in by_len.values():
                if len(group) > 1 and not self.prefill_chunk:
                    self._prefill_batch(group)
                else:
                    for req in group:
                        self._prefill(req)
        active = [s for s in range(self.max_batch) if self.slots[s] is not None]
        if active:
            self._decode(active)

    def _prefill_batch(self, group):
        """One forward for G same-length prompts (a 1-at-a-time prefill is
        GEMM-starved: [1, S0] activations; [G, S0] runs the same layer
        GEMMs at G-fold arithmetic intensity)."""
        from kubetorch_amd.models.llama import KVCache

        S0 = group[0].prompt.numel()
        G = len(group)
        batch = torch.stack([req.prompt for req in group])
        cache = KVCache(self.cfg, G, S0, self.device, self.dtype)
        logits = self.model._forward_cached(batch, cache)
        for j, req in enumerate(group):
            for i in range(self.cfg.n_layers):
                self.k[i][req.slot, :, :S0] = cache.k[i][j, :, :S0]
                self.v[i][req.slot, :, :S0] = cache.v[i][j, :, :S0]
            self.lens[req.slot] = S0
            self._host_lens[req.slot] = S0
            req.out = req.prompt.tolist()
            self._emit(req, logits[j])

    def _prefill(self, req):
        """Run the prompt through a throwaway per-request cache, then copy
        the KV rows into this request's slot."""
        from kubetorch_amd.models.llama import KVCache

        S0 = req.prompt.numel()
        cache = KVCache(self.cfg, 1, S0, self.device, self.dtype)
        pc = self.prefill_chunk
        if pc and pc < S0:
            for s0 in range(0, S0, pc):
                logits = self.model._forward_cached(
                    req.prompt[s0:s0 + pc].view(1, -1), cache)
        else:
            logits = self.model._forward_cached(req.prompt.view(1, -1), cache)
        for i in range(self.cfg.n_layers):
            self.k[i][req.slot, :, :S0] = cache.k[i][0, :, :S0]
            self.v[i][req.slot, :, :S0] = cache.v[i][0, :, :S0]
        self.lens[req.slot] = S0
        self._host_lens[req.slot] = S0
        req.out = req.prompt.tolist()
        self._emit(req, logits[0])

    def _sample(self, req, logits):
        if req.temperature > 0:
            probs = torch.softmax(logits.float() / req.temperature, -1)
            return int(torch.multinomial(probs, 1))
        return int(logits.argmax(-1))

    def _emit(self, req, logits):
        nxt = self._sample(req, logits)
        req.out.append(nxt)
        req.next_tok = nxt
        done = (len(req.out) - req.prompt.numel() >= req.max_new_tokens
                or (req.stop_token is not None and nxt == req.stop_token))
        if done:
            self.finished[req.rid] = req.out
            self.slots[req.slot] = None
            self.lens[req.slot] = 0
            self._host_lens[req.slot] = 0

    def _decode(self, active):
        if self._use_graph:
            return self._decode_graph(active)
        return self._decode_eager(active)

    # -- hipGraph path -------------------------------------------------------
    def _decode_math(self, L):
        """One full-batch decode step over horizon L as a pure function of
        the static buffers (_g_toks, lens, _g_act) — capturable: fixed
        shapes, no host syncs, no data-dependent control flow. Inactive
        slots decode garbage (their mask exposes only cache row 0) and are
        ignored at emit time; their cache rows are overwritten by the next
        prefill."""
        cfg, m = self.cfg, self.model
        B = self.max_batch
        hd, Hq, Hkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
        eps = cfg.norm_eps
        lens = self.lens
        cos = m.rope_cos[lens].view(B, 1, 1, hd // 2)
        sin = m.rope_sin[lens].view(B, 1, 1, hd // 2)
        ar = torch.arange(L, device=self.device)
        mask = (ar.view(1, L) <= lens.view(B, 1)).view(B, 1, 1, L)
        h = m.embed(self._g_toks).view(B, 1, cfg.dim)
        for i, layer in enumerate(m.layers):
            n1 = ops.rmsnorm(h, layer.attn_norm.weight, eps)
            qkv = layer.attn.wqkv(n1)
            q, k, v = qkv.split([Hq * hd, Hkv * hd, Hkv * hd], dim=-1)
            q = self._rope1(q.view(B, 1, Hq, hd), cos, sin)
            k = self._rope1(k.view(B, 1, Hkv, hd), cos, sin)
            v = v.view(B, 1, Hkv, hd)
            self.k[i][self._g_rows, :, lens] = k[:, 0]
            self.v[i][self._g_rows, :, lens] = v[:, 0]
            o = layer.attn._sdpa_masked(q.transpose(1, 2),
                                        self.k[i][:, :, :L],
                                        self.v[i][:, :, :L], mask)
            h = h + layer.attn.wo(o.transpose(1, 2).reshape(B, 1, -1))
            n2 = ops.rmsnorm(h, layer.mlp_norm.weight, eps)
            h = h + layer.mlp(n2)
        logits = m.lm_head(ops.rmsnorm(h, m.norm.weight, eps))[:, -1]
        self.lens.add_(self._g_act)  # active slots advance one position
        return logits

    def _bucket_for(self, need):
        """Smallest power-of-two horizon >= need (>=128), capped at
        max_len. The per-bucket graphs keep attention reads proportional
        to the actual active horizon instead of the full cache."""
        L = 128
        while L < need:
            L *= 2
        return min(L, self.max_len)

    def _ensure_graph(self, L):
        g = self._graphs.get(L)
        if g is not None:
            return g
        if self._g_toks is None:
            self._g_toks = torch.zeros(self.max_batch, dtype=torch.long,
                                       device=self.device)
            self._g_act = torch.zeros(self.max_batch, dtype=torch.long,
                                      device=self.device)
            self._g_pool = torch.cuda.graph_pool_handle()
        lens_snapshot = self.lens.clone()
        for _ in range(2):  # warm up allocator/workspaces pre-capture
            self._decode_math(L)
        self.lens.copy_(lens_snapshot)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self._g_pool):
            logits = self._decode_math(L)
        self._graphs[L] = (graph, logits)
        return self._graphs[L]

    def _decode_graph(self, active):
        need = max(self._host_lens[s] for s in active) + 1
        graph, logits = self._ensure_graph(self._bucket_for(need))
        toks = [0] * self.max_batch
        act = [0] * self.max_batch
        for s in active:
            toks[s] = self.slots[s].next_tok
            act[s] = 1
            self._host_lens[s] += 1
        self._g_toks.copy_(torch.tensor(toks, dtype=torch.long),
                           non_blocking=True)
        self._g_act.copy_(torch.tensor(act, dtype=torch.long),
                          non_blocking=True)
        graph.replay()
        for s in active:
            self._emit(self.slots[s], logits[s])

    # -- eager path (CPU, or KT_DECODE_GRAPH=0) ------------------------------
    def _decode_eager(self, active):
        cfg = self.cfg
        m = self.model
        idx = torch.tensor(active, device=self.device)
        toks = torch.tensor([self.slots[s].next_tok for s in active],
                            device=self.device)
        lens = self.lens[idx]                      # position of the new token
        B = len(active)
        hd, Hq, Hkv = cfg.head_dim, cfg.n_heads, cfg.n_kv_heads
        eps = cfg.norm_eps
        # per-slot RoPE rows (fp32 tables)
        cos = m.rope_cos[lens].view(B, 1, 1, hd // 2)
        sin = m.rope_sin[lens].view(B, 1, 1, hd // 2)
        L = int(lens.max().item()) + 1             # longest horizon this step
        # attention mask over the padded cache: slot b sees [0, lens[b]]
        ar = torch.arange(L, device=self.device)
        mask = (ar.view(1, L) <= lens.view(B, 1)).view(B, 1, 1, L)

        h = m.embed(toks).view(B, 1, cfg.dim)
        for i, layer in enumerate(m.layers):
            n1 = ops.rmsnorm(h, layer.attn_norm.weight, eps)
            qkv = layer.attn.wqkv(n1)
            q, k, v = qkv.split([Hq * hd, Hkv * hd, Hkv * hd], dim=-1)
            q = self._rope1(q.view(B, 1, Hq, hd), cos, sin)
            k = self._rope1(k.view(B, 1, Hkv, hd), cos, sin)
            v = v.view(B, 1, Hkv, hd)
            # append to the slot caches at each slot's own position
            # (advanced indexing: [idx, :, lens] -> [B, n_kv, hd] rows)
            self.k[i][idx, :, lens] = k[:, 0]
            self.v[i][idx, :, lens] = v[:, 0]
            kf = self.k[i][idx, :, :L]
            vf = self.v[i][idx, :, :L]
            o = layer.attn._sdpa_masked(q.transpose(1, 2), kf, vf, mask)
            h = h + layer.attn.wo(o.transpose(1, 2).reshape(B, 1, -1))
            n2 = ops.rmsnorm(h, layer.mlp_norm.weight, eps)
            h = h + layer.mlp(n2)
        logits = m.lm_head(ops.rmsnorm(h, m.norm.weight, eps))[:, -1]
        self.lens[idx] += 1
        for j, s in enumerate(active):
            self._emit(self.slots[s], logits[j])

    @staticmethod
    def _rope1(x, cos, sin):
        # single-position rotate-half, per-row tables (fp32 math, same
        # formula as ops._rope_ref so numerics match the prefill path)
        D = x.shape[-1]
        xf = x.float()
        x1, x2 = xf[..., :D // 2], xf[..., D // 2:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).to(x.dtype)
