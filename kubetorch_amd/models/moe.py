"""Mixture-of-Experts MLP for the Llama family (Mixtral-style top-k).

Routing + dispatch is torch (memory-bound gather/scatter); each expert's
FFN uses the same fused SwiGLU op as the dense MLP. Two placements:

  * dense (default): all experts on every rank — token-grouped batched
    expert GEMMs, DP-compatible as-is.
  * expert parallel (ep_group): experts sharded across the group's ranks.
    The EP group is sequence-replicated (every rank in it runs the same
    tokens — the TP-style placement; compose with DP ACROSS groups, not
    inside one): each rank computes its local experts and the partial
    outputs are all-reduce summed. Correct on any backend (gloo CI
    included); the all_to_all token exchange for data-sharded EP (less
    traffic when top_k << n_experts/world) is a round-2 swap kept behind
    the same interface.

The reference ships no model code — this extends the
flagship family the launchers run, and is the EP workload the SPMD
launcher's process groups exist for.
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from kubetorch_amd import ops


class Expert(nn.Module):
    def __init__(self, dim, intermediate):
        super().__init__()
        self.w_gate_up = nn.Linear(dim, 2 * intermediate, bias=False)
        self.w_down = nn.Linear(intermediate, dim, bias=False)

    def forward(self, x):
        return self.w_down(ops.swiglu(self.w_gate_up(x)))


class MoEMLP(nn.Module):
    """Drop-in replacement for the dense MLP block.

    n_experts total experts, top_k per token, softmax-normalized over the
    selected k (Mixtral convention). With ep_group, rank r owns experts
    [r*n_local, (r+1)*n_local).
    """

    def __init__(self, cfg, n_experts=8, top_k=2, ep_group=None):
        super().__init__()
        import torch.distributed as dist

        self.dim = cfg.dim
        self.n_experts = n_experts
        self.top_k = top_k
        self.router = nn.Linear(cfg.dim, n_experts, bias=False)
        self.ep_group = ep_group
        if ep_group is not None and dist.is_initialized():
            self.ep_world = dist.get_world_size(ep_group)
            self.ep_rank = dist.get_rank(ep_group)
        else:
            self.ep_world, self.ep_rank = 1, 0
        if n_experts % self.ep_world:
            raise ValueError("n_experts must divide by the EP world size")
        self.n_local = n_experts // self.ep_world
        self.local_offset = self.ep_rank * self.n_local
        self.experts = nn.ModuleList(
            Expert(cfg.dim, cfg.intermediate) for _ in range(self.n_local))

    def forward(self, x):
        B, S, D = x.shape
        flat = x.reshape(-1, D)
        logits = self.router(flat)                         # [T, E]
        weights, sel = logits.topk(self.top_k, dim=-1)     # [T, k]
        weights = torch.softmax(weights.float(), dim=-1).to(x.dtype)
        if self.ep_world == 1:
            out = self._dispatch_local(flat, sel, weights, base=0)
        else:
            out = self._dispatch_ep(flat, sel, weights)
        return out.reshape(B, S, D)

    def _dispatch_local(self, flat, sel, weights, base, out=None):
        """Compute this module's experts for their assigned tokens and
        weighted-scatter into out. base maps local expert i -> global id."""
        if out is None:
            out = torch.zeros_like(flat)
        for i, expert in enumerate(self.experts):
            eid = base + i
            tok, kth = (sel == eid).nonzero(as_tuple=True)  # tokens; which k
            if tok.numel() == 0:
                continue
            y = expert(flat[tok])
            out.index_add_(0, tok, y * weights[tok, kth].unsqueeze(-1))
        return out

    def _dispatch_ep(self, flat, sel, weights):
        """Sequence-replicated EP: every rank in the group runs the same
        tokens (router weights identical across the group), computes only
        its local experts, then the partial outputs are summed."""
        import torch.distributed as dist

        out = self._dispatch_local(flat, sel, weights, base=self.local_offset)
        dist.all_reduce(out, op=dist.ReduceOp.SUM, group=self.ep_group)
        return out

    def aux_load_balance_loss(self, x):
        """Switch-style load-balance auxiliary loss (fraction-of-tokens x
        mean-router-prob per expert, scaled by n_experts)."""
        flat = x.reshape(-1, self.dim)
        probs = torch.softmax(self.router(flat).float(), dim=-1)   # [T, E]
        _, sel = probs.topk(self.top_k, dim=-1)
        counts = torch.zeros(self.n_experts, device=x.device)
        counts.scatter_add_(0, sel.reshape(-1),
                            torch.ones(sel.numel(), device=x.device))
        frac = counts / max(1, sel.numel())
        return self.n_experts * (frac * probs.mean(0)).sum()


def _init_linear(linear, std, seed):
    g = torch.Generator().manual_seed(seed)
    with torch.no_grad():
        w = torch.empty(linear.weight.shape, dtype=torch.float32)
        w.normal_(0.0, std, generator=g)
        linear.weight.copy_(w.to(linear.weight.dtype))


def convert_to_moe(model, n_experts=8, top_k=2, ep_group=None, seed=0):
    """Swap every Block's dense MLP for a MoEMLP. Weights are seeded per
    GLOBAL expert id, so every EP sharding of the same (seed, n_experts)
    holds identical experts — a 2-rank EP run computes exactly what the
    single-rank dense placement computes (tested)."""
    std = model.cfg.init_std
    p = next(model.parameters())
    for li, layer in enumerate(model.layers):
        moe = MoEMLP(model.cfg, n_experts=n_experts, top_k=top_k,
                     ep_group=ep_group)
        _init_linear(moe.router, std, seed * 100003 + li)
        for i, ex in enumerate(moe.experts):
            gid = moe.local_offset + i
            _init_linear(ex.w_gate_up, std,
                         seed * 100003 + li * 1009 + gid * 2 + 7)
            _init_linear(ex.w_down, std,
                         seed * 100003 + li * 1009 + gid * 2 + 8)
        layer.mlp = moe.to(device=p.device, dtype=p.dtype)
    return model
