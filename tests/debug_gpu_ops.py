"""One-shot GPU debug: compare each HIP op against its CPU fp32 reference
and probe the tiny model layer by layer for NaNs. Not a pytest file."""
import torch
import torch.nn.functional as F

from kubetorch_amd import ops

BF16 = torch.bfloat16
torch.manual_seed(0)


def diff(name, a, b):
    a, b = a.float().cpu(), b.float().cpu()
    d = (a - b).abs()
    denom = b.abs().clamp(min=1e-3)
    print(f"{name:24s} max_abs={d.max().item():.4e} max_rel={(d/denom).max().item():.4e} "
          f"nan_a={a.isnan().any().item()} nan_b={b.isnan().any().item()}")


# rmsnorm
N, H = 33, 256
x = torch.randn(N, H, dtype=BF16, device="cuda")
w = torch.randn(H, dtype=BF16, device="cuda")
y, ir = ops._ext().rmsnorm_fwd(x, w, 1e-5)
yr, irr = ops._rmsnorm_ref_fwd(x.cpu(), w.cpu(), 1e-5)
diff("rmsnorm fwd", y, yr)
diff("rmsnorm invrms", ir, irr)
dy = torch.randn_like(x)
dx, dw = ops._ext().rmsnorm_bwd(dy, x, w, ir)
xr = x.cpu().float().requires_grad_(True)
wr = w.cpu().float().requires_grad_(True)
yy = xr * torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5) * wr
yy.backward(dy.cpu().float())
diff("rmsnorm dx", dx, xr.grad)
diff("rmsnorm dw", dw, wr.grad)

# rope
B, S, Hh, D = 2, 64, 4, 64
cos, sin = ops.precompute_rope(S, D, base=10000.0, device="cuda")
xq = torch.randn(B, S, Hh, D, dtype=BF16, device="cuda")
o = ops._ext().rope(xq.view(B * S, Hh, D), cos, sin, S, 1.0).view(B, S, Hh, D)
orf = ops._rope_ref(xq.cpu(), cos.cpu(), sin.cpu(), 1.0)
diff("rope fwd", o, orf)

# swiglu
gu = torch.randn(65, 256, dtype=BF16, device="cuda")
diff("swiglu fwd", ops._ext().swiglu_fwd(gu), ops._swiglu_ref_fwd(gu.cpu()))
do = torch.randn(65, 128, dtype=BF16, device="cuda")
dgu = ops._ext().swiglu_bwd(do, gu)
gr = gu.cpu().float().requires_grad_(True)
I = 128
yy = F.silu(gr[..., :I]) * gr[..., I:]
yy.backward(do.cpu().float())
diff("swiglu bwd", dgu, gr.grad)

# cross entropy
Nr, V = 128, 1024
logits = torch.randn(Nr, V, dtype=BF16, device="cuda") * 4
tg = torch.randint(0, V, (Nr,), device="cuda")
lcl = logits.clone()
loss = ops._ext().cross_entropy_fwd_(lcl, tg, 1.0, -100)
ref_loss = F.cross_entropy(logits.float().cpu(), tg.cpu(), reduction="none")
diff("ce loss", loss, ref_loss)
lr32 = logits.float().cpu().requires_grad_(True)
F.cross_entropy(lr32, tg.cpu(), reduction="sum").backward()
diff("ce grad", lcl, lr32.grad)

# adamw
n = 1003
p = torch.randn(n, dtype=BF16, device="cuda")
g = torch.randn(n, dtype=BF16, device="cuda")
m = torch.zeros(n, device="cuda")
v = torch.zeros(n, device="cuda")
pc, mc, vc = p.cpu().clone(), m.cpu().clone(), v.cpu().clone()
ops._ext().adamw_(p, g, m, v, 1e-2, 0.9, 0.95, 1e-8, 0.1, 1, 0.5)
ops.adamw_(pc, g.cpu(), mc, vc, 1e-2, 0.9, 0.95, 1e-8, 0.1, 1, 0.5)
diff("adamw p", p, pc)
diff("adamw m", m, mc)
diff("adamw v", v, vc)

# tiny model NaN probe
from kubetorch_amd.models import Llama, llama_tiny

cfg = llama_tiny()
prev = torch.get_default_dtype()
torch.set_default_dtype(torch.bfloat16)
with torch.device("cuda"):
    model = Llama(cfg)
torch.set_default_dtype(prev)
xt = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda")
h = model.embed(xt)
print("embed nan:", h.isnan().any().item())
cos_t = model.rope_cos[:128]
sin_t = model.rope_sin[:128]
for i, layer in enumerate(model.layers):
    h = layer(h, cos_t, sin_t)
    print(f"layer{i} nan: {h.isnan().any().item()} absmax={h.float().abs().max().item():.3f}")
logits = model.lm_head(model.norm(h))
print("logits nan:", logits.isnan().any().item(), "absmax:", logits.float().abs().max().item())
yt = torch.randint(0, cfg.vocab_size, (2, 128), device="cuda")
loss = ops.fused_cross_entropy(logits.clone(), yt)
print("tiny loss:", loss.item())
loss2 = F.cross_entropy(logits.float().view(-1, cfg.vocab_size), yt.view(-1))
print("ref loss:", loss2.item())
print("done")
