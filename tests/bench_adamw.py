"""Isolated fused-AdamW bandwidth probe: one 256 MB bucket, no overlap.
22 B of HBM traffic per param (p/g bf16 r + p w, m/v fp32 rw).
PYTHONPATH=. python tests/bench_adamw.py"""
import time

import torch

from kubetorch_amd import ops

n = 134_217_728  # 256 MB of bf16 params
p = torch.randn(n, dtype=torch.bfloat16, device="cuda")
g = torch.randn(n, dtype=torch.bfloat16, device="cuda")
m = torch.zeros(n, dtype=torch.float32, device="cuda")
v = torch.zeros(n, dtype=torch.float32, device="cuda")

def run():
    ops.adamw_(p, g, m, v, 1e-4, 0.9, 0.95, 1e-8, 0.1, 10, 1.0)

for _ in range(3):
    run()
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 20
for _ in range(iters):
    run()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
gb = n * 22 / 1e9
print(f"adamw 256MB bucket: {dt*1e3:.3f} ms  {gb/dt/1000:.2f} TB/s "
      f"({n/dt/1e9:.1f} Gelem/s)")
