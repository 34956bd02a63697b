"""GPU dispatch assets: a single-GPU train fn and an RL trainer/inference
pair with CUDA tensors (run via the local driver on a GPU box)."""
import torch


def train_tiny_llama(steps=3):
    from kubetorch_amd.models import Llama, llama_tiny
    from kubetorch_amd.parallel import FlatDDP

    dev = torch.device("cuda", 0)
    cfg = llama_tiny()
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    try:
        with torch.device(dev):
            model = Llama(cfg)
    finally:
        torch.set_default_dtype(prev)
    eng = FlatDDP(model, lr=1e-3, bucket_mb=8)
    x = torch.randint(0, cfg.vocab_size, (2, 128), device=dev)
    y = torch.randint(0, cfg.vocab_size, (2, 128), device=dev)
    losses = []
    for _ in range(steps):
        loss = model.loss(x, y)
        loss.backward()
        eng.step()
        losses.append(loss.item())
    torch.cuda.synchronize()
    return losses


class CudaTrainer:
    def __init__(self):
        torch.manual_seed(3)
        self.w = torch.randn(256, 256, device="cuda", dtype=torch.bfloat16)

    def publish(self):
        import kubetorch_amd as kt

        self.w += 1.0
        kt.put("rlgpu/w", self.w)
        return float(self.w.float().sum().item())


class CudaInference:
    def pull(self):
        import kubetorch_amd as kt

        dest = torch.zeros(256, 256, device="cuda", dtype=torch.bfloat16)
        kt.get("rlgpu/w", dest)
        torch.cuda.synchronize()
        return float(dest.float().sum().item())
