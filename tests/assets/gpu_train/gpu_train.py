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
    def pull(self, delay=0.0):
        import os
        import time

        import kubetorch_amd as kt

        dest = torch.zeros(256, 256, device="cuda", dtype=torch.bfloat16)
        kt.get("rlgpu/w", dest)
        torch.cuda.synchronize()
        if delay:
            time.sleep(delay)  # simulated inference load (autoscale signal)
        return {"sum": float(dest.float().sum().item()),
                "pod": os.environ.get("POD_NAME", "?")}


def elastic_gpu_step(v, delay=0.0):
    """Distributed step for the GPU elastic drill: every rank does real
    CUDA work, then all ranks agree on the result through a per-call
    process group (gloo on a 1-GPU box: both worker pods share cuda:0, so
    RCCL can't host two ranks; the PG lifecycle is identical). The delay
    keeps the call in flight long enough to kill a pod mid-step."""
    import os
    import time

    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    dev = torch.device("cuda", 0)
    a = torch.full((512, 512), float(v), device=dev, dtype=torch.bfloat16)
    out = (a @ torch.eye(512, device=dev, dtype=torch.bfloat16)).float().mean()
    if delay:
        time.sleep(delay)
    if world > 1:
        if dist.is_initialized():
            dist.destroy_process_group()  # fresh group per call (elastic)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        t = torch.tensor([out.item()])
        dist.all_reduce(t)
        dist.destroy_process_group()
        return {"rank": rank, "world": world, "sum": float(t.item())}
    return {"rank": rank, "world": world, "sum": float(out.item())}


def rccl_group_lifecycle(cycles=3):
    """In-step RCCL group destroy/reinit on gfx950 (world=1 on a 1-GPU
    box): validates the create -> collective -> destroy -> recreate cycle
    the per-call elastic model relies on, with device tensors over the
    nccl(=RCCL) backend."""
    import torch.distributed as dist

    results = []
    for i in range(cycles):
        store = dist.TCPStore("127.0.0.1", 29650 + i, 1, True)
        dist.init_process_group("nccl", store=store, rank=0, world_size=1)
        t = torch.full((1024,), float(i + 1), device="cuda")
        dist.all_reduce(t)
        dist.barrier()
        results.append(float(t.sum().item()))
        dist.destroy_process_group()
        del store
    return results
