"""Distributed training: the SPMD launcher stands up torch.distributed over
RCCL/xGMI (one rank per MI355X GPU). The training function is ordinary
torch.distributed code — RANK/WORLD_SIZE/MASTER_* come from the launcher."""
import kubetorch_amd as kt


def train(steps: int = 50):
    import torch
    import torch.distributed as dist

    from kubetorch_amd.models import Llama, llama3_8b
    from kubetorch_amd.parallel import FlatDDP, init_distributed

    rank, world, local_rank = init_distributed()
    # production shape on MI355X; degrades to a tiny model on gloo/CPU so
    # the same file demos the full path without a GPU
    on_gpu = torch.cuda.is_available()
    if on_gpu:
        dev = torch.device("cuda", local_rank)
        cfg = llama3_8b(max_seq_len=4096)
    else:
        from kubetorch_amd.models import llama_tiny

        dev = torch.device("cpu")
        cfg = llama_tiny(max_seq_len=256)
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    with torch.device(dev):
        model = Llama(cfg)
    torch.set_default_dtype(prev)
    engine = FlatDDP(model, lr=1e-4, clip_norm=1.0)
    engine.broadcast_params(src=0)

    from kubetorch_amd.data import ShardedLoader, TokenDataset, synthetic_tokens
    from kubetorch_amd.parallel import warmup_cosine

    seq = cfg.max_seq_len
    ds = TokenDataset(synthetic_tokens(cfg.vocab_size, seq * (512 if on_gpu
                                                              else 16)), seq)
    loader = ShardedLoader(ds, batch=4 if on_gpu else 1, rank=rank,
                           world=world, device=dev)  # pinned H2D prefetch
    it = iter(loader)
    for step in range(steps):
        try:
            x, y = next(it)
        except StopIteration:
            loader.set_epoch(loader.epoch + 1)
            it = iter(loader)
            x, y = next(it)
        loss = model.loss(x, y)
        loss.backward()
        engine.step(lr=warmup_cosine(step, 1e-4, 10, steps))
        if rank == 0 and step % 10 == 0:
            print(f"step {step}: loss {loss.item():.4f} "
                  f"grad_norm {engine.last_grad_norm:.2f}")
    if dist.is_initialized():
        dist.destroy_process_group()
    return {"rank": rank, "final_loss": loss.item()}


if __name__ == "__main__":
    import torch

    if torch.cuda.is_available():
        compute = kt.Compute(gpus=8, memory="640Gi").distribute(
            "pytorch", workers=2)     # 2 pods x 8 ranks over RCCL/xGMI
    else:
        compute = kt.Compute(cpus=1).distribute(
            "pytorch", workers=2, num_proc=1)  # local demo: 2 ranks, gloo
    remote = kt.fn(train).to(compute)
    import os

    steps = int(os.environ.get("KT_EX_STEPS", "50"))
    results = remote(steps, kt_timeout=3600)
    print(results)
