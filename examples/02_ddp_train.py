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
    dev = torch.device("cuda", local_rank)
    cfg = llama3_8b(max_seq_len=4096)
    prev = torch.get_default_dtype()
    torch.set_default_dtype(torch.bfloat16)
    with torch.device(dev):
        model = Llama(cfg)
    torch.set_default_dtype(prev)
    engine = FlatDDP(model, lr=1e-4, clip_norm=1.0)
    engine.broadcast_params(src=0)

    from kubetorch_amd.data import ShardedLoader, TokenDataset, synthetic_tokens
    from kubetorch_amd.parallel import warmup_cosine

    ds = TokenDataset(synthetic_tokens(cfg.vocab_size, 4096 * 512), 4096)
    loader = ShardedLoader(ds, batch=4, rank=rank, world=world,
                           device=dev)  # pinned one-ahead H2D prefetch
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
    remote = kt.fn(train).to(
        kt.Compute(gpus=8, memory="640Gi").distribute("pytorch", workers=2)
    )
    results = remote(50, kt_timeout=3600)   # 2 pods x 8 ranks
    print(results)
