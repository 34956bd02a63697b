"""Decorator + CLI workflow: `kt deploy examples/04_decorators_and_cli.py`
deploys every decorated callable; then `kt call <user>-embed --args '[..]'`,
`kt logs <user>-embed`, `kt teardown -p <user>`."""
import kubetorch_amd as kt


@kt.compute(cpus=2, memory="4Gi", inactivity_ttl="30m")
@kt.autoscale(min_scale=0, max_scale=4, target=10, metric="concurrency")
def embed(texts: list):
    return [hash(t) % 1000 for t in texts]


@kt.compute(gpus=8)
@kt.distribute("pytorch", workers=4)
def big_train(steps: int):
    import torch.distributed as dist

    from kubetorch_amd.parallel import init_distributed

    rank, world, _ = init_distributed()
    # ... your training loop ...
    if dist.is_initialized():
        dist.destroy_process_group()
    return rank
