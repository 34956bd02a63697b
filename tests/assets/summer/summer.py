import os


def summer(a, b):
    print(f"summing {a}+{b}")
    return a + b


def rank_env():
    return {
        "rank": int(os.environ.get("RANK", "-1")),
        "world_size": int(os.environ.get("WORLD_SIZE", "-1")),
        "local_rank": int(os.environ.get("LOCAL_RANK", "-1")),
        "node_rank": int(os.environ.get("NODE_RANK", "-1")),
        "master_addr": os.environ.get("MASTER_ADDR"),
        "pod_ips": os.environ.get("POD_IPS", ""),
    }


def boom():
    raise ValueError("intentional failure")


def gloo_allreduce(x):
    """DDP-style allreduce through the SPMD env contract (gloo on CPU)."""
    import torch
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group("gloo")
    t = torch.tensor([float(x)])
    dist.all_reduce(t)
    out = t.item()
    rank = dist.get_rank()
    dist.destroy_process_group()
    return {"rank": rank, "sum": out}


class Counter:
    def __init__(self, start=0):
        self.value = start

    def add(self, n=1):
        self.value += n
        return self.value

    def get(self):
        return self.value


def slow_echo(x, delay=10):
    import time

    time.sleep(delay)
    return x


def deploy_child_and_call():
    """In-cluster client: a deployed fn that deploys + calls ANOTHER fn
    (pod-from-pod; reference: test_pod_from_pod.py)."""
    import kubetorch_amd as kt
    from tests_child_helper import double  # resolved via synced workdir

    child = kt.fn(double, name="child-of-pod").to(kt.Compute(cpus=1))
    try:
        return child(21)
    finally:
        child.teardown()


def pod_name():
    return os.environ.get("POD_NAME", "unknown")


def read_env(key):
    return os.environ.get(key)


def read_secret_file(secret_name, key):
    d = os.environ.get(f"KT_SECRET_MOUNT_{secret_name}")
    if not d:
        return None
    with open(os.path.join(d, key)) as f:
        return f.read()


def write_volume(vol_env, fname, content):
    d = os.environ.get(vol_env)
    if not d:
        return None
    with open(os.path.join(d, fname), "w") as f:
        f.write(content)
    return d


def read_volume(vol_env, fname):
    d = os.environ.get(vol_env)
    if not d:
        return None
    with open(os.path.join(d, fname)) as f:
        return f.read()
