"""Standalone GPU IPC debug (run on a GPU box): exercises _share_cuda_ /
_new_shared_cuda across processes with full stderr."""
import multiprocessing as mp
import os
import sys
import time
import traceback

SOCK = "/tmp/dbg_pd.sock"


def putter(q):
    try:
        os.environ["KT_GPU_DATA_SOCK"] = SOCK
        import kubetorch_amd.data_store.pod_data_server as pds

        pds.SOCK_PATH = SOCK
        pds.LOCK_PATH = SOCK + ".lock"
        import torch

        t = torch.arange(1024, dtype=torch.bfloat16, device="cuda") * 0.5
        print("putter: tensor made", flush=True)
        h = t.untyped_storage()._share_cuda_()
        print("putter: _share_cuda_ ok:", type(h), len(h), flush=True)
        from kubetorch_amd.data_store import gpu_store

        gpu_store._client = None
        gpu_store.put("gpu_w", t)
        print("putter: published", flush=True)
        q.put("published")
        time.sleep(120)
    except BaseException:
        traceback.print_exc()
        q.put("FAILED")


if __name__ == "__main__":
    os.environ["KT_STORE_ROOT"] = "/tmp/dbg_store"
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=putter, args=(q,))
    p.start()
    msg = q.get(timeout=300)
    print("main: got", msg)
    if msg == "published":
        os.environ["KT_GPU_DATA_SOCK"] = SOCK
        import kubetorch_amd.data_store.pod_data_server as pds

        pds.SOCK_PATH = SOCK
        pds.LOCK_PATH = SOCK + ".lock"
        import torch

        from kubetorch_amd.data_store import gpu_store

        gpu_store._client = None
        dest = torch.zeros(1024, dtype=torch.bfloat16, device="cuda")
        gpu_store.get("gpu_w", dest)
        exp = torch.arange(1024, dtype=torch.bfloat16, device="cuda") * 0.5
        print("main: match =", torch.equal(dest, exp))
    p.terminate()
    p.join()
