"""GPU tensor store facade: kt.put/get for CUDA tensors and state dicts.

put(key, tensor|state_dict): register the live tensors with the node's
pod-data-server (hipIpc, zero-copy) and publish {key -> source host} to the
metadata service. get(key, dest): same-node -> IPC device copy inside the
daemon; cross-node -> per-transfer RCCL broadcast between daemons. Packed
mode (BroadcastWindow.pack) moves a whole same-dtype state dict as one flat
buffer = one collective (xGMI-friendly). Reference parity:
data_store/gpu_transfer.py + pod_data_server client."""
import json
import os

import httpx

from kubetorch_amd import constants as C
from kubetorch_amd.data_store.pod_data_server import (
    PodDataClient,
    _my_ip,
    export_tensor,
    import_tensor,
)

_client = None


def _pd_client():
    global _client
    if _client is None:
        _client = PodDataClient()
    return _client


def _my_host(cli):
    port = cli.ping().get("tcp_port", C.GPU_DATA_SERVER_TCP_PORT)
    return f"{_my_ip()}:{port}"


# -- metadata ------------------------------------------------------------------
def _meta_root():
    root = os.path.join(
        os.environ.get("KT_STORE_ROOT", os.path.expanduser("~/.ktamd/store")),
        ".meta")
    os.makedirs(root, exist_ok=True)
    return root


def _meta_path(key):
    return os.path.join(_meta_root(), key.strip("/").replace("/", "_") + ".json")


def publish_meta(key, meta):
    url = os.environ.get("KT_STORE_URL")
    if url:
        httpx.post(f"{url}/meta/{key}", json=meta, timeout=10).raise_for_status()
    else:
        with open(_meta_path(key), "w") as f:
            json.dump(meta, f)


def get_meta(key):
    url = os.environ.get("KT_STORE_URL")
    if url:
        r = httpx.get(f"{url}/meta/{key}", timeout=10)
        if r.status_code == 404:
            raise KeyError(key)
        r.raise_for_status()
        return r.json()
    p = _meta_path(key)
    if not os.path.exists(p):
        raise KeyError(key)
    with open(p) as f:
        return json.load(f)


def _flatten(sd):
    return {k: sd[k] for k in sorted(sd)}


# key -> (flat buffer, signature): packed publish buffers stay allocated
# and IPC-registered so repeat publishes only run the pack kernel
_PACKED_CACHE = {}


# -- public api ----------------------------------------------------------------
def put(key, src, window=None):
    """Publish a tensor or state dict. Tensors stay in the owner's memory
    (zero-copy hipIpc registration); only metadata goes to the store.
    With window.world_size > 2, serves a W-party broadcast group instead."""
    import torch

    if window is not None and getattr(window, "world_size", 2) > 2:
        return put_broadcast(key, src, window)
    cli = _pd_client()
    if isinstance(src, torch.Tensor):
        cli.register(key, src)
        publish_meta(key, {"kind": "tensor",
                           "host": _my_host(cli),
                           "meta": {"shape": tuple(src.shape),
                                    "dtype": str(src.dtype).split(".")[-1]}})
        return {"key": key, "locale": "local"}
    sd = _flatten(src)
    pack = bool(window and getattr(window, "pack", False))
    if pack:
        from kubetorch_amd import ops as kt_ops

        dtypes = {t.dtype for t in sd.values()}
        if len(dtypes) != 1:
            raise ValueError("packed mode requires a single dtype")
        # one pack kernel instead of torch.cat's per-tensor copies; segment
        # starts 16B-aligned so the kernel runs pure uint4 vectors.
        # Re-publishes of the same key/shapes (the RL weight-sync hot
        # loop) reuse the cached flat buffer AND its hipIpc registration:
        # the r01-measured 14 GB/s packed put was dominated by the fresh
        # 256 MB allocation + hipIpcGetMemHandle per publish, not the copy.
        sig = (tuple(t.numel() for t in sd.values()),
               next(iter(dtypes)), len(sd))
        cached = _PACKED_CACHE.get(key)
        if cached is not None and cached[1] == sig:
            flat = cached[0]
            kt_ops.pack_tensors([t.contiguous() for t in sd.values()],
                                flat=flat)
        else:
            flat, _ = kt_ops.pack_tensors(
                [t.contiguous() for t in sd.values()])
            cli.register(f"{key}/__packed__", flat)
            _PACKED_CACHE[key] = (flat, sig)
    else:
        for sub, t in sd.items():
            cli.register(f"{key}/{sub}", t)
    publish_meta(key, {
        "kind": "state_dict",
        "host": _my_host(cli),
        "packed": pack,
        "entries": {sub: {"shape": tuple(t.shape),
                          "dtype": str(t.dtype).split(".")[-1],
                          "numel": t.numel()}
                    for sub, t in sd.items()},
    })
    return {"key": key, "locale": "local", "packed": pack}


def get(key, dest, window=None):
    """Receive into pre-allocated dest tensor / state dict (shapes must
    match the published metadata). With window.world_size > 2, joins the
    W-party broadcast group."""
    import torch

    if window is not None and getattr(window, "world_size", 2) > 2:
        return get_broadcast(key, dest, window)
    cli = _pd_client()
    meta = get_meta(key)
    source = meta["host"]
    local = source.split(":")[0] in (_my_ip(), "127.0.0.1")

    def fetch(k, d):
        if local:
            cli.get_local(k, d)
        else:
            cli.fetch_remote(k, d, source)

    if isinstance(dest, torch.Tensor):
        if meta["kind"] != "tensor":
            raise TypeError(f"{key} is a {meta['kind']}, dest is a tensor")
        fetch(key, dest)
        return dest
    sd = _flatten(dest)
    entries = meta["entries"]
    missing = set(sd) - set(entries)
    if missing:
        raise KeyError(f"dest keys not in stored state dict: {sorted(missing)}")
    if meta.get("packed"):
        from kubetorch_amd import ops as kt_ops

        dt = getattr(torch, next(iter(entries.values()))["dtype"])
        first = next(iter(sd.values()))
        numels = [e["numel"] for e in entries.values()]
        offs, total = kt_ops.aligned_offsets(numels, dt.itemsize)
        flat = torch.empty(total, dtype=dt, device=first.device)
        fetch(f"{key}/__packed__", flat)
        # scatter with one unpack kernel where possible (publish order)
        wanted = [(sub, e, o) for (sub, e), o
                  in zip(entries.items(), offs) if sub in sd]
        kt_ops.unpack_tensors(
            flat, [sd[sub] for sub, _e, _o in wanted],
            offsets=[o for _s, _e, o in wanted])
        return dest
    for sub in sd:
        fetch(f"{key}/{sub}", sd[sub])
    return dest


# -- W-party broadcast (one RCCL/gloo group, fan-out to many receivers) ------
def _join_group(group_id, world_size, master=None, port=None):
    url = os.environ.get("KT_STORE_URL")
    body = {"group_id": group_id, "world_size": world_size}
    if master:
        body["master"] = master
        body["port"] = port
    if url:
        r = httpx.post(f"{url}/broadcast/join", json=body, timeout=10)
        r.raise_for_status()
        return r.json()
    # local fallback: file-based coordination with an exclusive lock
    import fcntl

    path = os.path.join(_meta_root(),
                        "bcast_" + group_id.replace("/", "_") + ".json")
    with open(path + ".lock", "w") as lockf:
        fcntl.flock(lockf, fcntl.LOCK_EX)
        rec = {"world_size": world_size, "members": 0, "master": None,
               "port": None}
        if os.path.exists(path):
            with open(path) as f:
                rec = json.load(f)
        if master:
            rec.update(master=master, port=port, world_size=world_size,
                       members=0)
            rank = 0
        else:
            rec["members"] += 1
            rank = rec["members"]
        with open(path, "w") as f:
            json.dump(rec, f)
        return {"rank": rank, "world_size": rec["world_size"],
                "master": rec["master"], "port": rec["port"]}


def put_broadcast(key, src, window):
    """Serve `src` (tensor or state dict) to window.world_size-1 receivers
    through ONE collective group (RCCL on GPU / gloo on CPU). Blocks in the
    daemon until all receivers join (window.timeout)."""
    import torch

    cli = _pd_client()
    sd = {key: src} if isinstance(src, torch.Tensor) else \
        {f"{key}/{s}": t for s, t in _flatten(src).items()}
    for k2, t in sd.items():
        cli.register(k2, t)
    r = cli.request({"cmd": "serve_bcast", "keys": sorted(sd),
                     "world_size": window.world_size})
    if not r["ok"]:
        raise RuntimeError(r["error"])
    meta = {
        "kind": "tensor" if isinstance(src, torch.Tensor) else "state_dict",
        "host": _my_host(cli),
        "broadcast": {"group_id": key, "port": r["port"],
                      "world_size": window.world_size},
    }
    if not isinstance(src, torch.Tensor):
        meta["entries"] = {s: {"shape": tuple(t.shape),
                               "dtype": str(t.dtype).split(".")[-1],
                               "numel": t.numel()}
                          for s, t in _flatten(src).items()}
    publish_meta(key, meta)
    _join_group(key, window.world_size, master=r["ip"], port=r["port"])
    return {"key": key, "broadcast": True, "world_size": window.world_size}


def get_broadcast(key, dest, window, poll=0.2):
    """Join the broadcast group for `key` and receive into dest."""
    import time

    import torch

    cli = _pd_client()
    deadline = time.time() + (window.timeout if window else 300)
    info = None
    while time.time() < deadline:
        info = _join_group(key, window.world_size)
        if info.get("master"):
            break
        time.sleep(poll)
    if not info or not info.get("master"):
        raise TimeoutError(f"broadcast source for {key!r} never registered")
    dests = ([dest] if isinstance(dest, torch.Tensor)
             else [t for _, t in sorted(_flatten(dest).items())])
    resp = cli.request({
        "cmd": "join_bcast",
        "dests": [export_tensor(d) for d in dests],
        "rank": info["rank"], "world_size": info["world_size"],
        "port": info["port"], "master_ip": info["master"].split(":")[0],
    })
    if not resp["ok"]:
        raise RuntimeError(resp["error"])
    if resp.get("payloads"):
        for d, p in zip(dests, resp["payloads"]):
            d.copy_(import_tensor(p))
    return dest


def rm(key):
    cli = _pd_client()
    try:
        meta = get_meta(key)
    except KeyError:
        return
    if meta["kind"] == "tensor":
        cli.unregister(key)
    else:
        if meta.get("packed"):
            cli.unregister(f"{key}/__packed__")
        for sub in meta.get("entries", {}):
            cli.unregister(f"{key}/{sub}")
    url = os.environ.get("KT_STORE_URL")
    if url:
        httpx.delete(f"{url}/meta/{key}", timeout=10)
    else:
        p = _meta_path(key)
        if os.path.exists(p):
            os.remove(p)
