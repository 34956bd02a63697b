"""kt.put / kt.get / kt.ls / kt.rm — the user-facing data-store facade.

Routes by payload type (reference parity: data_store/data_store_cmds.py):
  * GPU tensors / state dicts -> gpu_store (hipIpc registration + RCCL
    transfer through the per-node pod-data-server)
  * filesystem paths          -> file store (HTTP store service, or the
    local directory store when no store service is configured)
"""
import io
import os
import shutil
import tarfile

from kubetorch_amd.utils.tar import safe_extractall

import httpx

from kubetorch_amd import constants as C

LOCAL_STORE_ROOT = os.environ.get(
    "KT_STORE_ROOT", os.path.expanduser("~/.ktamd/store"))


def _store_url():
    return os.environ.get("KT_STORE_URL")


def _is_gpu_data(obj):
    try:
        import torch

        if isinstance(obj, torch.Tensor):
            return obj.is_cuda
        if isinstance(obj, dict) and obj:
            vals = list(obj.values())
            return all(isinstance(v, torch.Tensor) and v.is_cuda for v in vals)
    except ImportError:
        pass
    return False


def _local_path(key):
    root = os.path.abspath(LOCAL_STORE_ROOT)
    p = os.path.abspath(os.path.join(root, key.strip("/")))
    if p != root and not p.startswith(root + os.sep):
        raise ValueError("key escapes store root")
    return p


_EXCLUDE = ("__pycache__", ".git", ".pytest_cache")


def _local_manifest(src):
    out = {}
    for root, dirs, files in os.walk(src):
        dirs[:] = [d for d in dirs if d not in _EXCLUDE]
        for f in files:
            if f.endswith(".pyc"):
                continue
            full = os.path.join(root, f)
            st = os.stat(full)
            out[os.path.relpath(full, src)] = [st.st_size, round(st.st_mtime, 3)]
    return out


def put_delta(key, src, timeout=C.HTTP_TIMEOUT):
    """Incremental dir upload: diff the store's manifest against local
    size+mtime, upload only changed files, delete removed ones (the
    rsync-style hot loop; reference: RsyncClient). Falls back to a full
    tar upload on the first sync."""
    url = _store_url()
    src = os.path.expanduser(str(src))
    local = _local_manifest(src)
    r = httpx.get(f"{url}/manifest/{key}", timeout=timeout)
    remote = r.json().get("files", {}) if r.status_code == 200 else {}
    if not remote:
        return put(key, src, timeout=timeout, _delta=False)
    changed = [p for p, meta in local.items() if remote.get(p) != meta]
    removed = [p for p in remote if p not in local]
    for rel in changed:
        with open(os.path.join(src, rel), "rb") as f:
            httpx.put(f"{url}/files/{key}/{rel}", content=f.read(),
                      headers={"X-KT-Mtime": str(local[rel][1])},
                      timeout=timeout).raise_for_status()
    if removed:
        httpx.request("DELETE", f"{url}/manifest/{key}",
                      json={"paths": removed}, timeout=timeout)
    return {"key": key, "locale": "store", "changed": len(changed),
            "removed": len(removed)}


def _self_host():
    host = os.environ.get("KT_SELF_HOST")
    if host:
        return host
    import socket

    try:
        ip = socket.gethostbyname(socket.gethostname())
    except socket.gaierror:
        ip = "127.0.0.1"
    return f"{ip}:{os.environ.get('KT_SERVER_PORT', C.SERVER_PORT)}"


def localfs_registry_path():
    """Per-pod registry of locally-served keys; the pod's http_server
    process reads it to answer /localfiles/{key} (worker processes and
    the server share the pod filesystem)."""
    import tempfile

    tag = os.environ.get("KT_SERVER_PORT", str(C.SERVER_PORT))
    return os.path.join(tempfile.gettempdir(), f"kt-localfs-{tag}.json")


def _register_local_serve(key, path):
    import json

    reg_path = localfs_registry_path()
    reg = {}
    if os.path.exists(reg_path):
        try:
            with open(reg_path) as f:
                reg = json.load(f)
        except (OSError, ValueError):
            reg = {}
    reg[key.strip("/")] = path
    tmp = reg_path + ".tmp"
    with open(tmp, "w") as f:
        json.dump(reg, f)
    os.replace(tmp, reg_path)


def put(key, src, window=None, timeout=C.HTTP_TIMEOUT, _delta=True,
        locale="store"):
    """Store a file/dir (by path) or publish GPU tensors under `key`.

    locale="store" copies the data into the namespace store pod;
    locale="local" registers the key WITHOUT copying (zero-copy): this
    pod serves the file on demand and getters fetch it p2p, falling back
    to the store (and de-registering the source) if this pod dies.
    (Reference parity: data_store_client.py local locale, :175/:325.)

    Batch mode (reference: put(key=[...], src=[...])): lists of keys and
    sources are stored pairwise."""
    if isinstance(key, (list, tuple)):
        srcs = src if isinstance(src, (list, tuple)) else [src] * len(key)
        if len(srcs) != len(key):
            raise ValueError(f"{len(key)} keys but {len(srcs)} sources")
        return [put(k, s, window=window, timeout=timeout, _delta=_delta,
                    locale=locale) for k, s in zip(key, srcs)]
    if _is_gpu_data(src):
        from kubetorch_amd.data_store import gpu_store

        return gpu_store.put(key, src, window=window)
    src = os.path.expanduser(str(src))
    if not os.path.exists(src):
        raise FileNotFoundError(src)
    url = _store_url()
    if locale == "local":
        if url is None:
            raise ValueError("locale='local' needs a store service for the "
                             "key registry (KT_STORE_URL)")
        src_abs = os.path.abspath(src)
        _register_local_serve(key, src_abs)
        httpx.post(f"{url}/meta/localfs/{key}",
                   json={"host": _self_host(), "path": src_abs,
                         "is_dir": os.path.isdir(src_abs)},
                   timeout=timeout).raise_for_status()
        return {"key": key, "locale": "local", "host": _self_host()}
    if url is not None and _delta and os.path.isdir(src):
        return put_delta(key, src, timeout=timeout)
    if url is None:
        dest = _local_path(key)
        if os.path.isdir(src):
            if os.path.exists(dest):
                shutil.rmtree(dest)
            shutil.copytree(src, dest,
                            ignore=shutil.ignore_patterns(
                                "__pycache__", ".git", "*.pyc"))
        else:
            os.makedirs(os.path.dirname(dest), exist_ok=True)
            shutil.copy2(src, dest)
        return {"key": key, "locale": "local-dir"}
    if os.path.isdir(src):
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w:gz") as tar:
            tar.add(src, arcname=".",
                    filter=lambda ti: None if "__pycache__" in ti.name
                    or "/.git/" in ti.name or ti.name.endswith(".pyc") else ti)
        r = httpx.put(f"{url}/files/{key}", content=buf.getvalue(),
                      headers={"X-KT-Tar": "1"}, timeout=timeout)
    else:
        with open(src, "rb") as f:
            r = httpx.put(f"{url}/files/{key}", content=f.read(), timeout=timeout)
    r.raise_for_status()
    return {"key": key, "locale": "store"}


def get(key, dest=None, window=None, timeout=C.HTTP_TIMEOUT,
        contents=False):
    """Fetch a stored file/dir to `dest` path, or receive GPU tensors into
    `dest` tensors/state dict.

    Batch mode (reference: get(key=[...])): a list of keys fetches each
    (dest may be a matching list or None). `contents=True` returns a
    single file's bytes directly instead of writing a path."""
    if isinstance(key, (list, tuple)):
        dests = dest if isinstance(dest, (list, tuple)) else [dest] * len(key)
        if len(dests) != len(key):
            raise ValueError(f"{len(key)} keys but {len(dests)} dests")
        return [get(k, d, window=window, timeout=timeout, contents=contents)
                for k, d in zip(key, dests)]
    if contents:
        import tempfile

        with tempfile.TemporaryDirectory() as td:
            out = get(key, os.path.join(td, "f"), window=window,
                      timeout=timeout)
            if os.path.isdir(out):
                raise ValueError(
                    f"contents=True needs a file key; {key!r} is a dir")
            with open(out, "rb") as f:
                return f.read()
    if dest is not None and _is_gpu_data(dest):
        from kubetorch_amd.data_store import gpu_store

        return gpu_store.get(key, dest, window=window)
    if window is not None and _store_url() is not None:
        # fs key + BroadcastWindow -> rolling tree broadcast (W pods share
        # the download through completed peers instead of the store)
        return get_broadcast(key, dest, fanout=window.fanout, timeout=timeout)
    url = _store_url()
    if url is None:
        srcp = _local_path(key)
        if not os.path.exists(srcp):
            raise KeyError(f"no such key: {key}")
        if dest is None:
            return srcp
        dest = os.path.expanduser(dest)
        if os.path.isdir(srcp):
            if os.path.exists(dest):
                shutil.rmtree(dest)
            shutil.copytree(srcp, dest)
        else:
            os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
            shutil.copy2(srcp, dest)
        return dest
    # p2p-first: a local-locale source serves the key directly; if the
    # source is gone, de-register it and fall back to the store copy
    # (reference: get w/ remove_source retry, data_store_client.py:325)
    m = httpx.get(f"{url}/meta/localfs/{key}", timeout=timeout)
    if m.status_code == 200:
        source = m.json().get("host")
        try:
            return _fetch_from_peer(f"http://{source}", key, dest,
                                    timeout, route="localfiles")
        except (httpx.HTTPError, OSError):
            httpx.delete(f"{url}/meta/localfs/{key}", timeout=timeout)
    r = httpx.get(f"{url}/files/{key}", timeout=timeout)
    if r.status_code == 404:
        raise KeyError(f"no such key: {key}")
    r.raise_for_status()
    dest = os.path.expanduser(dest or os.path.basename(key))
    if r.headers.get("X-KT-Tar") == "1":
        os.makedirs(dest, exist_ok=True)
        with tarfile.open(fileobj=io.BytesIO(r.content), mode="r:gz") as tar:
            safe_extractall(tar, dest)
    else:
        os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
        with open(dest, "wb") as f:
            f.write(r.content)
    return dest


def _fetch_from_peer(source, key, dest, timeout, route="files"):
    r = httpx.get(f"{source}/{route}/{key}", timeout=timeout)
    r.raise_for_status()
    dest = os.path.expanduser(dest or os.path.basename(key))
    if r.headers.get("X-KT-Tar") == "1":
        os.makedirs(dest, exist_ok=True)
        with tarfile.open(fileobj=io.BytesIO(r.content), mode="r:gz") as tar:
            safe_extractall(tar, dest)
    else:
        os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
        with open(dest, "wb") as f:
            f.write(r.content)
    return dest


def get_broadcast(key, dest=None, serve=True, fanout=None, timeout=C.HTTP_TIMEOUT,
                  poll=0.2, max_wait=600.0):
    """Tree-broadcast get: W pods fetching the same key form a rolling tree
    (store -> first `fanout` pods -> their children, exponential fan-out)
    so the store serves O(fanout) downloads total instead of W. Reference
    parity: DataStoreClient broadcast get + join_fs_broadcast.

    serve=True registers this process as a source for later joiners (lazy
    singleton BcastFileServer). Returns the local dest path."""
    import time as _time

    from kubetorch_amd.data_store import fileserve

    url = _store_url()
    if url is None:  # local-dir mode: nothing to offload
        return get(key, dest, timeout=timeout)
    if fanout is None:
        fanout = int(os.environ.get("KT_FS_BCAST_FANOUT", "50"))
    key = key.strip("/")
    deadline = _time.monotonic() + max_wait
    while True:
        r = httpx.post(f"{url}/fsbcast/join",
                       json={"key": key, "fanout": fanout}, timeout=timeout)
        r.raise_for_status()
        j = r.json()
        if not j.get("wait"):
            break
        if _time.monotonic() > deadline:
            raise TimeoutError(f"fs broadcast join timed out for {key}")
        _time.sleep(poll)
    parent = j["parent"]
    try:
        if j["source"] == "store":
            out = get(key, dest, timeout=timeout)
        else:
            out = _fetch_from_peer(j["source"], key, dest, timeout)
    except Exception:
        # release the slot so the tree doesn't wedge on a failed child
        httpx.post(f"{url}/fsbcast/complete",
                   json={"key": key, "parent": parent}, timeout=timeout)
        raise
    serve_url = None
    if serve:
        fileserve.register_local(key, out)
        serve_url = fileserve.ensure_server().url
    httpx.post(f"{url}/fsbcast/complete",
               json={"key": key, "parent": parent, "url": serve_url},
               timeout=timeout)
    return out


def ls(prefix=""):
    url = _store_url()
    if url is None:
        base = _local_path(prefix) if prefix else LOCAL_STORE_ROOT
        out = []
        if os.path.isdir(base):
            for root, _d, files in os.walk(base):
                for f in files:
                    full = os.path.join(root, f)
                    out.append({"key": os.path.relpath(full, LOCAL_STORE_ROOT),
                                "size": os.path.getsize(full)})
        return out
    r = httpx.get(f"{url}/ls", params={"prefix": prefix})
    r.raise_for_status()
    return r.json()["entries"]


def rm(key, prefix=False):
    """Delete a key (dirs recurse). prefix=True deletes every key under
    the given prefix (reference: rm(prefix=) bulk cleanup)."""
    if prefix:
        for e in ls(key):
            rm(e["key"])
        return {"ok": True, "prefix": key}
    url = _store_url()
    if url is None:
        p = _local_path(key)
        if os.path.isdir(p):
            shutil.rmtree(p, ignore_errors=True)
        elif os.path.exists(p):
            os.remove(p)
        return {"ok": True}
    r = httpx.delete(f"{url}/files/{key}")
    r.raise_for_status()
    return r.json()


def sync_workdir_from_store():
    """Pod-side: pull the synced working dir for this workload and remap the
    callable pointers to it. Returns the local root or None."""
    key = os.environ.get("KT_WORKDIR_KEY")
    if not key:
        return None
    if _store_url() is None and not os.path.isdir(_local_path(key)):
        return None  # local mode without a store: original paths are valid
    root = os.environ.get("KT_WORKDIR_BASE", "/workdir")
    dest = os.path.join(root, key.strip("/").replace("/", "_"))
    try:
        # many pods pulling the same workdir (reload fan-out) form the
        # broadcast tree instead of all hitting the store
        workers = int(os.environ.get("KT_NUM_WORKERS", "1"))
        if workers >= 4 and _store_url() is not None:
            get_broadcast(key, dest)
        else:
            get(key, dest)
    except KeyError:
        return None
    rel = os.environ.get("KT_REL_PATH")
    if rel:
        os.environ[C.ENV_PROJECT_ROOT] = dest
        os.environ[C.ENV_FILE_PATH] = os.path.join(dest, rel)
    return dest
