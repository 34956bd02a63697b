"""Namespace data-store service: file store + key metadata + log store.

One pod per namespace (reference runs rsyncd+metadata+Loki images; this is a
single from-scratch FastAPI service):
    PUT  /files/{key...}       upload (tar stream or single file)
    GET  /files/{key...}       download
    GET  /ls?prefix=           list keys
    DELETE /files/{key...}
    POST /meta/{key...}        publish key metadata (e.g. GPU source ip)
    GET  /meta/{key...}
    DELETE /meta/{key...}
    POST /logs/push            pod log batches (ring per service)
    GET  /logs/tail            query by service/request_id
"""
import io
import json
import os
import tarfile

from kubetorch_amd.utils.tar import safe_extractall
import threading
import time

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from kubetorch_amd import constants as C

DATA_ROOT = os.environ.get("KT_STORE_ROOT",
                           os.path.expanduser("~/.ktamd/store"))

app = FastAPI()
_meta = {}
_meta_lock = threading.Lock()
_logs = {}
_logs_lock = threading.Lock()
_LOG_RING = 50000


def _path_for(key):
    root = os.path.abspath(DATA_ROOT)
    p = os.path.abspath(os.path.join(root, key.strip("/")))
    # separator-aware: "/data/storeX" must not pass a "/data/store" root
    if p != root and not p.startswith(root + os.sep):
        raise ValueError("key escapes store root")
    return p


@app.get("/health")
def health():
    return {"status": "ok", "root": DATA_ROOT}


@app.put("/files/{key:path}")
async def put_file(key: str, request: Request):
    path = _path_for(key)
    os.makedirs(os.path.dirname(path), exist_ok=True)
    body = await request.body()
    if request.headers.get("X-KT-Tar") == "1":
        os.makedirs(path, exist_ok=True)
        with tarfile.open(fileobj=io.BytesIO(body), mode="r:gz") as tar:
            safe_extractall(tar, path)
    else:
        with open(path, "wb") as f:
            f.write(body)
        mtime = request.headers.get("X-KT-Mtime")
        if mtime:  # preserve client mtime so delta-sync manifests compare
            try:
                os.utime(path, (float(mtime), float(mtime)))
            except ValueError:
                pass
    return {"ok": True, "bytes": len(body)}


@app.get("/files/{key:path}")
def get_file(key: str):
    path = _path_for(key)
    if os.path.isdir(path):
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w:gz") as tar:
            tar.add(path, arcname=".")
        return Response(buf.getvalue(), media_type="application/gzip",
                        headers={"X-KT-Tar": "1"})
    if not os.path.exists(path):
        return JSONResponse({"error": "not found"}, status_code=404)

    def stream():
        with open(path, "rb") as f:
            while chunk := f.read(1 << 20):
                yield chunk

    return StreamingResponse(stream(), media_type="application/octet-stream")


@app.get("/manifest/{key:path}")
def manifest(key: str):
    """File manifest {relpath: [size, mtime]} under a key — the client diffs
    against it and uploads only changed files (delta sync, the rsync-style
    hot loop)."""
    base = _path_for(key)
    out = {}
    if os.path.isdir(base):
        for root, _dirs, files in os.walk(base):
            for f in files:
                full = os.path.join(root, f)
                st = os.stat(full)
                out[os.path.relpath(full, base)] = [st.st_size,
                                                    round(st.st_mtime, 3)]
    return {"files": out}


@app.delete("/manifest/{key:path}")
async def delete_files(key: str, request: Request):
    """Delete a list of relative paths under key (delta-sync removals)."""
    body = await request.json()
    base = _path_for(key)
    for rel in body.get("paths", []):
        p = os.path.abspath(os.path.join(base, rel))
        if (p == base or p.startswith(base + os.sep)) and os.path.isfile(p):
            os.remove(p)
    return {"ok": True}


@app.get("/ls")
def ls(prefix: str = ""):
    base = _path_for(prefix) if prefix else DATA_ROOT
    out = []
    if os.path.isdir(base):
        for root, _dirs, files in os.walk(base):
            for f in files:
                full = os.path.join(root, f)
                out.append({
                    "key": os.path.relpath(full, DATA_ROOT),
                    "size": os.path.getsize(full),
                    "mtime": os.path.getmtime(full),
                })
    elif os.path.exists(base):
        out.append({"key": prefix, "size": os.path.getsize(base),
                    "mtime": os.path.getmtime(base)})
    return {"entries": out}


@app.delete("/files/{key:path}")
def rm(key: str):
    import shutil

    path = _path_for(key)
    if os.path.isdir(path):
        shutil.rmtree(path, ignore_errors=True)
    elif os.path.exists(path):
        os.remove(path)
    return {"ok": True}


@app.post("/meta/{key:path}")
async def put_meta(key: str, request: Request):
    body = await request.json()
    with _meta_lock:
        _meta[key] = {**body, "updated": time.time()}
    return {"ok": True}


@app.get("/meta/{key:path}")
def get_meta(key: str):
    with _meta_lock:
        m = _meta.get(key)
    if m is None:
        return JSONResponse({"error": "not found"}, status_code=404)
    return m


@app.delete("/meta/{key:path}")
def del_meta(key: str):
    with _meta_lock:
        _meta.pop(key, None)
    return {"ok": True}


_bcast = {}
_bcast_lock = threading.Lock()


@app.post("/broadcast/join")
async def broadcast_join(request: Request):
    """Coordinate a W-party broadcast group: the source registers with
    rank 0 (+ rendezvous host/port); receivers join and get assigned the
    next rank. (Reference parity: metadata server join_broadcast.)"""
    body = await request.json()
    gid = body["group_id"]
    with _bcast_lock:
        rec = _bcast.get(gid)
        if rec is None:
            rec = _bcast[gid] = {"world_size": body.get("world_size", 2),
                                 "members": 0, "master": None, "port": None,
                                 "created": time.time()}
        if body.get("master"):
            rec["master"] = body["master"]
            rec["port"] = body.get("port")
            rec["world_size"] = body.get("world_size", rec["world_size"])
            rec["members"] = 0  # fresh generation
            rank = 0
        else:
            rec["members"] += 1
            rank = rec["members"]
        return {"rank": rank, "world_size": rec["world_size"],
                "master": rec["master"], "port": rec["port"]}


@app.get("/broadcast/status")
def broadcast_status(group_id: str):
    with _bcast_lock:
        rec = _bcast.get(group_id)
    if rec is None:
        return JSONResponse({"error": "not found"}, status_code=404)
    return rec


@app.delete("/broadcast/{group_id}")
def broadcast_complete(group_id: str):
    with _bcast_lock:
        _bcast.pop(group_id, None)
    return {"ok": True}


_fsb = {}
_fsb_lock = threading.Lock()


@app.post("/fsbcast/join")
async def fsbcast_join(request: Request):
    """Filesystem tree-broadcast coordination (reference parity:
    metadata-server join_fs_broadcast + pod-data-server fs-broadcast
    tracking). Each joiner is assigned a source with spare capacity —
    a pod that already completed the download (preferred, to offload the
    store) or the store itself. fanout bounds concurrent children per
    source. Returns {"source": url|"store", "parent": id} or
    {"wait": true} when every source is saturated."""
    body = await request.json()
    key = body["key"].strip("/")
    fanout = int(body.get("fanout", 50))
    with _fsb_lock:
        rec = _fsb.setdefault(key, {"sources": [], "active": {"store": 0},
                                    "created": time.time()})
        for url in rec["sources"]:  # completed pods first: offload the store
            if rec["active"].get(url, 0) < fanout:
                rec["active"][url] = rec["active"].get(url, 0) + 1
                return {"source": url, "parent": url}
        if rec["active"]["store"] < max(1, fanout):
            rec["active"]["store"] += 1
            return {"source": "store", "parent": "store"}
        return {"wait": True}


@app.post("/fsbcast/complete")
async def fsbcast_complete(request: Request):
    """A joiner finished downloading: release its parent's slot and (if it
    advertises a serve url) register it as a source for later joiners."""
    body = await request.json()
    key = body["key"].strip("/")
    with _fsb_lock:
        rec = _fsb.get(key)
        if rec is None:
            return {"ok": True}
        parent = body.get("parent")
        if parent in rec["active"] and rec["active"][parent] > 0:
            rec["active"][parent] -= 1
        url = body.get("url")
        if url and url not in rec["sources"]:
            rec["sources"].append(url)
    return {"ok": True}


@app.get("/fsbcast/status")
def fsbcast_status(key: str):
    with _fsb_lock:
        rec = _fsb.get(key.strip("/"))
    if rec is None:
        return JSONResponse({"error": "not found"}, status_code=404)
    return rec


@app.delete("/fsbcast/{key:path}")
def fsbcast_clear(key: str):
    with _fsb_lock:
        _fsb.pop(key.strip("/"), None)
    return {"ok": True}


@app.post("/logs/push")
async def logs_push(request: Request):
    body = await request.json()
    service = body.get("service", "unknown")
    with _logs_lock:
        ring = _logs.setdefault(service, [])
        ring.extend(body.get("entries", []))
        if len(ring) > _LOG_RING:
            _logs[service] = ring[-_LOG_RING:]
    return {"ok": True}


@app.get("/logs/tail")
def logs_tail(service: str, request_id: str = None, since_ts: float = 0,
              limit: int = 1000):
    with _logs_lock:
        ring = list(_logs.get(service, []))
    out = [e for e in ring if e.get("ts", 0) >= since_ts
           and (request_id is None or e.get("request_id") == request_id)]
    return {"entries": out[-limit:]}


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=C.DATA_STORE_PORT)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--root", default=DATA_ROOT)
    args = ap.parse_args()
    globals()["DATA_ROOT"] = args.root
    os.makedirs(args.root, exist_ok=True)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
