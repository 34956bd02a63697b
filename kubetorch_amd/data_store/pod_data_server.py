"""Per-node pod-data-server: GPU tensor store + transfer service.

Reference parity: data_store/pod_data_server.py (2950 LoC). MI355X design:
  * App processes register live GPU tensors with the per-node daemon via
    hipIpc handles (torch-ROCm `UntypedStorage._share_cuda_` == hipIpc
    under HSA dmabuf IPC) — zero-copy: the daemon maps the app's HBM.
  * Same-node transfers are a single device-to-device `copy_` inside the
    daemon between two mapped storages (xGMI/HBM path, no RCCL).
  * Cross-node transfers form a per-transfer 2..W-rank process group
    (TCPStore rendezvous; backend "nccl" == RCCL on GPU boxes, gloo for
    CPU tensors/tests) and broadcast — concurrent groups, no global state.
  * CPU tensors take the same protocol with inline bytes (tests/fallback).

Wire protocol: length-prefixed pickles over a Unix socket (local clients)
and TCP :29400 (server<->server). Trusted in-cluster, like the reference.
"""
import os
import pickle
import socket
import struct
import threading
import time
import traceback

from kubetorch_amd import constants as C

SOCK_PATH = os.environ.get("KT_GPU_DATA_SOCK", "/tmp/kt-amd-gpu-data.sock")
LOCK_PATH = SOCK_PATH + ".lock"


# -- framing -------------------------------------------------------------------
def send_msg(conn, obj):
    data = pickle.dumps(obj)
    conn.sendall(struct.pack("!I", len(data)) + data)


def recv_msg(conn):
    hdr = _recv_exact(conn, 4)
    if hdr is None:
        return None
    (n,) = struct.unpack("!I", hdr)
    data = _recv_exact(conn, n)
    return pickle.loads(data) if data is not None else None


def _recv_exact(conn, n):
    buf = b""
    while len(buf) < n:
        chunk = conn.recv(n - len(buf))
        if not chunk:
            return None
        buf += chunk
    return buf


# -- ipc helpers ---------------------------------------------------------------
def export_tensor(t):
    """Serialize a tensor for registration: hipIpc handle for GPU tensors,
    inline bytes for CPU. Returns a dict payload."""
    import torch

    meta = {"shape": tuple(t.shape), "dtype": str(t.dtype).split(".")[-1],
            "device": "cuda" if t.is_cuda else "cpu"}
    if t.is_cuda:
        # torch's canonical IPC reduction (hipIpcMemHandle + ref-counter +
        # event under the hood); raw _new_shared_cuda without the ref-count
        # machinery segfaults the consumer on ROCm.
        from torch.multiprocessing.reductions import reduce_tensor

        func, args = reduce_tensor(t.contiguous().detach())
        assert func.__name__ == "rebuild_cuda_tensor", func
        return {"meta": meta, "cuda_reduced": args}
    return {"meta": meta, "bytes": t.contiguous().cpu().numpy().tobytes()}


def import_tensor(payload):
    """Rebuild a tensor from an export payload (maps hipIpc memory)."""
    import torch

    meta = payload["meta"]
    dtype = getattr(torch, meta["dtype"])
    if "cuda_reduced" in payload:
        from torch.multiprocessing.reductions import rebuild_cuda_tensor

        torch.cuda.init()
        return rebuild_cuda_tensor(*payload["cuda_reduced"])
    t = torch.frombuffer(bytearray(payload["bytes"]), dtype=dtype)
    return t.reshape(meta["shape"]).clone()


# -- the daemon ----------------------------------------------------------------
class PodDataServer:
    def __init__(self, sock_path=SOCK_PATH, tcp_port=C.GPU_DATA_SERVER_TCP_PORT):
        self.sock_path = sock_path
        self.tcp_port = tcp_port
        self.registry = {}   # key -> {"tensor": t, "meta": {...}}
        self.lock = threading.Lock()
        self._stop = threading.Event()
        self._next_pg_port = C.RCCL_PORT_RANGE[0]
        self.failures = 0

    # -- lifecycle --
    def start(self):
        if os.path.exists(self.sock_path):
            os.unlink(self.sock_path)
        self.usock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.usock.bind(self.sock_path)
        self.usock.listen(64)
        self.tsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.tsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        for cand in range(self.tcp_port, self.tcp_port + 32):
            try:
                self.tsock.bind(("0.0.0.0", cand))
                self.tcp_port = cand
                break
            except OSError:
                continue
        else:
            raise OSError(f"no free TCP port in {self.tcp_port}..+32")
        self.tsock.listen(64)
        threading.Thread(target=self._accept_loop, args=(self.usock,),
                         daemon=True).start()
        threading.Thread(target=self._accept_loop, args=(self.tsock,),
                         daemon=True).start()

    def serve_forever(self):
        self.start()
        while not self._stop.wait(1.0):
            # exit when our socket file is removed (e.g. the owning test/
            # pod dir was cleaned up) — prevents orphaned daemons piling up
            if not os.path.exists(self.sock_path):
                return

    def _accept_loop(self, sock):
        while not self._stop.is_set():
            try:
                conn, _ = sock.accept()
            except OSError:
                return
            threading.Thread(target=self._handle_conn, args=(conn,),
                             daemon=True).start()

    def _handle_conn(self, conn):
        try:
            while True:
                msg = recv_msg(conn)
                if msg is None:
                    return
                try:
                    out = self._dispatch(msg)
                except Exception as e:  # noqa: BLE001
                    self.failures += 1
                    out = {"ok": False, "error": str(e),
                           "traceback": traceback.format_exc()}
                send_msg(conn, out)
        finally:
            conn.close()

    def _alloc_pg_port(self):
        with self.lock:
            port = self._next_pg_port
            self._next_pg_port += 1
            if self._next_pg_port >= C.RCCL_PORT_RANGE[1]:
                self._next_pg_port = C.RCCL_PORT_RANGE[0]
            return port

    # -- ops --
    def _dispatch(self, msg):
        cmd = msg["cmd"]
        if cmd == "ping":
            return {"ok": True, "keys": len(self.registry),
                    "failures": self.failures, "tcp_port": self.tcp_port}
        if cmd == "register":
            t = import_tensor(msg["payload"])
            with self.lock:
                self.registry[msg["key"]] = {"tensor": t,
                                             "meta": msg["payload"]["meta"],
                                             "ts": time.time()}
            return {"ok": True}
        if cmd == "unregister":
            with self.lock:
                self.registry.pop(msg["key"], None)
            return {"ok": True}
        if cmd == "list":
            with self.lock:
                return {"ok": True,
                        "keys": {k: v["meta"] for k, v in self.registry.items()}}
        if cmd == "get_local":
            # same-node zero-copy: dest is the caller's mapped tensor
            entry = self.registry.get(msg["key"])
            if entry is None:
                return {"ok": False, "error": f"key {msg['key']!r} not found"}
            src = entry["tensor"]
            if "dest" in msg and "cuda_reduced" in msg["dest"]:
                # hipIpc-mapped dest: device-to-device copy, caller sees it
                dest = import_tensor(msg["dest"])
                dest.copy_(src)
                import torch

                torch.cuda.synchronize(dest.device)
                return {"ok": True}
            # CPU dest (or no dest): ship bytes back inline
            return {"ok": True, "payload": export_tensor(src.cpu())}
        if cmd == "serve_bcast":
            # rank 0 of a per-transfer PG: serve registered keys
            keys = msg["keys"]
            tensors = []
            for k in keys:
                e = self.registry.get(k)
                if e is None:
                    return {"ok": False, "error": f"key {k!r} not found"}
                tensors.append(e["tensor"])
            port = self._alloc_pg_port()
            world = msg.get("world_size", 2)
            th = threading.Thread(
                target=self._run_bcast, args=(tensors, 0, world, port, None),
                daemon=True)
            th.start()
            return {"ok": True, "port": port, "ip": _my_ip()}
        if cmd == "join_bcast":
            # rank>0: receive into dests. GPU dests are hipIpc-mapped so the
            # caller sees the result directly; CPU dests travel back inline.
            dests = [import_tensor(p) for p in msg["dests"]]
            err = self._run_bcast(dests, msg["rank"], msg["world_size"],
                                  msg["port"], msg["master_ip"])
            if err:
                return {"ok": False, "error": err}
            out = {"ok": True}
            if dests and not dests[0].is_cuda:
                out["payloads"] = [export_tensor(d) for d in dests]
            return out
        return {"ok": False, "error": f"unknown cmd {cmd!r}"}

    def _run_bcast(self, tensors, rank, world, port, master_ip):
        """Per-transfer process group broadcast (rank 0 = source). Uses the
        RCCL backend for GPU tensors, gloo for CPU. Concurrent-safe: each
        transfer has its own TCPStore + PG (reference 'concurrent' mode)."""
        import datetime

        import torch
        import torch.distributed as dist

        try:
            is_cuda = tensors[0].is_cuda
            backend = "nccl" if is_cuda else "gloo"
            store = dist.TCPStore(
                master_ip or _my_ip(), port, world, is_master=(rank == 0),
                timeout=datetime.timedelta(seconds=C.RCCL_TRANSFER_TIMEOUT),
            )
            pg = dist.ProcessGroupGloo(
                store, rank, world,
                datetime.timedelta(seconds=C.RCCL_TRANSFER_TIMEOUT),
            ) if backend == "gloo" else _new_nccl_pg(store, rank, world)
            for t in tensors:
                work = pg.broadcast([t], dist.BroadcastOptions())
                work.wait()
            if is_cuda:
                torch.cuda.synchronize(tensors[0].device)
            return None
        except Exception as e:  # noqa: BLE001
            self.failures += 1
            return f"{type(e).__name__}: {e}"


def _new_nccl_pg(store, rank, world):
    import datetime

    import torch.distributed as dist

    opts = dist.ProcessGroupNCCL.Options()
    opts._timeout = datetime.timedelta(seconds=C.RCCL_TRANSFER_TIMEOUT)
    return dist.ProcessGroupNCCL(store, rank, world, opts)


def _my_ip():
    try:
        return socket.gethostbyname(socket.gethostname())
    except socket.gaierror:
        return "127.0.0.1"


# -- client --------------------------------------------------------------------
class PodDataClient:
    """App-process client of the node's pod-data-server (auto-starts it)."""

    def __init__(self, sock_path=SOCK_PATH, autostart=True):
        self.sock_path = sock_path
        if autostart:
            ensure_server(sock_path)
        self._conn = None
        self._lock = threading.Lock()

    def _connect(self):
        if self._conn is None:
            self._conn = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            self._conn.connect(self.sock_path)
        return self._conn

    # commands that block in the daemon (collective joins) get a dedicated
    # connection so they don't starve other callers of the shared one.
    _BLOCKING = ("join_bcast", "serve_bcast")

    def request(self, msg, retries=2):
        if msg.get("cmd") in self._BLOCKING:
            conn = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            conn.connect(self.sock_path)
            try:
                send_msg(conn, msg)
                resp = recv_msg(conn)
                if resp is None:
                    raise ConnectionError("server closed connection")
                return resp
            finally:
                conn.close()
        with self._lock:
            for attempt in range(retries + 1):
                try:
                    conn = self._connect()
                    send_msg(conn, msg)
                    resp = recv_msg(conn)
                    if resp is None:
                        raise ConnectionError("server closed connection")
                    return resp
                except (ConnectionError, BrokenPipeError, FileNotFoundError):
                    self._conn = None
                    if attempt == retries:
                        raise
                    ensure_server(self.sock_path)
                    time.sleep(0.5)

    # high-level ops
    def register(self, key, tensor):
        r = self.request({"cmd": "register", "key": key,
                          "payload": export_tensor(tensor)})
        if not r["ok"]:
            raise RuntimeError(r["error"])

    def unregister(self, key):
        self.request({"cmd": "unregister", "key": key})

    def get_local(self, key, dest):
        r = self.request({"cmd": "get_local", "key": key,
                          "dest": export_tensor(dest)})
        if not r["ok"]:
            raise KeyError(r.get("error", key))
        if not dest.is_cuda and "payload" in r:
            dest.copy_(import_tensor(r["payload"]))
        return dest

    def fetch_remote(self, key, dest, source_host):
        """Pull `key` from another node's server into dest: ask the source
        to serve a 2-rank broadcast, join as rank 1."""
        ip, _, port = source_host.partition(":")
        s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        s.connect((ip, int(port or C.GPU_DATA_SERVER_TCP_PORT)))
        try:
            send_msg(s, {"cmd": "serve_bcast", "keys": [key], "world_size": 2})
            r = recv_msg(s)
            if not r or not r["ok"]:
                raise KeyError(r.get("error") if r else "no response")
            resp = self.request({
                "cmd": "join_bcast", "dests": [export_tensor(dest)],
                "rank": 1, "world_size": 2, "port": r["port"],
                "master_ip": ip,
            })
            if not resp["ok"]:
                raise RuntimeError(resp["error"])
            if not dest.is_cuda and resp.get("payloads"):
                dest.copy_(import_tensor(resp["payloads"][0]))
            return dest
        finally:
            s.close()

    def ping(self):
        return self.request({"cmd": "ping"})


def ensure_server(sock_path=SOCK_PATH, tcp_port=None, timeout=120.0):
    """Start the per-node daemon if not running (file-lock singleton)."""
    import subprocess
    import sys

    probe = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    try:
        probe.connect(sock_path)
        probe.close()
        return
    except OSError:
        pass
    import fcntl

    with open(LOCK_PATH, "w") as lockf:
        fcntl.flock(lockf, fcntl.LOCK_EX)
        try:
            probe = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            try:
                probe.connect(sock_path)
                probe.close()
                return
            except OSError:
                pass
            env = dict(os.environ)
            env["KT_GPU_DATA_SOCK"] = sock_path
            cmd = [sys.executable, "-m",
                   "kubetorch_amd.data_store.pod_data_server",
                   "--sock", sock_path]
            if tcp_port:
                cmd += ["--tcp-port", str(tcp_port)]
            subprocess.Popen(cmd, env=env, stdout=subprocess.DEVNULL,
                             stderr=subprocess.DEVNULL, start_new_session=True)
        finally:
            fcntl.flock(lockf, fcntl.LOCK_UN)
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            probe = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
            probe.connect(sock_path)
            probe.close()
            return
        except OSError:
            time.sleep(0.1)
    raise RuntimeError("pod-data-server failed to start")


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--tcp-port", type=int, default=C.GPU_DATA_SERVER_TCP_PORT)
    ap.add_argument("--sock", default=SOCK_PATH)
    args = ap.parse_args()
    server = PodDataServer(sock_path=args.sock, tcp_port=args.tcp_port)
    server.serve_forever()


if __name__ == "__main__":
    main()
