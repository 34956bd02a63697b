"""Peer file server for filesystem tree broadcast.

When W pods all need the same large key (model weights, datasets), pulling
W copies from the namespace store serializes on its NIC. The reference
solves this with a rolling fs-broadcast tree (pod_data_server.py
fs-broadcast completion tracking + metadata join_fs_broadcast, fanout ~50):
each pod that finishes downloading becomes a source for up to `fanout`
later pods, so total store egress is O(1) and the fan-out is exponential.

This module is the "become a source" half: a tiny threaded HTTP server that
serves keys a pod has registered locally (files raw, directories as tar.gz
with the same X-KT-Tar convention as the store). The coordination half
lives in data_store/server.py (/fsbcast/*) and the client half in
data_store/commands.py (get_broadcast).
"""
import io
import os
import socket
import tarfile
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

_registry = {}
_registry_lock = threading.Lock()


def register_local(key, path):
    """Make `path` servable to broadcast children under `key`."""
    with _registry_lock:
        _registry[key.strip("/")] = os.path.abspath(path)


def lookup_local(key):
    with _registry_lock:
        return _registry.get(key.strip("/"))


class _Handler(BaseHTTPRequestHandler):
    def log_message(self, *a):  # quiet
        pass

    def do_GET(self):
        if not self.path.startswith("/files/"):
            self.send_error(404)
            return
        key = self.path[len("/files/"):].strip("/")
        path = lookup_local(key)
        if path is None or not os.path.exists(path):
            self.send_error(404)
            return
        if os.path.isdir(path):
            buf = io.BytesIO()
            with tarfile.open(fileobj=buf, mode="w:gz") as tar:
                tar.add(path, arcname=".")
            body = buf.getvalue()
            self.send_response(200)
            self.send_header("X-KT-Tar", "1")
        else:
            with open(path, "rb") as f:
                body = f.read()
            self.send_response(200)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


class BcastFileServer:
    """Threaded HTTP file server over the broadcast registry. One per pod,
    started lazily the first time the pod completes a broadcast get."""

    def __init__(self, host="0.0.0.0", port=0, advertise_host=None):
        self._srv = ThreadingHTTPServer((host, port), _Handler)
        self.port = self._srv.server_address[1]
        self.host = advertise_host or _local_ip()
        self._thread = threading.Thread(target=self._srv.serve_forever,
                                        daemon=True)
        self._thread.start()

    @property
    def url(self):
        return f"http://{self.host}:{self.port}"

    def close(self):
        self._srv.shutdown()
        self._srv.server_close()


def _local_ip():
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect(("10.255.255.255", 1))
        ip = s.getsockname()[0]
        s.close()
        return ip
    except OSError:
        return "127.0.0.1"


_server = None
_server_lock = threading.Lock()


def ensure_server():
    """Lazy per-process singleton server (pods reuse one across keys)."""
    global _server
    with _server_lock:
        if _server is None:
            _server = BcastFileServer()
        return _server
