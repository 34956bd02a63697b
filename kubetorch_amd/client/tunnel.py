"""TCP tunnel for out-of-cluster clients (reference parity:
data_store/websocket_tunnel.py). A local TCP listener bridges every
connection to an in-cluster service THROUGH the controller's public port,
so clients behind firewalls reach the data store / worker pods with only
the controller exposed:

    with TcpTunnel("kubetorch-data-store", 8080) as t:
        os.environ["KT_STORE_URL"] = f"http://127.0.0.1:{t.local_port}"
        kt.put("key", "./dir")

Each connection uses two chunked-HTTP streams via the controller
(/controller/tunnel/{id}/up and /down) — no WebSocket stack required.
"""
import socket
import threading

import httpx

from kubetorch_amd.globals import controller_client


class TcpTunnel:
    def __init__(self, service, port, namespace="default", local_port=0,
                 controller_url=None):
        self.service = service
        self.port = port
        self.namespace = namespace
        self.controller_url = controller_url or controller_client().base_url
        self._lsock = socket.socket()
        self._lsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._lsock.bind(("127.0.0.1", local_port))
        self._lsock.listen(16)
        self.local_port = self._lsock.getsockname()[1]
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._accept_loop, daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        try:
            self._lsock.close()
        except OSError:
            pass

    __enter__ = lambda self: self.start()  # noqa: E731

    def __exit__(self, *exc):
        self.stop()

    def _accept_loop(self):
        while not self._stop.is_set():
            try:
                conn, _ = self._lsock.accept()
            except OSError:
                return
            threading.Thread(target=self._bridge, args=(conn,),
                             daemon=True).start()

    def _bridge(self, conn):
        base = self.controller_url.rstrip("/")
        try:
            r = httpx.post(f"{base}/controller/tunnel/open",
                           json={"namespace": self.namespace,
                                 "service": self.service, "port": self.port},
                           timeout=30)
            r.raise_for_status()
            tid = r.json()["tunnel_id"]
        except Exception:
            conn.close()
            return

        def up():
            def gen():
                while True:
                    try:
                        data = conn.recv(65536)
                    except OSError:
                        return
                    if not data:
                        return
                    yield data

            try:
                httpx.post(f"{base}/controller/tunnel/{tid}/up",
                           content=gen(), timeout=None)
            except Exception:
                pass

        def down():
            try:
                with httpx.stream(
                        "GET", f"{base}/controller/tunnel/{tid}/down",
                        timeout=None) as resp:
                    for chunk in resp.iter_bytes():
                        if chunk:
                            conn.sendall(chunk)
            except Exception:
                pass
            finally:
                try:
                    conn.shutdown(socket.SHUT_WR)
                except OSError:
                    pass
                conn.close()

        threading.Thread(target=up, daemon=True).start()
        threading.Thread(target=down, daemon=True).start()
