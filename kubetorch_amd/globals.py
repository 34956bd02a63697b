"""Process-wide singletons: ControllerClient (retrying HTTP client of the
controller API) and service URL resolution (in-cluster vs local vs
port-forward). (Reference parity: python_client/kubetorch/globals.py.)"""
import os
import threading
import time

import httpx

from kubetorch_amd import constants as C
from kubetorch_amd.config import config

RETRY_STATUS = (502, 503)
MAX_ATTEMPTS = 5


class ControllerClient:
    """HTTP client of the controller with retry/backoff on 502/503 and
    connection errors (reference: globals.py:372-501)."""

    def __init__(self, base_url=None):
        self._base_url = base_url
        self._client = httpx.Client(timeout=C.HTTP_TIMEOUT)

    @property
    def base_url(self):
        if self._base_url:
            return self._base_url
        url = config.api_url
        if url:
            self._base_url = url.rstrip("/")
            return self._base_url
        # local mode: in-process controller
        from kubetorch_amd.controller.local import ensure_local_controller

        self._base_url = ensure_local_controller()
        return self._base_url

    def _request(self, method, path, **kw):
        last = None
        backoff = 0.5
        for _ in range(MAX_ATTEMPTS):
            try:
                r = self._client.request(method, self.base_url + path, **kw)
                if r.status_code in RETRY_STATUS:
                    last = RuntimeError(f"{r.status_code}: {r.text[:200]}")
                else:
                    return r
            except (httpx.ConnectError, httpx.ReadTimeout, httpx.RemoteProtocolError) as e:
                last = e
            time.sleep(backoff)
            backoff = min(backoff * 2, 8)
        raise RuntimeError(f"controller unreachable after {MAX_ATTEMPTS} attempts: {last}")

    def deploy(self, name, namespace, manifest=None, metadata=None,
               service_config=None, launch_id=None, timeout=None):
        r = self._request(
            "POST", "/controller/deploy",
            json={"name": name, "namespace": namespace, "manifest": manifest,
                  "metadata": metadata, "service_config": service_config,
                  "launch_id": launch_id},
            timeout=timeout or C.LAUNCH_TIMEOUT,
        )
        data = r.json()
        if r.status_code != 200:
            from kubetorch_amd.exceptions import LaunchError

            raise LaunchError(str(data))
        return data

    def put_volume(self, volume, namespace):
        r = self._request("POST", f"/controller/volumes/{namespace}",
                          json=volume.to_pvc_manifest(namespace))
        r.raise_for_status()
        return r.json()

    def list_volumes(self, namespace):
        return self._request(
            "GET", f"/controller/volumes/{namespace}").json().get("volumes", [])

    def delete_volume(self, name, namespace):
        return self._request(
            "DELETE", f"/controller/volumes/{namespace}/{name}").json()

    def put_secret(self, secret, namespace):
        """Create/update a Secret through the controller (reference:
        controller-side kubernetes_secrets_client)."""
        r = self._request(
            "POST", f"/controller/secrets/{namespace}",
            json={"name": secret.name, "k8s_name": secret.k8s_name,
                  "values": secret.values, "as_env": secret.as_env,
                  "mount_path": secret.mount_path})
        r.raise_for_status()
        return r.json()

    def list_secrets(self, namespace):
        return self._request(
            "GET", f"/controller/secrets/{namespace}").json().get("secrets", [])

    def delete_secret(self, name, namespace):
        return self._request(
            "DELETE", f"/controller/secrets/{namespace}/{name}").json()

    def register_workload(self, name, namespace, **body):
        return self._request(
            "POST", "/controller/workload",
            json={"name": name, "namespace": namespace, **body},
        ).json()

    def get_workload(self, name, namespace):
        r = self._request("GET", f"/controller/workload/{namespace}/{name}")
        return r.json() if r.status_code == 200 else None

    def list_workloads(self, namespace):
        return self._request("GET", f"/controller/workloads/{namespace}").json()

    def delete_workload(self, name, namespace):
        return self._request(
            "DELETE", f"/controller/workload/{namespace}/{name}"
        ).json()

    def service_events(self, name, namespace, since=0.0):
        """Launch/lifecycle events for a service's pods (streamed by
        Module.to while waiting for readiness)."""
        r = self._request(
            "GET", f"/controller/events/{namespace}/{name}",
            params={"since": since},
        )
        return r.json().get("events", [])


_controller = None
_lock = threading.Lock()


def controller_client() -> ControllerClient:
    global _controller
    with _lock:
        if _controller is None:
            _controller = ControllerClient()
        return _controller


class PortForward:
    """kubectl port-forward lifecycle for out-of-cluster clients
    (reference: globals.py:123-300). No-op in local mode."""

    def __init__(self, target, namespace, local_port, remote_port):
        self.target = target
        self.namespace = namespace
        self.local_port = local_port
        self.remote_port = remote_port
        self.proc = None

    def start(self, timeout=20):
        import shutil
        import socket
        import subprocess

        if shutil.which("kubectl") is None:
            raise RuntimeError("kubectl not found; port-forward unavailable")
        self.proc = subprocess.Popen(
            ["kubectl", "-n", self.namespace, "port-forward", self.target,
             f"{self.local_port}:{self.remote_port}"],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        )
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                s = socket.create_connection(("127.0.0.1", self.local_port), 1)
                s.close()
                return self
            except OSError:
                if self.proc.poll() is not None:
                    raise RuntimeError("kubectl port-forward exited")
                time.sleep(0.2)
        raise RuntimeError("port-forward did not become ready")

    def stop(self):
        if self.proc is not None and self.proc.poll() is None:
            self.proc.terminate()

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()


def service_url(service_name, namespace, hosts=None):
    """Resolve the base URL for a deployed service. Local mode / explicit
    hosts -> first pod; in-cluster -> the K8s Service DNS."""
    if hosts:
        return f"http://{hosts[0]}"
    if os.environ.get("KUBERNETES_SERVICE_HOST"):
        return f"http://{service_name}.{namespace}.svc.cluster.local:{C.SERVER_PORT}"
    return f"http://{service_name}.{namespace}:{C.SERVER_PORT}"
