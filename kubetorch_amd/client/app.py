"""App: deploy an arbitrary CLI/HTTP app (reference parity:
resources/compute/app.py). The command ships as the pod's main process; an
optional health path gates readiness."""
import os

from kubetorch_amd.client.module import Module, sanitize_name
from kubetorch_amd.resources.compute import Compute
from kubetorch_amd.config import config


class App(Module):
    module_type = "app"

    def __init__(self, command, name=None, compute=None, health_path=None,
                 port=None):
        pointers = {
            "name": name or "app",
            "file_path": "",
            "rel_path": "",
            "project_root": os.getcwd(),
        }
        super().__init__(pointers, name=name, compute=compute)
        self.command = command
        self.health_path = health_path
        self.port = port

    def metadata(self):
        md = super().metadata()
        md["command"] = self.command
        md["health_path"] = self.health_path
        md["port"] = self.port
        return md

    def to(self, compute: Compute = None, **kw):
        if compute is not None:
            self.compute = compute
        if self.compute is None:
            self.compute = Compute(cpus=1)
        # app-mode pods run the user command directly
        if self.compute._raw_manifest is None and not self.compute.local:
            manifest = self.compute.to_manifest(self.name,
                                                username=config.username)
            container = manifest["spec"]["template"]["spec"]["containers"][0]
            container["command"] = ["bash", "-lc", self.command]
            self.compute._raw_manifest = manifest
        return super().to()

    def _wait_ready(self, timeout=None, reloaded=False):
        if not self.health_path:
            return  # fire-and-forget command (kt run)
        # an app serves ITS OWN http endpoint: poll health_path for any
        # non-5xx answer (our /ready route doesn't exist there)
        import time as _t

        import httpx

        from kubetorch_amd.globals import controller_client

        deadline = _t.time() + (timeout or 900)
        path = "/" + self.health_path.lstrip("/")
        while _t.time() < deadline:
            try:
                w = controller_client().get_workload(self.name, self.namespace)
                pods = (w or {}).get("pods") or self.service_hosts
            except Exception:
                pods = self.service_hosts
            for host in pods:
                port = self.port or host.split(":")[-1]
                url = f"http://{host.split(':')[0]}:{port}{path}"
                try:
                    if httpx.get(url, timeout=3).status_code < 500:
                        return
                except httpx.HTTPError:
                    pass
            _t.sleep(0.5)
        from kubetorch_amd.exceptions import LaunchError

        raise LaunchError(
            f"app {self.name} health path {path} not answering "
            f"after {timeout or 900}s")


    def wait(self, timeout=900, poll=1.0, follow=True, printer=print):
        """Foreground mode: block until the app's pods exit, streaming the
        app's stdout/stderr while waiting (reference: App deploy --follow
        / _wait_for_app_exit). Returns True if the app finished within
        the timeout."""
        import time as _t

        from kubetorch_amd.globals import controller_client

        cc = controller_client()
        offset = 0
        deadline = _t.time() + timeout

        def drain():
            nonlocal offset
            if not follow:
                return
            try:
                r = cc._request(
                    "GET",
                    f"/controller/podlogs/{self.namespace}/{self.name}",
                    params={"offset": offset}).json()
            except Exception:
                return
            if r.get("text"):
                for line in r["text"].splitlines():
                    printer(f"[{self.name}] {line}")
            offset = r.get("offset", offset)

        while _t.time() < deadline:
            drain()
            w = cc.get_workload(self.name, self.namespace)
            pods = (w or {}).get("pods") or []
            if not pods:
                drain()  # final tail
                return True
            _t.sleep(poll)
        return False


def app(command, name=None, health_path=None, port=None):
    return App(command, name=sanitize_name(name) if name else None,
               health_path=health_path, port=port)
