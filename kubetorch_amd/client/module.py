"""Module: shared deploy/call machinery for Fn/Cls/App.

Deploy path (reference parity: module.py:486 Module.to + compute.py:2006):
  1. name resolution with username prefix
  2. sync the working dir to the data store (k8s) — skipped in local mode
  3. build module metadata (pointers + dispatch + procs)
  4. controller /deploy: apply manifest, hot-reload warm pods, ack barrier
  5. wait for /ready?launch_id on the service
The hot loop (re-.to() with changed code) reuses warm pods: rsync delta +
reload broadcast, no pod restart."""
import base64
import pickle
import re
import time
import uuid

from kubetorch_amd import constants as C
from kubetorch_amd.client.http_client import HTTPClient, shared_client
from kubetorch_amd.resources.compute import Compute
from kubetorch_amd.config import config
from kubetorch_amd.exceptions import LaunchError
from kubetorch_amd.globals import controller_client


def sanitize_name(name):
    name = re.sub(r"[^a-z0-9-]", "-", name.lower()).strip("-")
    return name[:63]


class Module:
    module_type = "fn"

    def __init__(self, pointers, name=None, compute=None, init_args=None):
        self.pointers = pointers
        self._name = name
        self.compute = compute
        self.init_args = init_args
        self.service_hosts = []
        self.launch_id = None
        self._http = None
        self._exec_token = None  # per-deploy shared secret for run_bash
        self.stream_logs = bool(config.get("stream_logs"))

    @property
    def name(self):
        if getattr(self, "_bound_full_name", None):
            return self._bound_full_name  # bound to an existing service
        base = self._name or self.pointers["name"]
        return sanitize_name(f"{config.username}-{base}")

    @property
    def namespace(self):
        return (self.compute.namespace if self.compute else config.namespace)

    # -- deploy ----------------------------------------------------------------
    def metadata(self):
        md = {
            "module_name": self.name,
            "callable_name": self.pointers["name"],
            "module_type": self.module_type,
            "file_path": self.pointers["file_path"],
            "rel_path": self.pointers.get("rel_path"),
            "project_root": self.pointers["project_root"],
            "workdir_key": f"{self.namespace}/{self.name}/workdir",
            "distributed_config": (self.compute.distributed_config
                                   if self.compute else None),
            "image_setup": (self.compute.image_setup_contents()
                            if self.compute else ""),
            "allowed_serialization": ",".join(
                self.compute.allowed_serialization) if self.compute else "json",
        }
        if self._exec_token:
            md["exec_token"] = self._exec_token
        if self.init_args is not None:
            md["init_args"] = base64.b64encode(
                pickle.dumps(self.init_args)).decode()
        return md

    def _sync_workdir(self, md):
        if self.pointers.get("remote"):
            return  # remote_dir mode: the code is baked into the image
        if self.compute is not None and self.compute.local:
            return  # same filesystem: pods import the original paths
        try:
            from kubetorch_amd.data_store import commands as ds

            ds.put(md["workdir_key"], src=self.pointers["project_root"])
        except Exception:
            pass  # no data store deployed: pods must have the code baked in

    def _reload_prefixes(self, reload_prefixes):
        """Default service-lookup order: username, current git branch, prod
        (reference: Module.to get_if_exists / reload_prefixes)."""
        if reload_prefixes:
            return list(reload_prefixes)
        prefixes = [config.username]
        try:
            import subprocess

            br = subprocess.run(
                ["git", "rev-parse", "--abbrev-ref", "HEAD"],
                capture_output=True, text=True, timeout=5,
            ).stdout.strip()
            if br and br != "HEAD":
                prefixes.append(br)
        except Exception:
            pass
        prefixes.append("prod")
        return prefixes

    def _get_existing_service(self, reload_prefixes=None):
        """Find an already-deployed service for this module under the
        fallback prefixes. Returns the workload dict or None."""
        base = self._name or self.pointers["name"]
        candidates = [sanitize_name(f"{p}-{base}")
                      for p in self._reload_prefixes(reload_prefixes)]
        try:
            workloads = controller_client().list_workloads(
                self.namespace).get("workloads", [])
        except Exception:
            return None
        by_name = {w["name"]: w for w in workloads}
        for cand in candidates:
            if cand in by_name:
                return by_name[cand]
        return None

    def to(self, compute: Compute = None, init_args=None,
           get_if_exists=False, reload_prefixes=None):
        """Deploy (or hot-reload) this module onto the compute.

        get_if_exists=True: before launching, look for an existing service
        under the fallback prefixes (username -> git branch -> prod, or an
        explicit reload_prefixes list) and bind to it instead of deploying
        (reference parity: Module.to get_if_exists)."""
        if compute is not None:
            self.compute = compute
        if init_args is not None:
            self.init_args = init_args
        if self.compute is None:
            self.compute = Compute(cpus=1)
        if get_if_exists:
            w = self._get_existing_service(reload_prefixes)
            if w is not None:
                # bind exactly to the found service name (bypasses the
                # username prefixing in .name)
                found = w["name"]
                self._bound_full_name = found
                self.launch_id = w.get("launch_id")
                self.service_hosts = w.get("pods") or []
                if not self.service_hosts:
                    try:
                        full = controller_client()._request(
                            "GET",
                            f"/controller/workload/{self.namespace}/{found}",
                        ).json()
                        self.service_hosts = full.get("pods") or []
                    except Exception:
                        pass
                self._http = None
                if self.stream_logs:
                    print(f"[kt] reusing existing service {found}")
                return self
        t0 = time.time()
        self._exec_token = uuid.uuid4().hex
        md = self.metadata()
        self._sync_workdir(md)
        # secrets and PVCs must exist before pods reference them
        # (reference: _upload_secrets_list / Volume create before apply)
        for s in (self.compute.secrets or []):
            controller_client().put_secret(s, self.namespace)
        for v in (self.compute.volumes or []):
            if getattr(v, "needs_create", False):
                controller_client().put_volume(v, self.namespace)
        launch_id = uuid.uuid4().hex[:12]
        manifest = self.compute.to_manifest(self.name, username=config.username,
                                            module=self.pointers["name"])
        service_config = {"kind": self.compute.kind}
        if self.compute.endpoint is not None:
            service_config["endpoint"] = self.compute.endpoint.to_service_config()
        resp = controller_client().deploy(
            name=self.name, namespace=self.namespace, manifest=manifest,
            metadata=md, launch_id=launch_id,
            service_config=service_config,
            timeout=self.compute.launch_timeout,
        )
        self.launch_id = resp.get("launch_id", launch_id)
        self.service_hosts = resp.get("hosts") or []
        self._http = None
        self._wait_ready(timeout=self.compute.launch_timeout,
                         reloaded=bool(resp.get("reloaded_pods")))
        elapsed = time.time() - t0
        if self.stream_logs:
            print(f"[kt] {self.name} ready in {elapsed:.2f}s "
                  f"({len(self.service_hosts) or self.compute.replicas} pods)")
        return self

    async def to_async(self, compute=None, init_args=None):
        import asyncio

        return await asyncio.to_thread(self.to, compute, init_args)

    def _base_url(self):
        from kubetorch_amd.globals import service_url

        default = service_url(self.name, self.namespace, self.service_hosts)
        ep = self.compute.endpoint if self.compute else None
        if ep is not None:
            return ep.resolve(default_url=default, hosts=self.service_hosts)
        return default

    @property
    def http(self) -> HTTPClient:
        if self._http is None:
            self._http = HTTPClient(self._base_url(), self.pointers["name"])
        return self._http

    def _wait_ready(self, timeout=C.LAUNCH_TIMEOUT, reloaded=False):
        """Poll /ready?launch_id until the pod finished loading this deploy,
        streaming service events (pod scheduled/started/probe failures —
        reference parity: K8s launch-event streaming) while waiting. If pods
        were hot-reloaded through the controller ack barrier, they are ready
        by construction, but poll once to verify."""
        deadline = time.time() + timeout
        delay = 0.05
        ev_since = 0.0
        last_ev_poll = 0.0
        while time.time() < deadline:
            if self.http.is_ready(launch_id=None if not reloaded else self.launch_id):
                # accept pods that don't carry a launch_id (direct-env launch)
                return
            if self.stream_logs and time.time() - last_ev_poll > 1.0:
                last_ev_poll = time.time()
                ev_since = self._print_events(ev_since)
            time.sleep(delay)
            delay = min(delay * 1.5, 2.0)
        self._print_events(ev_since)  # surface the failure reason
        raise LaunchError(f"service {self.name} not ready after {timeout}s")

    def _print_events(self, since):
        try:
            evs = controller_client().service_events(
                self.name, self.namespace, since=since)
        except Exception:
            return since
        for e in evs:
            tag = "!" if e.get("type") == "Warning" else "·"
            pod = f" {e['pod']}" if e.get("pod") else ""
            print(f"[kt] {tag}{pod} {e.get('reason', '')}: "
                  f"{e.get('message', '')}")
            since = max(since, e.get("ts", since))
        return since

    # -- lifecycle ---------------------------------------------------------------
    def teardown(self):
        controller_client().delete_workload(self.name, self.namespace)
        self.service_hosts = []
        self._http = None

    def logs(self, since=0, limit=1000):
        return self.http.logs(since=since, limit=limit)

    def workload(self):
        return controller_client().get_workload(self.name, self.namespace)

    # -- post-launch helpers (reference: compute.py ssh/pip_install/run_bash) --
    def run_bash(self, command, timeout=600):
        """Run a bash command inside the service's (first) pod. Requires the
        per-deploy exec token (set on the pod at deploy time); a service bound
        via get_if_exists has no token and cannot exec."""
        if not self._exec_token:
            raise PermissionError(
                "run_bash requires the exec token from this client's own "
                "deploy (.to()); a service bound with get_if_exists cannot "
                "be exec'd remotely")
        r = shared_client().post(self.http.base_url + "/exec",
                                 json={"command": command, "timeout": timeout},
                                 headers={"X-KT-Exec-Token": self._exec_token},
                                 timeout=timeout + 30)
        r.raise_for_status()
        return r.json()

    def ssh(self, command=None, pod_index=0):
        """Shell access to a service pod (reference: Compute.ssh). With
        `command`: run it and return {returncode, stdout, stderr} (exec
        route, token-gated). Without: attach an interactive shell —
        kubectl exec -it in-cluster; on the local driver pods are local
        processes, so a command is required."""
        if command is not None:
            return self.run_bash(command)
        if self.compute is not None and self.compute.local:
            raise RuntimeError(
                "interactive ssh targets a cluster pod; local-driver pods "
                "are subprocesses on this machine — pass command=... "
                "instead")
        import subprocess

        pods = (self.workload() or {}).get("pods") or self.service_hosts
        if not pods:
            raise RuntimeError(f"no pods found for {self.name}")
        pod_name = f"{self.name}-{pod_index}"
        return subprocess.call(
            ["kubectl", "-n", self.namespace, "exec", "-it",
             pod_name, "--", "bash"])

    def pip_install(self, packages, extra_args=""):
        if isinstance(packages, str):
            packages = [packages]
        import shlex
        import sys

        pkgs = " ".join(shlex.quote(p) for p in packages)
        return self.run_bash(
            f"{shlex.quote(sys.executable)} -m pip install {pkgs} {extra_args}")

    def _lb_client(self):
        """Round-robin over the service's live pods for autoscaled
        (knative-kind) services — the role the K8s Service/Knative
        activator plays in-cluster. Pod list refreshed from the
        controller every 2 s so scale-ups receive traffic."""
        now = time.time()
        if now - getattr(self, "_lb_ts", 0) > 2.0:
            try:
                w = controller_client().get_workload(self.name, self.namespace)
                hosts = w.get("pods") or []
            except Exception:
                hosts = []
            self._lb_hosts = hosts or self.service_hosts
            self._lb_ts = now
        if not self._lb_hosts:
            return self.http  # no live pods known: fall back to the
            # deploy-time URL (its retry loop rides out a scale-from-zero)
        self._lb_i = getattr(self, "_lb_i", -1) + 1
        host = self._lb_hosts[self._lb_i % len(self._lb_hosts)]
        return HTTPClient(f"http://{host}", self.pointers["name"])

    def _call(self, args, kwargs, method=None, **opts):
        client = self.http
        if (self.compute is not None and self.compute.kind == "knative"
                and self.compute.local):
            client = self._lb_client()
        return client.call(
            args=args, kwargs=kwargs, method=method,
            stream_logs=opts.pop("stream_logs", self.stream_logs),
            **opts,
        )
