"""Controller drivers: how workload manifests become running pods.

* K8sDriver  — applies manifests through kubectl / the K8s API (in-cluster).
* LocalDriver — "pods" are local http_server subprocesses on localhost
  ports. This is the fake-cluster mode that makes the whole control plane
  testable end-to-end without Kubernetes (and the `kt` local dev story);
  the reference only fakes the supervisor layer (LOCAL_IPS), we fake the
  pod layer too.
"""
import json
import os
import shutil
import socket
import subprocess
import sys
import time

from kubetorch_amd import constants as C


def desired_replicas(manifest):
    """Replica count across manifest kinds: plain spec.replicas, or the
    sum over a Kubeflow training job's replica specs."""
    spec = manifest.get("spec", {})
    if "replicas" in spec:
        return spec["replicas"] or 1
    for key in ("pytorchReplicaSpecs", "tfReplicaSpecs", "mxReplicaSpecs",
                "xgbReplicaSpecs"):
        if key in spec:
            return sum(rs.get("replicas", 1)
                       for rs in spec[key].values()) or 1
    if "headGroupSpec" in spec:  # RayCluster: head + worker groups
        return 1 + sum(g.get("replicas", 0)
                       for g in spec.get("workerGroupSpecs", []))
    return 1


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class LocalPod:
    def __init__(self, name, port, proc):
        self.name = name
        self.port = port
        self.proc = proc
        self.log_path = None

    @property
    def host(self):
        return f"127.0.0.1:{self.port}"

    def alive(self):
        return self.proc.poll() is None

    def kill(self):
        if self.alive():
            self.proc.terminate()
            try:
                self.proc.wait(5)
            except subprocess.TimeoutExpired:
                self.proc.kill()


class LocalDriver:
    """Spawns one http_server subprocess per replica; KT_LOCAL_IPS carries
    the peer list (discovery), KT_CONTROLLER_URL points back at us."""

    def __init__(self, controller_url=None):
        self.controller_url = controller_url
        self.services = {}  # (ns, name) -> [LocalPod]
        self.events = {}    # (ns, name) -> [event dict] (launch streaming)
        self._event_state = {}  # pod name -> last seen liveness
        self._master_ports = {}  # (ns, name) -> rendezvous port (stable
        # across pod replacement so re-joined ranks find the same master)
        self.secrets = {}  # (ns, k8s_name) -> {name, values, as_env, mount_path}
        # deploys, the pod monitor and the autoscaler all reconcile
        # concurrently; serialize applies so a service can't double-spawn
        import threading

        self._apply_lock = threading.Lock()

    def _event(self, namespace, name, reason, message, pod=None):
        self.events.setdefault((namespace, name), []).append({
            "ts": time.time(), "type": "Normal", "reason": reason,
            "message": message, "pod": pod,
        })

    def get_events(self, name, namespace, since=0.0):
        """Pod lifecycle events for a service (reference parity: K8s events
        streamed by the client during launch). Lazily emits exit events for
        pods that died since the last poll."""
        key = (namespace, name)
        for p in self.services.get(key, []):
            alive = p.alive()
            was = self._event_state.get(p.name)
            if was is None:
                self._event_state[p.name] = alive
            elif was and not alive:
                self._event_state[p.name] = False
                ev = self.events.setdefault(key, [])
                ev.append({"ts": time.time(), "type": "Warning",
                           "reason": "BackOff",
                           "message": f"pod exited rc={p.proc.returncode}",
                           "pod": p.name})
        return [e for e in self.events.get(key, []) if e["ts"] > since]

    def apply(self, manifest, namespace, metadata=None, launch_id=None):
        """Reconcile the service to the manifest's replica count. Warm
        (alive) pods are kept — a hot reload reaches them through the
        controller push channel — and only dead/missing pods are spawned
        (the K8s-Deployment-controller behavior the reference delegates to
        Kubernetes; here the control plane does its own reconciliation,
        which is what makes mid-step pod death auto-heal)."""
        with self._apply_lock:
            return self._apply_locked(manifest, namespace, metadata,
                                      launch_id)

    def _apply_locked(self, manifest, namespace, metadata, launch_id):
        name = manifest["metadata"]["name"]
        replicas = desired_replicas(manifest)
        key = (namespace, name)
        pods = self.services.get(key, [])
        alive = [p for p in pods if p.alive()]
        if len(alive) == replicas:
            self._event(namespace, name, "Reloaded",
                        f"hot reload into {len(alive)} warm pod(s)")
            self.services[key] = alive
            return [p.host for p in alive]  # warm pods: reload only
        if len(alive) > replicas:  # scale down: drop the newest extras
            for p in alive[replicas:]:
                p.kill()
            alive = alive[:replicas]
        n_new = replicas - len(alive)
        used_idx = {int(p.name.rsplit("-", 1)[1]) for p in alive}
        new_idx = [i for i in range(replicas + len(used_idx))
                   if i not in used_idx][:n_new]
        ports = [_free_port() for _ in range(n_new)]
        peer_list = ",".join([p.host for p in alive]
                             + [f"127.0.0.1:{p}" for p in ports])
        # one stable rendezvous port for the whole service lifetime
        master_port = self._master_ports.setdefault(key, _free_port())
        new_pods = []
        md = metadata or {}
        for i, port in zip(new_idx, ports):
            env = dict(os.environ)
            env.update({
                "KT_SERVER_PORT": str(port),
                "KT_SELF_HOST": f"127.0.0.1:{port}",
                C.ENV_LOCAL_IPS: peer_list,
                C.ENV_SERVICE_NAME: name,
                "POD_NAME": f"{name}-{i}",
                "POD_NAMESPACE": namespace,
                "POD_IP": "127.0.0.1",
                "KT_MASTER_PORT": str(master_port),
            })
            if launch_id:
                env[C.ENV_LAUNCH_ID] = str(launch_id)
            if self.controller_url:
                env["KT_CONTROLLER_URL"] = self.controller_url
                env["KT_PEERS_URL"] = self.controller_url
            for k, v in (md.get("env") or {}).items():
                env[k] = str(v)
            self._inject_secrets(manifest, namespace, env)
            self._inject_volumes(manifest, namespace, env)
            # module metadata -> env contract (same as controller push)
            if md.get("file_path"):
                env[C.ENV_FILE_PATH] = md["file_path"]
            if md.get("project_root"):
                env[C.ENV_PROJECT_ROOT] = md["project_root"]
            if md.get("callable_name"):
                env[C.ENV_CALLABLE_NAME] = md["callable_name"]
                env[C.ENV_MODULE_NAME] = md.get("module_name", md["callable_name"])
            if md.get("module_type"):
                env[C.ENV_MODULE_TYPE] = md["module_type"]
            if md.get("init_args"):
                env[C.ENV_INIT_ARGS] = md["init_args"]
            if md.get("distributed_config"):
                env[C.ENV_DISTRIBUTED_CONFIG] = json.dumps(md["distributed_config"])
            if md.get("module_type") == "app" and md.get("command"):
                # app mode: the user command IS the pod main process
                env["KT_APP_PORT"] = str(md.get("port") or port)
                cmd = ["bash", "-lc", md["command"]]
            else:
                cmd = [sys.executable, "-m",
                       "kubetorch_amd.serving.http_server",
                       "--port", str(port), "--host", "127.0.0.1"]
            # pod stdout/stderr goes to a per-pod log file (the stand-in
            # for `kubectl logs`; /controller/podlogs serves it)
            import tempfile

            logdir = os.path.join(tempfile.gettempdir(), "kt-pod-logs",
                                  namespace)
            os.makedirs(logdir, exist_ok=True)
            log_path = os.path.join(logdir, f"{name}-{i}.log")
            logf = open(log_path, "ab")
            proc = subprocess.Popen(cmd, env=env, stdout=logf, stderr=logf)
            logf.close()
            pod = LocalPod(f"{name}-{i}", port, proc)
            pod.log_path = log_path
            new_pods.append(pod)
            self._event(namespace, name, "Scheduled",
                        f"assigned 127.0.0.1:{port}", pod=f"{name}-{i}")
            self._event(namespace, name, "Started",
                        "container started", pod=f"{name}-{i}")
            self._event_state[f"{name}-{i}"] = True
        self.services[key] = alive + new_pods
        return [p.host for p in self.services[key]]

    def delete(self, name, namespace):
        for p in self.services.pop((namespace, name), []):
            p.kill()
        self.events.pop((namespace, name), None)
        self._master_ports.pop((namespace, name), None)

    # -- secrets (reference: controller-side kubernetes_secrets_client) ----
    def apply_secret(self, spec, namespace):
        k8s_name = spec.get("k8s_name") or \
            f"kt-secret-{spec['name']}".lower().replace("_", "-")
        self.secrets[(namespace, k8s_name)] = dict(spec, k8s_name=k8s_name)

    def list_secrets(self, namespace):
        return [{"name": s["name"], "k8s_name": k,
                 "keys": sorted(s.get("values", {}))}
                for (ns, k), s in self.secrets.items() if ns == namespace]

    def delete_secret(self, name, namespace):
        k8s_name = f"kt-secret-{name}".lower().replace("_", "-")
        self.secrets.pop((namespace, k8s_name), None)
        self.secrets.pop((namespace, name), None)

    # -- volumes (local stand-in: host dirs) --------------------------------
    def _volume_root(self, namespace, claim):
        import tempfile

        return os.path.join(tempfile.gettempdir(), "kt-local-volumes",
                            namespace, claim)

    def apply_volume(self, manifest, namespace):
        claim = manifest.get("metadata", {}).get("name")
        os.makedirs(self._volume_root(namespace, claim), exist_ok=True)
        self.volumes = getattr(self, "volumes", {})
        self.volumes[(namespace, claim)] = manifest

    def list_volumes(self, namespace):
        vols = getattr(self, "volumes", {})
        return [{"name": c, "spec": m.get("spec", {})}
                for (ns, c), m in vols.items() if ns == namespace]

    def delete_volume(self, name, namespace):
        import shutil as _sh

        getattr(self, "volumes", {}).pop((namespace, name), None)
        _sh.rmtree(self._volume_root(namespace, name), ignore_errors=True)

    def _inject_volumes(self, manifest, namespace, env):
        """Expose PVC mounts to local pods as host dirs via
        KT_VOLUME_MOUNT_<NAME> (a pod subprocess cannot be given a real
        mount namespace; same convention as secrets)."""
        try:
            spec = (manifest.get("spec", {}).get("template", {})
                    .get("spec", {})) or {}
            claims = {v["name"]: v["persistentVolumeClaim"]["claimName"]
                      for v in spec.get("volumes", []) or []
                      if "persistentVolumeClaim" in v}
            c = (spec.get("containers") or [{}])[0]
            for vm in c.get("volumeMounts", []) or []:
                claim = claims.get(vm["name"])
                if not claim:
                    continue
                d = self._volume_root(namespace, claim)
                os.makedirs(d, exist_ok=True)
                env["KT_VOLUME_MOUNT_" + vm["name"].upper()
                    .replace("-", "_")] = d
        except Exception:
            pass

    def _inject_secrets(self, manifest, namespace, env):
        """Materialize the pod template's secret references into the local
        pod's environment: envFrom secretRefs become env vars; secret
        volumes are written under a scratch dir with
        KT_SECRET_MOUNT_<NAME>=<dir> pointing at it (the stand-in for the
        in-cluster file mount)."""
        try:
            spec = (manifest.get("spec", {}).get("template", {})
                    .get("spec", {})) or {}
            containers = spec.get("containers") or [{}]
            c = containers[0]
            for ref in c.get("envFrom", []) or []:
                sname = ref.get("secretRef", {}).get("name")
                sec = self.secrets.get((namespace, sname))
                if sec:
                    for k, v in sec.get("values", {}).items():
                        env[k] = str(v)
            vol_secrets = {v["name"]: v["secret"]["secretName"]
                           for v in spec.get("volumes", []) or []
                           if "secret" in v}
            for vm in c.get("volumeMounts", []) or []:
                sname = vol_secrets.get(vm["name"])
                if not sname:
                    continue
                sec = self.secrets.get((namespace, sname))
                if not sec:
                    continue
                import tempfile

                d = os.path.join(tempfile.gettempdir(), "kt-local-secrets",
                                 namespace, sname)
                os.makedirs(d, exist_ok=True)
                os.chmod(d, 0o700)
                for k, v in sec.get("values", {}).items():
                    path = os.path.join(d, k)
                    with open(path, "w") as f:
                        f.write(str(v))
                    os.chmod(path, 0o600)
                env_key = "KT_SECRET_MOUNT_" + \
                    sec["name"].upper().replace("-", "_")
                env[env_key] = d
        except Exception:
            pass

    def pods(self, name, namespace):
        return [p.host for p in self.services.get((namespace, name), [])
                if p.alive()]

    def pod_logs(self, name, namespace, offset=0):
        """Raw pod process stdout/stderr (the kubectl-logs stand-in; app
        pods' command output lands here). Returns (text, new_offset)."""
        chunks = []
        end = offset
        for p in self.services.get((namespace, name), []):
            if not p.log_path or not os.path.exists(p.log_path):
                continue
            with open(p.log_path, "rb") as f:
                f.seek(offset)
                data = f.read()
            if data:
                chunks.append(data.decode(errors="replace"))
                end = max(end, offset + len(data))
        return "".join(chunks), end

    def teardown_all(self):
        for key in list(self.services):
            self.delete(key[1], key[0])


class K8sDriver:
    """Applies manifests via kubectl (the controller pod has RBAC for this;
    reference ships this logic inside its closed-source controller image).
    apply() verifies the rollout actually converged (kubectl rollout
    status for Deployments, pod readiness for job kinds) instead of
    fire-and-forgetting the manifest."""

    ROLLOUT_KINDS = {"Deployment", "StatefulSet", "DaemonSet"}

    def __init__(self, kubectl="kubectl",
                 rollout_timeout=None):
        self.kubectl = kubectl
        self.rollout_timeout = rollout_timeout or int(
            os.environ.get("KT_ROLLOUT_TIMEOUT", "600"))

    def available(self):
        return shutil.which(self.kubectl) is not None

    def apply(self, manifest, namespace, metadata=None, launch_id=None):
        payload = json.dumps(manifest)
        r = subprocess.run(
            [self.kubectl, "-n", namespace, "apply", "-f", "-"],
            input=payload.encode(), capture_output=True,
        )
        if r.returncode != 0:
            raise RuntimeError(
                f"kubectl apply failed rc={r.returncode}: "
                f"{r.stderr.decode(errors='replace')[-800:]}")
        kind = manifest.get("kind", "")
        name = manifest.get("metadata", {}).get("name", "")
        if kind in self.ROLLOUT_KINDS:
            rs = subprocess.run(
                [self.kubectl, "-n", namespace, "rollout", "status",
                 f"{kind.lower()}/{name}",
                 f"--timeout={self.rollout_timeout}s"],
                capture_output=True,
            )
            if rs.returncode != 0:
                raise RuntimeError(
                    f"rollout of {kind}/{name} did not converge within "
                    f"{self.rollout_timeout}s: "
                    f"{rs.stderr.decode(errors='replace')[-800:]}")
            return self.pods(name, namespace)
        return []

    def delete(self, name, namespace, kind="deployment"):
        subprocess.run(
            [self.kubectl, "-n", namespace, "delete", kind, name,
             "--ignore-not-found"],
            check=True, capture_output=True,
        )

    # -- KubetorchWorkload CRs: durable workload registry ---------------------
    # The reference controller persists its registry as KubetorchWorkload
    # objects so a controller restart loses nothing; HUB rehydrates from
    # these at startup (charts/.../crds.yaml is the schema).
    def persist_workload(self, w):
        md = w.get("metadata") or {}
        cr = {
            "apiVersion": "kubetorch.amd.com/v1",
            "kind": "KubetorchWorkload",
            "metadata": {
                "name": w["name"], "namespace": w["namespace"],
                "labels": {C.SERVICE_LABEL: w["name"]},
                # full registry record (schema-free) for exact rehydration
                "annotations": {C.LABEL_PREFIX + "/state": json.dumps(w)},
            },
            "spec": {
                "serviceConfig": {
                    "kind": (w.get("service_config") or {}).get("kind", ""),
                    "name": w["name"],
                },
                "module": {
                    "type": md.get("module_type", "fn"),
                    "dispatch": ("spmd" if md.get("distributed_config")
                                 else "regular"),
                    "pointers": {k: str(md[k]) for k in
                                 ("callable_name", "file_path",
                                  "project_root") if md.get(k)},
                },
            },
        }
        r = subprocess.run(
            [self.kubectl, "-n", w["namespace"], "apply", "-f", "-"],
            input=json.dumps(cr).encode(), capture_output=True)
        if r.returncode != 0:
            raise RuntimeError(r.stderr.decode(errors="replace")[-500:])

    def load_workloads(self):
        """Rehydrate the registry records from KubetorchWorkload CRs."""
        out = subprocess.run(
            [self.kubectl, "get", "kubetorchworkloads", "-A", "-o", "json"],
            capture_output=True)
        if out.returncode != 0:
            return []
        records = []
        for it in json.loads(out.stdout or b"{}").get("items", []):
            raw = (it.get("metadata", {}).get("annotations", {})
                   .get(C.LABEL_PREFIX + "/state"))
            if raw:
                try:
                    records.append(json.loads(raw))
                except ValueError:
                    pass
        return records

    def delete_workload_cr(self, name, namespace):
        subprocess.run(
            [self.kubectl, "-n", namespace, "delete", "kubetorchworkload",
             name, "--ignore-not-found"], capture_output=True)

    def pods(self, name, namespace):
        out = subprocess.run(
            [self.kubectl, "-n", namespace, "get", "pods", "-l",
             f"{C.SERVICE_LABEL}={name}", "-o", "json"],
            check=True, capture_output=True,
        )
        items = json.loads(out.stdout).get("items", [])
        hosts = []
        for it in items:
            ip = it.get("status", {}).get("podIP")
            if ip:
                hosts.append(f"{ip}:{C.SERVER_PORT}")
        return hosts

    def apply_volume(self, manifest, namespace):
        r = subprocess.run(
            [self.kubectl, "-n", namespace, "apply", "-f", "-"],
            input=json.dumps(manifest).encode(), capture_output=True)
        if r.returncode != 0:
            raise RuntimeError(r.stderr.decode(errors="replace")[-500:])

    def list_volumes(self, namespace):
        out = subprocess.run(
            [self.kubectl, "-n", namespace, "get", "pvc", "-o", "json"],
            check=True, capture_output=True)
        return [{"name": it["metadata"]["name"],
                 "spec": it.get("spec", {})}
                for it in json.loads(out.stdout).get("items", [])]

    def delete_volume(self, name, namespace):
        subprocess.run(
            [self.kubectl, "-n", namespace, "delete", "pvc", name,
             "--ignore-not-found"],
            check=True, capture_output=True)

    def pod_logs(self, name, namespace, offset=0):
        out = subprocess.run(
            [self.kubectl, "-n", namespace, "logs", "-l",
             f"{C.SERVICE_LABEL}={name}", "--tail=1000"],
            capture_output=True, text=True)
        text = out.stdout if out.returncode == 0 else ""
        return text[offset:], len(text)

    def apply_secret(self, spec, namespace):
        import base64

        k8s_name = spec.get("k8s_name") or \
            f"kt-secret-{spec['name']}".lower().replace("_", "-")
        manifest = {
            "apiVersion": "v1", "kind": "Secret",
            "metadata": {"name": k8s_name, "namespace": namespace},
            "type": "Opaque",
            "data": {k: base64.b64encode(str(v).encode()).decode()
                     for k, v in (spec.get("values") or {}).items()},
        }
        subprocess.run(
            [self.kubectl, "-n", namespace, "apply", "-f", "-"],
            input=json.dumps(manifest).encode(), check=True,
            capture_output=True)

    def list_secrets(self, namespace):
        out = subprocess.run(
            [self.kubectl, "-n", namespace, "get", "secrets", "-o", "json"],
            check=True, capture_output=True)
        items = json.loads(out.stdout).get("items", [])
        return [{"name": it["metadata"]["name"].replace("kt-secret-", "", 1),
                 "k8s_name": it["metadata"]["name"],
                 "keys": sorted((it.get("data") or {}))}
                for it in items
                if it["metadata"]["name"].startswith("kt-secret-")]

    def delete_secret(self, name, namespace):
        k8s_name = f"kt-secret-{name}".lower().replace("_", "-")
        subprocess.run(
            [self.kubectl, "-n", namespace, "delete", "secret", k8s_name,
             "--ignore-not-found"],
            check=True, capture_output=True)

    def get_events(self, name, namespace, since=0.0):
        """K8s events for the service's objects (scheduling, image pulls,
        probe failures) — what the client streams during launch."""
        import datetime

        out = subprocess.run(
            [self.kubectl, "-n", namespace, "get", "events",
             "--sort-by=.lastTimestamp", "-o", "json"],
            capture_output=True,
        )
        if out.returncode != 0:
            return []
        evs = []
        for it in json.loads(out.stdout).get("items", []):
            obj = it.get("involvedObject", {}).get("name", "")
            if not obj.startswith(name):
                continue
            raw_ts = (it.get("lastTimestamp") or it.get("eventTime") or "")
            try:
                ts = datetime.datetime.fromisoformat(
                    raw_ts.replace("Z", "+00:00")).timestamp()
            except ValueError:
                ts = 0.0
            if ts <= since:
                continue
            evs.append({"ts": ts, "type": it.get("type", "Normal"),
                        "reason": it.get("reason", ""),
                        "message": it.get("message", ""), "pod": obj})
        return evs
