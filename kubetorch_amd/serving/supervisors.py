"""Execution & distribution supervisors.

Class tree (reference parity: serving/design.md:12-41, spmd_supervisor.py):
    ExecutionSupervisor                  -- subprocess pool, call -> worker 0
      DistributedSupervisor              -- + peer discovery, quorum, monitor
        SPMDSupervisor                   -- + multi-pod fan-out, per-rank env
Process-framework env contracts: TorchProcess (RCCL over xGMI), JaxProcess,
TensorflowProcess, generic SPMD.
"""
import base64
import json
import os
import pickle
import socket
from concurrent.futures import FIRST_EXCEPTION, ThreadPoolExecutor, wait

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import reconstruct_exception
from kubetorch_amd.serving import discovery
from kubetorch_amd.serving.process_pool import ProcessPool

TREE_THRESHOLD = int(os.environ.get("KT_TREE_THRESHOLD", "100"))
TREE_FANOUT = int(os.environ.get("KT_TREE_FANOUT", "50"))


# ---------------------------------------------------------------------------
# per-framework distributed env contracts
# ---------------------------------------------------------------------------
class SPMDProcess:
    """Generic rank-aware env (any SPMD program)."""

    @staticmethod
    def auto_num_proc():
        return 1

    @classmethod
    def env_vars(cls, worker_hosts, node_rank, local_rank, num_proc):
        world = len(worker_hosts) * num_proc
        return {
            "WORLD_SIZE": str(world),
            "RANK": str(node_rank * num_proc + local_rank),
            "LOCAL_RANK": str(local_rank),
            "NODE_RANK": str(node_rank),
            "LOCAL_WORLD_SIZE": str(num_proc),
            "POD_IPS": ",".join(h.split(":")[0] for h in worker_hosts),
            "KT_NUM_WORKERS": str(len(worker_hosts)),
        }

    @staticmethod
    def framework_cleanup():
        pass


class TorchProcess(SPMDProcess):
    """torch.distributed contract: one rank per GPU, backend "nccl" (= RCCL
    on ROCm) over xGMI intra-node. (Reference: spmd/pytorch_process.py.)"""

    @staticmethod
    def auto_num_proc():
        try:
            import torch

            n = torch.cuda.device_count()
            return n if n > 0 else 1
        except Exception:
            return 1

    @classmethod
    def env_vars(cls, worker_hosts, node_rank, local_rank, num_proc):
        env = super().env_vars(worker_hosts, node_rank, local_rank, num_proc)
        env["MASTER_ADDR"] = worker_hosts[0].split(":")[0]
        env["MASTER_PORT"] = os.environ.get("KT_MASTER_PORT",
                                            str(C.DEFAULT_MASTER_PORT))
        env.update(C.RCCL_ENV_DEFAULTS)
        return env

    @staticmethod
    def framework_cleanup():
        try:
            import torch.distributed as dist

            if dist.is_initialized():
                dist.destroy_process_group()
        except Exception:
            pass


class JaxProcess(SPMDProcess):
    @classmethod
    def env_vars(cls, worker_hosts, node_rank, local_rank, num_proc):
        env = super().env_vars(worker_hosts, node_rank, local_rank, num_proc)
        env["JAX_COORDINATOR_ADDRESS"] = worker_hosts[0].split(":")[0] + ":1234"
        env["JAX_PROCESS_ID"] = env["RANK"]
        env["JAX_NUM_PROCESSES"] = env["WORLD_SIZE"]
        env["JAX_LOCAL_DEVICE_IDS"] = str(local_rank)
        return env


class TensorflowProcess(SPMDProcess):
    @classmethod
    def env_vars(cls, worker_hosts, node_rank, local_rank, num_proc):
        env = super().env_vars(worker_hosts, node_rank, local_rank, num_proc)
        workers = [h.split(":")[0] + ":2222" for h in worker_hosts]
        env["TF_CONFIG"] = json.dumps(
            {"cluster": {"worker": workers},
             "task": {"type": "worker", "index": node_rank * num_proc + local_rank}}
        )
        return env


PROCESS_CLASSES = {
    "pytorch": TorchProcess,
    "torch": TorchProcess,
    "jax": JaxProcess,
    "tensorflow": TensorflowProcess,
    "spmd": SPMDProcess,
}


def _encode_call(args, kwargs):
    return base64.b64encode(pickle.dumps((args, kwargs or {}))).decode()


def _decode_resp(resp):
    if resp.get("ok"):
        return pickle.loads(base64.b64decode(resp["result"]))
    raise reconstruct_exception(resp["error"])


# ---------------------------------------------------------------------------
class ExecutionSupervisor:
    """Base supervisor: a subprocess pool; calls route to worker 0."""

    distributed = False

    def __init__(self, num_proc=1, eager_load=True):
        self.num_proc = num_proc
        self.pool = ProcessPool(num_proc, base_env_fn=self._base_env,
                                eager_load=eager_load)

    def _base_env(self, idx):
        return {}

    def call(self, args=(), kwargs=None, method=None, timeout=None,
             serialized_body=None, **_):
        if serialized_body is not None:
            # distributed-flagged supervisors without SPMD fan-out (Ray,
            # Monarch) receive the pre-serialized body from run_callable
            resp = self.pool.submit(0, serialized_body, method=method).result(
                timeout or C.HTTP_TIMEOUT * 10)
            return _decode_resp(resp)
        resp = self.pool.call(0, args, kwargs, method=method, timeout=timeout)
        return _decode_resp(resp)

    def healthy(self):
        return all(w.alive() for w in self.pool.workers)

    def cleanup(self):
        self.pool.terminate()

    def restart(self):
        self.pool.restart()


class DistributedSupervisor(ExecutionSupervisor):
    """+ peer discovery/quorum over DNS or KT_LOCAL_IPS and a background
    membership monitor that aborts in-flight calls on pod death."""

    distributed = True

    def __init__(self, num_proc=1, num_workers=None, service_name=None,
                 namespace=None, eager_load=True, quorum_timeout=C.QUORUM_TIMEOUT):
        self.num_workers = num_workers or int(os.environ.get("KT_NUM_WORKERS", "1"))
        self.service_name = service_name or os.environ.get(C.ENV_SERVICE_NAME)
        self.namespace = namespace or os.environ.get("POD_NAMESPACE", "default")
        self.quorum_timeout = quorum_timeout
        self.monitor = discovery.MembershipMonitor(self.service_name, self.namespace)
        super().__init__(num_proc=num_proc, eager_load=eager_load)

    def worker_hosts(self):
        return discovery.pod_ips(
            num_workers=self.num_workers, timeout=self.quorum_timeout,
            service_name=self.service_name, namespace=self.namespace,
        )

    def cleanup(self):
        self.monitor.stop()
        super().cleanup()


def _self_host():
    explicit = os.environ.get("KT_SELF_HOST")
    if explicit:
        return explicit
    try:
        ip = socket.gethostbyname(socket.gethostname())
    except socket.gaierror:
        ip = "127.0.0.1"
    return f"{ip}:{os.environ.get('KT_SERVER_PORT', C.SERVER_PORT)}"


class SPMDSupervisor(DistributedSupervisor):
    """The distributed launcher. The coordinator pod waits for quorum, sorts
    hosts with itself at rank 0, fans the call out to every pod (flat, or
    tree at >=100 workers) and every local rank, and aggregates the per-rank
    results into one list. (Reference parity: spmd/spmd_supervisor.py.)"""

    def __init__(self, framework="pytorch", num_proc=None, **kw):
        self.process_cls = PROCESS_CLASSES.get(framework, SPMDProcess)
        if num_proc in (None, "auto"):
            num_proc = self.process_cls.auto_num_proc()
        super().__init__(num_proc=num_proc, **kw)

    def _ordered_hosts(self):
        hosts = self.worker_hosts()
        me = _self_host()
        if me in hosts:
            hosts.remove(me)
        # coordinator puts itself at rank 0 for a stable MASTER_ADDR
        return [me] + sorted(hosts)

    def _local_env(self, hosts, node_rank, local_rank):
        return self.process_cls.env_vars(hosts, node_rank, local_rank,
                                         self.num_proc)

    def call(self, args=(), kwargs=None, method=None, timeout=None,
             distributed_subcall=False, workers="all", restart_procs=False,
             serialized_body=None, **_):
        body = serialized_body or _encode_call(args, kwargs)
        if restart_procs:
            self.pool.restart()

        if distributed_subcall:
            # non-coordinator worker: run local ranks; in tree mode also
            # relay to our subtree (fanout bounded per node)
            hosts = json.loads(os.environ.get("KT_SPMD_HOSTS", "[]")) or \
                self._ordered_hosts()
            me = _self_host()
            node_rank = hosts.index(me) if me in hosts else 0
            subtree = _.get("subtree") if isinstance(_.get("subtree"), list) \
                else None
            results = self._run_local_ranks(body, method, hosts, node_rank,
                                            timeout)
            if subtree and len(subtree) > 1:
                from kubetorch_amd.serving.remote_pool import (
                    call_worker_subcall,
                )

                rest = subtree[1:]
                with ThreadPoolExecutor(max_workers=min(16, len(rest))) as ex:
                    futs = [
                        ex.submit(call_worker_subcall, h, body, method, hosts,
                                  None, timeout)
                        for h in rest
                    ]
                    for f in futs:
                        r = f.result()
                        results.extend(r if isinstance(r, list) else [r])
            return results

        hosts = self._ordered_hosts()
        if workers == "any":
            hosts = hosts[:1]
        elif workers == "ready":
            from kubetorch_amd.serving.remote_pool import get_client

            def _up(h):
                try:
                    return get_client().get(f"http://{h}/health",
                                            timeout=3).status_code == 200
                except Exception:
                    return False

            hosts = [hosts[0]] + [h for h in hosts[1:] if _up(h)]
        if isinstance(workers, (list, tuple)):
            sel = []
            for w in workers:
                sel.append(hosts[w] if isinstance(w, int) else w)
            # always keep coordinator so MASTER_ADDR stays valid
            hosts = [hosts[0]] + [h for h in sel if h != hosts[0]]

        if not self.monitor._thread or not self.monitor._thread.is_alive():
            self.monitor.start(hosts)
        else:
            # membership changes BETWEEN calls are the expected elastic path
            # (a respawned pod joins this call's rendezvous); only changes
            # during the call should abort — re-baseline to this call's set
            self.monitor.rebase(hosts)
        aborted = []
        self.monitor.subscribe(aborted.append)
        executor = ThreadPoolExecutor(max_workers=4)
        try:
            remote_hosts = hosts[1:]
            futs = []
            # remote fan-out on the asyncio engine (200-concurrency cap,
            # one loop thread regardless of world size); flat, or tree for
            # very large worlds
            from kubetorch_amd.serving.remote_pool import fanout

            eng = fanout()
            if len(hosts) >= TREE_THRESHOLD:
                children = remote_hosts[:TREE_FANOUT]
                subtrees = [remote_hosts[i::TREE_FANOUT] for i in range(TREE_FANOUT)]
                for child, subtree in zip(children, subtrees):
                    futs.append(eng.submit(child, body, method, hosts,
                                           [child] + subtree[1:], timeout))
            else:
                for h in remote_hosts:
                    futs.append(eng.submit(h, body, method, hosts, None,
                                           timeout))
            local_fut = executor.submit(
                self._run_local_ranks, body, method, hosts, 0, timeout)
            futs.append(local_fut)

            done, pending = wait(futs, timeout=timeout,
                                 return_when=FIRST_EXCEPTION)
            if aborted:
                for f in pending:
                    f.cancel()
                raise aborted[0]
            for f in done:
                exc = f.exception()
                if exc is not None:
                    # a failed peer connection usually means a dead pod:
                    # force a membership check before reporting (reference:
                    # spmd_supervisor.py:413-429)
                    changed = self.monitor.check_now()
                    if aborted:
                        raise aborted[0]
                    if changed is not None:
                        raise changed
                    raise exc
            results = []
            local_results = local_fut.result()
            remote_results = []
            for f in futs[:-1]:
                r = f.result()
                remote_results.extend(r if isinstance(r, list) else [r])
            # rank order: coordinator local ranks first, then remote
            results.extend(local_results)
            results.extend(remote_results)
            return results
        finally:
            self.monitor.unsubscribe(aborted.append)
            executor.shutdown(wait=False)

    def _run_local_ranks(self, body, method, hosts, node_rank, timeout):
        futs = []
        for lr in range(self.num_proc):
            env = self._local_env(hosts, node_rank, lr)
            env["KT_SPMD_HOSTS"] = json.dumps(hosts)
            futs.append(self.pool.submit(lr, body, method=method, env=env))
        out = []
        for f in futs:
            out.append(_decode_resp(f.result(timeout)))
        return out


class RaySupervisor(ExecutionSupervisor):
    """Head-only Ray supervisor: starts `ray start --head` (or the command
    in KUBERAY_GEN_RAY_START_CMD) as a side process; calls run in a single
    worker that connects to the local GCS. Membership is Ray's business.
    (Reference parity: serving/ray_supervisor.py.)"""

    distributed = True

    def __init__(self, num_proc=1, **kw):
        import shutil
        import subprocess

        self._ray_proc = None
        cmd = os.environ.get("KUBERAY_GEN_RAY_START_CMD")
        if cmd is None and shutil.which("ray"):
            cmd = "ray start --head --block --dashboard-host=0.0.0.0"
        if cmd:
            self._ray_proc = subprocess.Popen(["bash", "-lc", cmd])
            self._wait_gcs()
        super().__init__(num_proc=1)

    def _wait_gcs(self, port=None, timeout=120):
        """Poll the Ray GCS port until live (reference: ray_supervisor.py
        GCS liveness check)."""
        import socket
        import time

        if port is None:
            port = int(os.environ.get("KT_RAY_GCS_PORT", "6379"))

        deadline = time.time() + timeout
        while time.time() < deadline:
            if self._ray_proc.poll() is not None:
                raise RuntimeError("ray head exited during startup")
            try:
                s = socket.create_connection(("127.0.0.1", port), 1)
                s.close()
                return
            except OSError:
                time.sleep(0.5)
        raise RuntimeError(f"Ray GCS not live on :{port} after {timeout}s")

    def cleanup(self):
        if self._ray_proc is not None and self._ray_proc.poll() is None:
            self._ray_proc.terminate()
        super().cleanup()


class MonarchSupervisor(DistributedSupervisor):
    """Single-controller actor-mesh launcher (reference parity:
    serving/monarch_supervisor.py). Every pod runs a `process_allocator`
    service; the controller pod (rank 0) exposes the full allocator
    address list as KT_MONARCH_HOSTS so user code can build a
    RemoteAllocator over all pods' meshes. The monarch wheel has no ROCm
    build in this image yet, so the allocator binary is resolved from
    KT_MONARCH_ALLOCATOR / PATH and a clear error is raised when absent
    (user code still needs `import monarch` to drive meshes)."""

    ALLOCATOR_PORT = 26600

    def __init__(self, num_workers=None, num_proc=None, **kw):
        import shutil
        import subprocess

        binary = os.environ.get("KT_MONARCH_ALLOCATOR") or \
            shutil.which("process_allocator")
        if not binary:
            raise NotImplementedError(
                "monarch's process_allocator binary is not on this image "
                "(no ROCm monarch wheel yet); install torchmonarch or set "
                "KT_MONARCH_ALLOCATOR, or use "
                "distribute('pytorch'|'spmd'|'ray')")
        self.allocator_port = int(os.environ.get("KT_MONARCH_PORT",
                                                 self.ALLOCATOR_PORT))
        self._alloc_proc = subprocess.Popen(
            [binary, f"--port={self.allocator_port}",
             "--program=monarch_bootstrap"])
        self._wait_allocator()
        # single controller process drives the meshes (Ray-style)
        super().__init__(num_proc=1, num_workers=num_workers,
                         quorum_timeout=kw.get("quorum_timeout",
                                               C.QUORUM_TIMEOUT))

    def _wait_allocator(self, timeout=60):
        import socket
        import time

        deadline = time.time() + timeout
        while time.time() < deadline:
            if self._alloc_proc.poll() is not None:
                raise RuntimeError(
                    f"process_allocator exited rc={self._alloc_proc.returncode}")
            try:
                s = socket.create_connection(("127.0.0.1",
                                              self.allocator_port), 1)
                s.close()
                return
            except OSError:
                time.sleep(0.25)
        raise RuntimeError(
            f"process_allocator not live on :{self.allocator_port}")

    def call(self, args=(), kwargs=None, method=None, timeout=None,
             serialized_body=None, **_):
        body = serialized_body or _encode_call(args, kwargs)
        hosts = self.worker_hosts()
        allocators = ",".join(
            f"{h.split(':')[0]}:{self.allocator_port}" for h in sorted(hosts))
        env = {"KT_MONARCH_HOSTS": allocators,
               "MONARCH_ALLOCATOR_PORT": str(self.allocator_port)}
        resp = self.pool.submit(0, body, method=method, env=env).result(
            timeout or C.HTTP_TIMEOUT * 10)
        return _decode_resp(resp)

    def cleanup(self):
        if getattr(self, "_alloc_proc", None) is not None \
                and self._alloc_proc.poll() is None:
            self._alloc_proc.terminate()
            try:
                self._alloc_proc.wait(5)
            except Exception:
                self._alloc_proc.kill()
        super().cleanup()


def supervisor_factory(distribution_type=None, **kw):
    """"local"/None -> ExecutionSupervisor; "pytorch"/"jax"/"tensorflow"/
    "spmd" -> SPMDSupervisor. (Reference: serving/supervisor_factory.py.)"""
    if distribution_type in (None, "", "local"):
        kw.pop("num_workers", None)
        kw.pop("framework", None)
        kw.pop("quorum_timeout", None)
        return ExecutionSupervisor(num_proc=kw.pop("num_proc", 1) or 1,
                                   **{k: v for k, v in kw.items()
                                      if k in ("eager_load",)})
    if distribution_type == "ray":
        return RaySupervisor()
    if distribution_type == "monarch":
        return MonarchSupervisor(**{k: v for k, v in kw.items()
                                    if k in ("num_workers", "num_proc",
                                             "quorum_timeout")})
    return SPMDSupervisor(framework=distribution_type, **kw)
