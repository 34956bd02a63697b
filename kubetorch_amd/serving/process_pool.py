"""Subprocess isolation: ProcessPool of ProcessWorkers.

User code runs in spawn-method subprocesses, never in the server process, so
a reload can kill and respawn workers without touching the HTTP server, and
each SPMD local rank owns a process (and on GPU nodes, a GPU).
(Reference parity: serving/process_pool.py, serving/process_worker.py.)
"""
import base64
import multiprocessing as mp
import os
import pickle
import queue as pyqueue
import threading
import traceback
import uuid
from concurrent.futures import Future, ThreadPoolExecutor

from kubetorch_amd.exceptions import package_exception

_WORKER_THREADS = 16


def _worker_main(idx, req_q, resp_q, base_env, eager_load, log_q=None):
    """Worker subprocess entrypoint: apply env, redirect stdout/stderr to the
    log queue, optionally eager-load the callable, then serve requests on a
    thread pool."""
    os.environ.update({k: str(v) for k, v in base_env.items()})
    os.environ.setdefault(
        "PYTHONBREAKPOINT", "kubetorch_amd.serving.pdb_ws.deep_breakpoint")
    import contextvars
    import io
    import sys

    rid_var = contextvars.ContextVar("rid", default=None)

    if log_q is not None:
        class _QW(io.TextIOBase):
            def __init__(self, source, orig):
                self.source = source
                self.orig = orig
                self._partial = ""

            def write(self, s):
                self._partial += s
                while "\n" in self._partial:
                    line, self._partial = self._partial.split("\n", 1)
                    if line.strip():
                        try:
                            log_q.put({"line": line, "source": f"worker{idx}",
                                       "request_id": rid_var.get()})
                        except Exception:
                            self.orig.write(line + "\n")
                return len(s)

            def flush(self):
                pass

        sys.stdout = _QW("stdout", sys.stdout)
        sys.stderr = _QW("stderr", sys.stderr)

    from kubetorch_amd.serving import loading

    if eager_load:
        try:
            loading.load_from_env()
        except Exception:
            traceback.print_exc()

    pool = ThreadPoolExecutor(max_workers=_WORKER_THREADS)

    def handle(req):
        rid = req["rid"]
        rid_var.set(req.get("request_id"))
        try:
            env = req.get("env") or {}
            os.environ.update({k: str(v) for k, v in env.items()})
            callable_obj = loading.load_from_env(fresh=req.get("fresh", False))
            method = req.get("method")
            target = getattr(callable_obj, method) if method else callable_obj
            args, kwargs = pickle.loads(base64.b64decode(req["body"]))
            result = target(*args, **kwargs)
            import inspect

            if inspect.iscoroutine(result):
                import asyncio

                result = asyncio.run(result)
            resp_q.put({"rid": rid, "ok": True,
                        "result": base64.b64encode(pickle.dumps(result)).decode()})
        except BaseException as e:  # noqa: BLE001 - ship everything back
            resp_q.put({"rid": rid, "ok": False, "error": package_exception(e)})

    while True:
        req = req_q.get()
        if req is None:
            break
        if req.get("cmd") == "ping":
            resp_q.put({"rid": req["rid"], "ok": True,
                        "result": base64.b64encode(pickle.dumps("pong")).decode()})
            continue
        pool.submit(handle, req)
    pool.shutdown(wait=False, cancel_futures=True)


class ProcessWorker:
    def __init__(self, idx, base_env=None, eager_load=True, log_q=None):
        self.idx = idx
        ctx = mp.get_context("spawn")
        self.req_q = ctx.Queue()
        self.resp_q = ctx.Queue()
        self.proc = ctx.Process(
            target=_worker_main,
            args=(idx, self.req_q, self.resp_q, base_env or {}, eager_load,
                  log_q),
            daemon=True,
        )
        self.proc.start()

    def alive(self):
        return self.proc.is_alive()

    def terminate(self):
        try:
            self.req_q.put(None)
        except Exception:
            pass
        self.proc.terminate()
        self.proc.join(5)
        if self.proc.is_alive():
            self.proc.kill()
            self.proc.join(5)


class ProcessPool:
    """N ProcessWorkers + a response-router thread matching request ids.
    (Reference parity: serving/process_pool.py:12,125,178.)"""

    def __init__(self, num_proc=1, base_env_fn=None, eager_load=True,
                 capture_logs=True):
        self.num_proc = num_proc
        self._base_env_fn = base_env_fn or (lambda idx: {})
        self._eager_load = eager_load
        self.log_q = mp.get_context("spawn").Queue() if capture_logs else None
        self.workers = []
        self._futures = {}
        self._lock = threading.Lock()
        self._routers = []
        self._stopped = False
        for i in range(num_proc):
            self._start_worker(i)

    def _start_worker(self, idx):
        w = ProcessWorker(idx, base_env=self._base_env_fn(idx),
                          eager_load=self._eager_load, log_q=self.log_q)
        if idx < len(self.workers):
            self.workers[idx] = w
        else:
            self.workers.append(w)
        t = threading.Thread(target=self._route, args=(w,), daemon=True)
        t.start()
        self._routers.append(t)

    def _route(self, worker):
        while not self._stopped:
            try:
                resp = worker.resp_q.get(timeout=1.0)
            except (pyqueue.Empty, EOFError, OSError):
                if not worker.alive():
                    # a crashed worker (segfault, OOM-kill) must FAIL its
                    # in-flight requests now, not let callers hang until
                    # the HTTP timeout. Only if THIS worker is still the
                    # pool's current worker for its slot — after a
                    # deliberate restart() the old router must not fail
                    # futures that belong to the replacement worker.
                    with self._lock:
                        current = (self.workers[worker.idx]
                                   if worker.idx < len(self.workers) else None)
                    if current is worker:
                        self._fail_worker_futures(
                            worker.idx,
                            f"worker {worker.idx} died "
                            f"(exitcode {worker.proc.exitcode})")
                    return
                continue
            with self._lock:
                entry = self._futures.pop(resp["rid"], None)
            fut = entry[0] if entry else None
            if fut is not None and not fut.done():
                fut.set_result(resp)

    def _fail_worker_futures(self, idx, msg):
        with self._lock:
            dead = [rid for rid, (f, widx) in self._futures.items()
                    if widx == idx]
            entries = [self._futures.pop(rid) for rid in dead]
        for fut, _w in entries:
            if not fut.done():
                fut.set_result(
                    {"ok": False,
                     "error": {"error_type": "PodTerminatedError",
                               "message": msg, "traceback": ""}})

    def submit(self, idx, body_b64, method=None, env=None, fresh=False,
               request_id=None):
        # self-heal: a worker that died while idle is respawned before the
        # request is enqueued (a dead worker's queue accepts puts but
        # nothing drains them)
        if not self.workers[idx].alive():
            with self._lock:
                needs = not self.workers[idx].alive()
            if needs:
                self._start_worker(idx)
        rid = uuid.uuid4().hex
        fut = Future()
        with self._lock:
            self._futures[rid] = (fut, idx)
        self.workers[idx].req_q.put(
            {"rid": rid, "body": body_b64, "method": method, "env": env,
             "fresh": fresh, "request_id": request_id}
        )
        return fut

    def call(self, idx, args=(), kwargs=None, method=None, env=None,
             timeout=None):
        body = base64.b64encode(pickle.dumps((args, kwargs or {}))).decode()
        resp = self.submit(idx, body, method=method, env=env).result(timeout)
        return resp

    def call_all(self, bodies_envs, method=None, timeout=None):
        """bodies_envs: list of (body_b64, env) per worker; returns futures."""
        futs = []
        for idx, (body, env) in enumerate(bodies_envs):
            futs.append(self.submit(idx, body, method=method, env=env))
        return futs

    def restart(self):
        self.terminate()
        self._stopped = False
        self._routers = []
        old = self.workers
        self.workers = []
        for i in range(self.num_proc):
            self._start_worker(i)

    def terminate(self):
        self._stopped = True
        for w in self.workers:
            w.terminate()
        with self._lock:
            for fut, _idx in self._futures.values():
                if not fut.done():
                    fut.set_result(
                        {"ok": False,
                         "error": {"error_type": "KubetorchError",
                                   "message": "worker pool terminated",
                                   "traceback": ""}}
                    )
            self._futures.clear()
