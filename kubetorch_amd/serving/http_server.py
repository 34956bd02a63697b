"""Worker pod HTTP server (FastAPI).

Endpoints (reference parity: serving/http_server.py):
    GET  /health                     liveness
    GET  /ready?launch_id=X          503 until the reload for X completed
    GET  /metrics                    prometheus exposition
    GET  /logs/tail?since=&request_id=   ring-buffer tail (JSON)
    WS   /logs/ws                    live log stream
    POST /call/{name}[/{method}]     run the configured callable
    POST /spmd/subcall               peer-to-peer distributed fan-in
    POST /reload                     hot reload (also driven by controller WS)
    GET  /app/status                 app-mode process status

The server process never runs user code: calls route through the supervisor
into spawn-method subprocesses (process_pool.py). Distributed calls fan out
per the supervisor (supervisors.py).
"""
import asyncio
import base64
import json
import os
import pickle
import threading
import time
import uuid
from contextlib import asynccontextmanager

from fastapi import FastAPI, Request, Response, WebSocket
from fastapi.responses import JSONResponse

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import package_exception
from kubetorch_amd.serving import log_capture, metrics
from kubetorch_amd.serving.supervisors import supervisor_factory

STATE = {
    "supervisor": None,
    "lock": threading.Lock(),
    "launch_id": None,
    "started_at": None,
    "terminating": False,
    "app_proc": None,
    "load_error": None,
}


def _distributed_config():
    raw = os.environ.get(C.ENV_DISTRIBUTED_CONFIG)
    if not raw:
        return {}
    try:
        return json.loads(raw)
    except json.JSONDecodeError:
        return {}


def get_supervisor(recreate=False):
    with STATE["lock"]:
        if STATE["supervisor"] is not None and not recreate:
            return STATE["supervisor"]
        if STATE["supervisor"] is not None:
            try:
                STATE["supervisor"].cleanup()
            except Exception:
                pass
            STATE["supervisor"] = None
        cfg = _distributed_config()
        dtype = cfg.get("type")
        kw = {}
        if dtype not in (None, "", "local"):
            kw = {
                "num_workers": cfg.get("workers"),
                "num_proc": cfg.get("num_proc"),
                "quorum_timeout": cfg.get("quorum_timeout", C.QUORUM_TIMEOUT),
            }
        else:
            kw = {"num_proc": cfg.get("num_proc") or 1}
        try:
            STATE["supervisor"] = supervisor_factory(dtype, **kw)
            STATE["load_error"] = None
        except Exception as e:
            STATE["load_error"] = package_exception(e)
            raise
        return STATE["supervisor"]


def apply_metadata(md: dict):
    """Apply module metadata pushed by the controller (env-var contract)."""
    mapping = {
        "module_name": C.ENV_MODULE_NAME,
        "callable_name": C.ENV_CALLABLE_NAME,
        "file_path": C.ENV_FILE_PATH,
        "project_root": C.ENV_PROJECT_ROOT,
        "init_args": C.ENV_INIT_ARGS,
        "module_type": C.ENV_MODULE_TYPE,
        "service_name": C.ENV_SERVICE_NAME,
        "service_dns": C.ENV_SERVICE_DNS,
        "workdir_key": "KT_WORKDIR_KEY",
        "rel_path": "KT_REL_PATH",
        "allowed_serialization": "KT_ALLOWED_SERIALIZATION",
        "exec_token": "KT_EXEC_TOKEN",
    }
    for key, env in mapping.items():
        if key in md and md[key] is not None:
            os.environ[env] = str(md[key])
    if md.get("distributed_config") is not None:
        os.environ[C.ENV_DISTRIBUTED_CONFIG] = json.dumps(md["distributed_config"])


_SUP_SIG_KEYS = (C.ENV_DISTRIBUTED_CONFIG, C.ENV_MODULE_NAME,
                 C.ENV_CALLABLE_NAME, C.ENV_FILE_PATH, C.ENV_PROJECT_ROOT,
                 C.ENV_INIT_ARGS, C.ENV_MODULE_TYPE, "KT_WORKDIR_KEY",
                 "KT_REL_PATH")


def _sup_sig():
    return tuple(os.environ.get(k) for k in _SUP_SIG_KEYS)


def do_reload(md: dict, launch_id=None):
    """Hot reload: apply metadata, re-sync code, recreate supervisor.
    The launch_id is only set after success so /ready gates correctly.

    A genuine reload (launch_id set) always recreates; a plain metadata
    push recreates only if the supervisor-relevant config actually changed
    (reference parity: hash of distributed config decides recreate,
    http_server.py:878-1137) — otherwise the registration-time metadata
    echo would terminate the pool under an in-flight first call."""
    before = _sup_sig()
    apply_metadata(md or {})
    rebuild = bool(launch_id) or STATE["supervisor"] is None \
        or _sup_sig() != before
    if rebuild:
        try:
            from kubetorch_amd.data_store import commands as ds

            ds.sync_workdir_from_store()
        except Exception:
            pass
    if md and md.get("image_setup"):
        from kubetorch_amd.serving import image_setup

        image_setup.cached_image_setup(md["image_setup"], app_state=STATE)
    if rebuild:
        from kubetorch_amd.serving import loading

        loading.clear_cache()
        get_supervisor(recreate=True)
    if launch_id:
        STATE["launch_id"] = str(launch_id)
        os.environ[C.ENV_LAUNCH_ID] = str(launch_id)


# -- controller push channel (pod side) ---------------------------------------
async def _controller_ws_loop(url):
    """Register with the controller hub and stream metadata/reload pushes.
    Implemented as a long-lived chunked-HTTP stream (NDJSON) rather than a
    raw WebSocket: same push semantics + ack barrier, reconnect w/ backoff.
    (Reference parity: http_server.py:206-497 ControllerWebSocket.)"""
    import httpx

    backoff = 1.0
    pod = {
        "pod_name": os.environ.get("POD_NAME", os.uname().nodename),
        "pod_ip": os.environ.get("POD_IP", "127.0.0.1"),
        "namespace": os.environ.get("POD_NAMESPACE", "default"),
        "service_name": os.environ.get(C.ENV_SERVICE_NAME, ""),
        "port": int(os.environ.get("KT_SERVER_PORT", C.SERVER_PORT)),
        "request_metadata": True,
    }
    while not STATE["terminating"]:
        try:
            async with httpx.AsyncClient(timeout=None) as client:
                async with client.stream("POST", url, json=pod) as resp:
                    backoff = 1.0
                    async for line in resp.aiter_lines():
                        if not line.strip():
                            continue
                        msg = json.loads(line)
                        action = msg.get("action")
                        if action in ("metadata", "reload"):
                            lid = msg.get("launch_id")
                            await asyncio.to_thread(
                                do_reload, msg.get("metadata", {}),
                                lid if action == "reload" else None,
                            )
                            if action == "reload":
                                ack_url = url.rsplit("/", 1)[0] + "/reload_ack"
                                await client.post(ack_url, json={
                                    "pod_name": pod["pod_name"],
                                    "launch_id": lid,
                                })
        except Exception:
            await asyncio.sleep(backoff)
            backoff = min(backoff * 2, 30.0)


@asynccontextmanager
async def lifespan(app: FastAPI):
    log_capture.install()
    STATE["started_at"] = time.time()
    from kubetorch_amd.serving import gpu_metrics

    gpu_metrics.register()
    pusher = metrics.MetricsPusher()
    pusher.start()
    stop_ev = threading.Event()
    ws_task = None
    controller_url = os.environ.get("KT_CONTROLLER_URL")
    if controller_url:
        ws_task = asyncio.create_task(
            _controller_ws_loop(controller_url.rstrip("/") + "/controller/pods/stream")
        )
    # initial launch_id from env (set by the launcher/local driver)
    STATE["launch_id"] = os.environ.get(C.ENV_LAUNCH_ID)
    # eagerly build the supervisor if a callable is configured; startup
    # errors are surfaced at call time, not crash time (reference behavior)
    if os.environ.get(C.ENV_FILE_PATH):
        try:
            get_supervisor()
        except Exception:
            pass

    def _drain_loop():
        # follow the CURRENT supervisor's log queue (reloads swap the pool)
        while not stop_ev.is_set():
            sup = STATE.get("supervisor")
            q = sup.pool.log_q if sup is not None else None
            if q is None:
                stop_ev.wait(0.5)
                continue
            try:
                item = q.get(timeout=0.5)
            except Exception:
                continue
            if item:
                log_capture.RING.append(
                    item.get("line", ""), source=item.get("source", "worker"),
                    request_id=item.get("request_id"))

    threading.Thread(target=_drain_loop, daemon=True).start()
    yield
    STATE["terminating"] = True
    stop_ev.set()
    if ws_task:
        ws_task.cancel()
    pusher.stop()
    if STATE["supervisor"] is not None:
        STATE["supervisor"].cleanup()


app = FastAPI(lifespan=lifespan)


@app.middleware("http")
async def metrics_middleware(request: Request, call_next):
    t0 = time.time()
    path = request.url.path.split("/")[1] or "root"
    # the in-flight gauge counts USER calls only — metrics/health polls
    # must not feed the autoscaler's concurrency signal
    is_call = path in ("call", "spmd")
    if is_call:
        metrics.ACTIVE_REQUESTS.inc()
    try:
        resp = await call_next(request)
        metrics.HTTP_REQUESTS.labels(path=path, status=resp.status_code).inc()
        return resp
    finally:
        if is_call:
            metrics.ACTIVE_REQUESTS.dec()
            metrics.touch_activity()
        metrics.HTTP_DURATION.labels(path=path).observe(time.time() - t0)


@app.get("/health")
def health():
    return {"status": "ok", "uptime": time.time() - (STATE["started_at"] or time.time())}


@app.get("/ready")
def ready(launch_id: str = None):
    if STATE["terminating"]:
        return JSONResponse({"ready": False, "reason": "terminating"}, status_code=503)
    if launch_id and STATE["launch_id"] != launch_id:
        return JSONResponse(
            {"ready": False, "reason": f"launch {STATE['launch_id']} != {launch_id}"},
            status_code=503,
        )
    return {"ready": True, "launch_id": STATE["launch_id"]}


@app.get("/metrics")
def metrics_route():
    return Response(metrics.exposition(), media_type="text/plain; version=0.0.4")


@app.get("/logs/tail")
def logs_tail(since: int = 0, request_id: str = None, limit: int = 1000):
    return {"entries": log_capture.RING.tail(since, request_id, limit),
            "seq": log_capture.RING.seq}


@app.websocket("/logs/ws")
async def logs_ws(ws: WebSocket):
    await ws.accept()
    params = ws.query_params
    since = int(params.get("since", 0))
    request_id = params.get("request_id")
    try:
        while True:
            entries = log_capture.RING.tail(since, request_id)
            for e in entries:
                await ws.send_json(e)
                since = e["seq"] + 1
            got = await asyncio.to_thread(log_capture.RING.wait_for, since, 5.0)
            if not got:
                await ws.send_json({"keepalive": True})
    except Exception:
        pass


def _validate_name(name):
    configured = os.environ.get(C.ENV_CALLABLE_NAME) or os.environ.get(C.ENV_MODULE_NAME)
    return configured is None or name in (configured, os.environ.get(C.ENV_MODULE_NAME))


class SerializationNotAllowed(Exception):
    pass


async def _parse_call(request: Request):
    ser = request.headers.get("X-Serialization", "json")
    # default-deny pickle: deserializing client bytes is code execution, so
    # the format must be on the deploy-time allowlist (reference parity:
    # KT_ALLOWED_SERIALIZATION, default "json")
    allowed = os.environ.get("KT_ALLOWED_SERIALIZATION", "json").split(",")
    if ser not in allowed:
        raise SerializationNotAllowed(
            f"serialization {ser!r} not allowed (allowed: {allowed}); deploy "
            f"with Compute(allowed_serialization=[...]) to permit it")
    body = await request.body()
    if ser == "pickle":
        payload = json.loads(body)
        return payload["body"], ser
    payload = json.loads(body) if body else {}
    args = payload.get("args", [])
    kwargs = payload.get("kwargs", {})
    return base64.b64encode(pickle.dumps((tuple(args), kwargs))).decode(), ser


def _serialize_result(result, ser):
    if ser == "pickle":
        return {"result": base64.b64encode(pickle.dumps(result)).decode()}
    try:
        json.dumps(result)
        return {"result": result}
    except (TypeError, ValueError):
        return {"result_pickle": base64.b64encode(pickle.dumps(result)).decode()}


@app.post("/call/{name}")
@app.post("/call/{name}/{method}")
async def run_callable(name: str, request: Request, method: str = None,
                       workers: str = None, restart_procs: bool = False):
    if not _validate_name(name):
        return JSONResponse(
            {"error": {"error_type": "KeyError",
                       "message": f"callable {name!r} not deployed here "
                                  f"(configured: {os.environ.get(C.ENV_CALLABLE_NAME)})",
                       "traceback": ""}},
            status_code=404,
        )
    if STATE["load_error"]:
        return JSONResponse({"error": STATE["load_error"]}, status_code=500)
    rid = request.headers.get("X-Request-ID") or uuid.uuid4().hex
    token = log_capture.request_id_var.set(rid)
    try:
        sup = get_supervisor()
        try:
            body_b64, ser = await _parse_call(request)
        except SerializationNotAllowed as e:
            return JSONResponse({"error": package_exception(e)}, status_code=400)
        sel = workers
        if workers and workers not in ("all", "any", "ready"):
            sel = json.loads(workers)
        result = await asyncio.to_thread(
            sup.call, serialized_body=body_b64, method=method,
            workers=sel or "all", restart_procs=restart_procs,
        ) if sup.distributed else await asyncio.to_thread(
            _call_simple, sup, body_b64, method, rid,
        )
        return _serialize_result(result, ser)
    except BaseException as e:  # noqa: BLE001
        return JSONResponse({"error": package_exception(e)}, status_code=500)
    finally:
        log_capture.request_id_var.reset(token)


def _call_simple(sup, body_b64, method, rid):
    from kubetorch_amd.serving.supervisors import _decode_resp

    resp = sup.pool.submit(0, body_b64, method=method, request_id=rid).result(
        C.HTTP_TIMEOUT * 10
    )
    return _decode_resp(resp)


@app.post("/spmd/subcall")
async def spmd_subcall(request: Request):
    payload = await request.json()
    try:
        sup = get_supervisor()
        if "hosts" in payload and payload["hosts"]:
            os.environ["KT_SPMD_HOSTS"] = json.dumps(payload["hosts"])
        result = await asyncio.to_thread(
            sup.call, serialized_body=payload["body"],
            method=payload.get("method"), distributed_subcall=True,
            subtree=payload.get("subtree"),
        )
        return {"result": base64.b64encode(pickle.dumps(result)).decode()}
    except BaseException as e:  # noqa: BLE001
        return JSONResponse({"error": package_exception(e)}, status_code=500)


@app.post("/reload")
async def reload_route(request: Request):
    payload = await request.json()
    try:
        await asyncio.to_thread(do_reload, payload.get("metadata", {}),
                                payload.get("launch_id"))
        return {"ok": True, "launch_id": STATE["launch_id"]}
    except BaseException as e:  # noqa: BLE001
        return JSONResponse({"error": package_exception(e)}, status_code=500)


@app.post("/exec")
async def exec_route(request: Request):
    """Run a bash command inside the pod (post-launch pip_install/run_bash
    helpers; reference: compute.py run_bash/pip_install). Gated by the
    per-deploy shared secret KT_EXEC_TOKEN: without it (or on mismatch) the
    route is disabled — an unauthenticated /exec would hand command execution
    to any peer that can reach the serving port."""
    import hmac
    import subprocess

    token = os.environ.get("KT_EXEC_TOKEN", "")
    presented = request.headers.get("X-KT-Exec-Token", "")
    if not token or not hmac.compare_digest(token, presented):
        return JSONResponse(
            {"error": {"error_type": "PermissionError",
                       "message": "exec disabled: missing or invalid "
                                  "X-KT-Exec-Token",
                       "traceback": ""}},
            status_code=403,
        )
    payload = await request.json()
    cmd = payload.get("command")
    if not cmd:
        return JSONResponse({"error": {"error_type": "ValueError",
                                       "message": "missing 'command'",
                                       "traceback": ""}}, status_code=400)
    try:
        res = await asyncio.to_thread(
            subprocess.run, ["bash", "-lc", cmd],
            capture_output=True, text=True,
            timeout=payload.get("timeout", 600),
        )
    except subprocess.TimeoutExpired as e:
        return JSONResponse({"error": package_exception(e)}, status_code=500)
    return {"returncode": res.returncode, "stdout": res.stdout[-20000:],
            "stderr": res.stderr[-20000:]}


@app.get("/localfiles/{key:path}")
def serve_local_file(key: str):
    """Serve a key this pod registered with locale='local' (zero-copy
    put: the data never moved to the store; peers fetch it from here).
    Only explicitly registered keys are served — this is not a general
    file server."""
    import io as _io
    import tarfile as _tarfile

    from kubetorch_amd.data_store.commands import localfs_registry_path

    reg_path = localfs_registry_path()
    reg = {}
    if os.path.exists(reg_path):
        try:
            with open(reg_path) as f:
                reg = json.load(f)
        except (OSError, ValueError):
            reg = {}
    path = reg.get(key.strip("/"))
    if not path or not os.path.exists(path):
        return JSONResponse({"error": f"key {key!r} not registered here"},
                            status_code=404)
    if os.path.isdir(path):
        buf = _io.BytesIO()
        with _tarfile.open(fileobj=buf, mode="w:gz") as tar:
            tar.add(path, arcname=".")
        return Response(buf.getvalue(), media_type="application/gzip",
                        headers={"X-KT-Tar": "1"})
    with open(path, "rb") as f:
        return Response(f.read(), media_type="application/octet-stream")


@app.get("/app/status")
def app_status():
    proc = STATE.get("app_proc")
    if proc is None:
        return {"running": False}
    rc = proc.poll()
    return {"running": rc is None, "returncode": rc}


def _install_sigterm_drain():
    """SIGTERM: mark terminating (readiness flips 503) and let in-flight
    requests drain before exiting (reference: TerminationCheckMiddleware,
    http_server.py:1184-1236).

    Once uvicorn's event loop is up it installs its own SIGTERM handler,
    whose graceful shutdown (stop accepting -> wait for in-flight
    responses -> run lifespan teardown, which cleans up the supervisor)
    gives the same drain semantics — validated by the kill-mid-call e2e
    drill. This handler covers the window before uvicorn starts and
    non-uvicorn embeddings of the app."""
    import signal

    def _drain_then_exit():
        # runs OFF the main thread: the asyncio loop must stay free to
        # deliver the in-flight responses we are waiting for
        drain_s = float(os.environ.get("KT_TERM_DRAIN_S", "25"))
        deadline = time.time() + drain_s
        while time.time() < deadline and \
                metrics.ACTIVE_REQUESTS._value.get() > 0:
            time.sleep(0.2)
        if STATE["supervisor"] is not None:
            try:
                STATE["supervisor"].cleanup()
            except Exception:
                pass
        os._exit(0)

    def handler(signum, frame):
        # flip readiness FIRST (new traffic stops routing here), then let
        # in-flight calls finish before tearing the pool down — a worker
        # killed mid-request fails the caller for no reason
        STATE["terminating"] = True
        threading.Thread(target=_drain_then_exit, daemon=True).start()

    try:
        signal.signal(signal.SIGTERM, handler)
    except ValueError:
        pass  # not the main thread (in-process test server)


def main():
    import argparse

    import uvicorn

    _install_sigterm_drain()

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("KT_SERVER_PORT", C.SERVER_PORT)))
    ap.add_argument("--host", default="0.0.0.0")
    args = ap.parse_args()
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
