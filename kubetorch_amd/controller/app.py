"""kubetorch_amd controller: workload registry + pod push hub + TTL reaper.

The reference ships this as a closed-source image
(ghcr.io/run-house/kubetorch-controller); this is a
from-scratch implementation. Endpoints:

    POST /controller/deploy           apply manifest + register + reload pods
    POST /controller/workload         register-only (BYO / selector-only)
    GET  /controller/workloads/{ns}
    GET|DELETE /controller/workload/{ns}/{name}
    POST /controller/pods/stream      pod registration -> NDJSON push stream
    POST /controller/pods/reload_ack  ack barrier for hot reload
    GET  /controller/config           cluster-wide client defaults
    GET  /controller/events/{ns}/{name}   pod lifecycle events (launch UX)
    GET  /controller/podlogs/{ns}/{name}  raw pod stdout (app log-follow)
    POST|GET|DELETE /controller/secrets/{ns}[/{name}]
    POST|GET|DELETE /controller/volumes/{ns}[/{name}]
    POST /controller/tunnel/open (+/up,/down)  external-client TCP bridge
    GET  /controller/debug/connections
    GET  /health

Background loops: TTL reaper (inactivity teardown), pod monitor (auto
re-provision to desired replicas), KPA autoscaler (knative-kind
workloads on the local driver).

Push semantics: one asyncio queue per connected pod; deploy broadcasts
{action: reload, metadata, launch_id} to the service's pods and waits for
every ack before returning (reference: provisioning/design.md:59-103).
Single-process by design — the hub state is in-memory (the reference's
controller has the same 1-worker constraint, design.md:372).
"""
import asyncio
import json
import os
import time
import uuid

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from kubetorch_amd import constants as C
from kubetorch_amd.controller.drivers import K8sDriver, LocalDriver


class Hub:
    def __init__(self):
        self.workloads = {}       # (ns, name) -> workload dict
        self.pod_queues = {}      # pod_name -> asyncio.Queue
        self.pod_info = {}        # pod_name -> registration dict
        self.acks = {}            # launch_id -> {"pending": set, "event": Event}
        mode = os.environ.get("KT_CONTROLLER_DRIVER", "auto")
        k8s = K8sDriver()
        if mode == "local" or (mode == "auto" and not k8s.available()):
            self.driver = LocalDriver()
            self.driver_name = "local"
        else:
            self.driver = k8s
            self.driver_name = "k8s"

    def set_url(self, url):
        if isinstance(self.driver, LocalDriver):
            self.driver.controller_url = url

    def pods_for_service(self, service, namespace):
        return [p for p, info in self.pod_info.items()
                if info.get("service_name") == service
                and info.get("namespace") == namespace]


HUB = Hub()


def _parse_ttl(raw):
    if not raw:
        return None
    raw = str(raw).strip().lower()
    mult = {"s": 1, "m": 60, "h": 3600, "d": 86400}.get(raw[-1])
    try:
        return int(float(raw[:-1]) * mult) if mult else int(float(raw))
    except ValueError:
        return None


async def _ttl_reaper(interval=None):
    """Tear down workloads whose pods report no activity for longer than
    their inactivity TTL (reference: controller TTL reaper fed by the
    kt_last_activity_timestamp metric)."""
    import httpx

    if interval is None:
        interval = float(os.environ.get("KT_TTL_REAPER_INTERVAL", "30"))

    grace = float(os.environ.get("KT_TTL_GRACE", "60"))
    while True:
        await asyncio.sleep(interval)
        for (ns, name), w in list(HUB.workloads.items()):
            ttl = _parse_ttl(
                w.get("manifest", {}).get("metadata", {})
                .get("annotations", {}).get(C.INACTIVITY_TTL_ANNOTATION))
            if not ttl:
                continue
            if time.time() - w.get("created", 0) < ttl + grace:
                continue  # launch grace: pods report no activity until the
                          # first call lands
            last = w.get("updated", 0)
            try:
                pods = HUB.driver.pods(name, ns)
                # activity is per-pod: the service is idle only when EVERY
                # pod is (one busy replica must keep the workload alive)
                async with httpx.AsyncClient(timeout=5) as client:
                    for pod in pods:
                        r = await client.get(f"http://{pod}/metrics")
                        for line in r.text.splitlines():
                            if line.startswith("kt_last_activity_timestamp"):
                                last = max(last, float(line.split()[-1]))
            except Exception:
                continue
            if time.time() - last > ttl:
                HUB.workloads.pop((ns, name), None)
                try:
                    await asyncio.to_thread(HUB.driver.delete, name, ns)
                except Exception:
                    pass


def _parse_duration(raw, default=0.0):
    if raw in (None, ""):
        return default
    raw = str(raw).strip().lower()
    mult = {"s": 1, "m": 60, "h": 3600}.get(raw[-1])
    try:
        return float(raw[:-1]) * mult if mult else float(raw)
    except ValueError:
        return default


def _autoscale_spec(manifest):
    """Extract the Knative KPA annotations from a knative-Service manifest
    (None for non-autoscaled workloads)."""
    ann = (manifest.get("spec", {}).get("template", {})
           .get("metadata", {}).get("annotations", {}))
    if "autoscaling.knative.dev/target" not in ann:
        return None
    g = lambda k, d: ann.get(f"autoscaling.knative.dev/{k}", d)  # noqa: E731
    return {
        "target": max(1, int(float(g("target", 100)))),
        "metric": g("metric", "concurrency"),
        "min": max(1, int(g("min-scale", 1) or 1)),
        "max": int(g("max-scale", 0) or 0),
        "scale_down_delay": _parse_duration(g("scale-down-delay", "0s")),
        "window": _parse_duration(g("window", "60s"), 60.0),
    }


async def _autoscaler(interval=None):
    """KPA-style concurrency autoscaler for the local driver (the role
    Knative's autoscaler plays in-cluster; the chart ships Knative config
    for the K8s path — charts/kubetorch-amd/knative/). Scales a workload's
    replica count on the summed kt_active_requests of its pods:
    desired = ceil(in_flight / target), clamped to [min, max], with
    scale-down held back by scale-down-delay."""
    import httpx

    if interval is None:
        interval = float(os.environ.get("KT_AUTOSCALER_INTERVAL", "1.0"))
    last_above = {}   # (ns, name) -> ts the signal last justified >min
    while True:
        await asyncio.sleep(interval)
        if HUB.driver_name != "local":
            continue
        for (ns, name), w in list(HUB.workloads.items()):
            manifest = w.get("manifest") or {}
            spec = _autoscale_spec(manifest)
            if spec is None:
                continue
            try:
                pods = HUB.driver.pods(name, ns)
            except Exception:
                continue
            in_flight = 0.0
            async with httpx.AsyncClient(timeout=3) as client:
                for p in pods:
                    try:
                        r = await client.get(f"http://{p}/metrics")
                        for line in r.text.splitlines():
                            if line.startswith("kt_active_requests"):
                                in_flight += float(line.split()[-1])
                    except Exception:
                        pass
            want = -(-int(in_flight) // spec["target"])  # ceil
            want = max(spec["min"], want)
            if spec["max"]:
                want = min(spec["max"], want)
            cur = w.get("desired_replicas", len(pods) or 1)
            key = (ns, name)
            if want >= cur:
                last_above[key] = time.time()
            if want > cur or (want < cur and time.time()
                              - last_above.get(key, 0) > spec["scale_down_delay"]):
                w["desired_replicas"] = want
                scaled = dict(manifest)
                scaled.setdefault("spec", {})
                scaled["spec"] = dict(scaled["spec"], replicas=want)
                try:
                    await asyncio.to_thread(
                        HUB.driver.apply, scaled, ns, w.get("metadata"),
                        w.get("launch_id"))
                    HUB.driver._event(ns, name, "Autoscaled",
                                      f"{cur} -> {want} replicas "
                                      f"(in_flight={int(in_flight)}, "
                                      f"target={spec['target']})")
                except Exception:
                    pass


async def _pod_monitor(interval=None):
    """Auto re-provision: watch every registered workload and re-apply its
    manifest when live pods < desired replicas (the reconciliation loop a
    K8s Deployment controller provides; the local driver needs its own so
    a killed worker pod auto-heals without a client re-deploy — BASELINE
    config 4's 'auto re-provision'). Respawned pods join the next call's
    rendezvous (per-call process groups = elastic re-join)."""
    if interval is None:
        interval = float(os.environ.get("KT_POD_MONITOR_INTERVAL", "1.0"))
    backoff = {}  # (ns, name) -> last respawn ts
    while True:
        await asyncio.sleep(interval)
        if HUB.driver_name != "local":
            continue  # K8s reconciles its own Deployments
        for (ns, name), w in list(HUB.workloads.items()):
            manifest = w.get("manifest") or {}
            if not manifest:
                continue
            if (w.get("metadata") or {}).get("module_type") == "app":
                continue  # apps run to completion; exit is not a fault
            from kubetorch_amd.controller.drivers import desired_replicas

            desired = w.get("desired_replicas") or desired_replicas(manifest)
            try:
                alive = len(HUB.driver.pods(name, ns))
            except Exception:
                continue
            if alive >= desired:
                continue
            if time.time() - backoff.get((ns, name), 0) < 3.0:
                continue
            backoff[(ns, name)] = time.time()
            try:
                # reconcile to the CURRENT desired count — knative-kind
                # manifests carry no spec.replicas (the autoscaler owns
                # desired_replicas), so stamp it into the applied copy
                scaled = dict(manifest)
                scaled["spec"] = dict(scaled.get("spec", {}),
                                      replicas=desired)
                await asyncio.to_thread(
                    HUB.driver.apply, scaled, ns, w.get("metadata"),
                    w.get("launch_id"))
                HUB.driver._event(ns, name, "Respawned",
                                  f"re-provisioned {desired - alive} pod(s)")
            except Exception:
                pass


from contextlib import asynccontextmanager


@asynccontextmanager
async def _lifespan(app):
    if hasattr(HUB.driver, "load_workloads"):
        # controller restart: rehydrate the registry from the durable
        # KubetorchWorkload CRs (reference parity: the CRD IS the registry)
        try:
            for w in await asyncio.to_thread(HUB.driver.load_workloads):
                HUB.workloads.setdefault(
                    (w["namespace"], w["name"]), w)
        except Exception:
            pass
    task = asyncio.create_task(_ttl_reaper())
    mon = asyncio.create_task(_pod_monitor())
    scaler = asyncio.create_task(_autoscaler())
    yield
    task.cancel()
    mon.cancel()
    scaler.cancel()


app = FastAPI(lifespan=_lifespan)


@app.get("/health")
def health():
    return {"status": "ok", "driver": HUB.driver_name}


@app.get("/controller/config")
def cluster_config():
    """Cluster-wide client defaults (the chart mounts the kubetorch-config
    ConfigMap at /etc/kubetorch/config.yaml; KT_CLUSTER_CONFIG JSON
    overrides for tests/BYO)."""
    import json as _json

    raw = os.environ.get("KT_CLUSTER_CONFIG")
    if raw:
        try:
            return {"config": _json.loads(raw)}
        except ValueError:
            pass
    path = os.environ.get("KT_CLUSTER_CONFIG_PATH",
                          "/etc/kubetorch/config.yaml")
    if os.path.exists(path):
        try:
            import yaml

            with open(path) as f:
                return {"config": yaml.safe_load(f) or {}}
        except Exception:
            pass
    return {"config": {}}


@app.post("/controller/deploy")
async def deploy(request: Request):
    """Apply the manifest and hot-reload the workload's pods.
    Body: {namespace, name, manifest, metadata, service_config}."""
    body = await request.json()
    ns = body.get("namespace", "default")
    name = body["name"]
    manifest = body.get("manifest") or {}
    metadata = body.get("metadata") or {}
    launch_id = body.get("launch_id") or uuid.uuid4().hex[:12]

    key = (ns, name)
    existing = HUB.workloads.get(key)
    HUB.workloads[key] = {
        "name": name,
        "namespace": ns,
        "manifest": manifest,
        "metadata": metadata,
        "service_config": body.get("service_config") or {},
        "launch_id": launch_id,
        "created": existing["created"] if existing else time.time(),
        "updated": time.time(),
    }

    hosts = []
    if manifest:
        try:
            hosts = await asyncio.to_thread(
                HUB.driver.apply, manifest, ns, metadata, launch_id
            )
        except Exception as e:
            return JSONResponse(
                {"error": f"driver apply failed: {e}"}, status_code=500
            )
        if hasattr(HUB.driver, "persist_workload"):
            # durable registry: a controller restart rehydrates from the
            # KubetorchWorkload CRs (best-effort — CRD may not be installed)
            try:
                await asyncio.to_thread(
                    HUB.driver.persist_workload, HUB.workloads[key])
            except Exception:
                pass
        if HUB.driver_name == "k8s" and manifest.get("kind") == "Deployment":
            # route + discovery Services (the reference controller creates
            # these; a custom endpoint selector narrows call routing)
            from kubetorch_amd.provisioning.manifests import (
                build_service_manifests,
            )

            ep = (body.get("service_config") or {}).get("endpoint") or {}
            svc, headless = build_service_manifests(
                name, ns, selector=ep.get("selector"))
            for m in (svc, headless):
                try:
                    await asyncio.to_thread(HUB.driver.apply, m, ns)
                except Exception:
                    pass

    # hot-reload connected pods of this service and wait for acks
    pods = HUB.pods_for_service(name, ns)
    acked = []
    if pods:
        ev = asyncio.Event()
        HUB.acks[launch_id] = {"pending": set(pods), "event": ev}
        msg = {"action": "reload", "metadata": metadata, "launch_id": launch_id}
        for p in pods:
            await HUB.pod_queues[p].put(msg)
        try:
            await asyncio.wait_for(ev.wait(), timeout=C.RELOAD_ACK_TIMEOUT)
            acked = pods
        except asyncio.TimeoutError:
            acked = [p for p in pods
                     if p not in HUB.acks[launch_id]["pending"]]
        finally:
            HUB.acks.pop(launch_id, None)

    return {"ok": True, "launch_id": launch_id, "hosts": hosts,
            "reloaded_pods": acked, "driver": HUB.driver_name}


@app.post("/controller/volumes/{ns}")
async def put_volume(ns: str, request: Request):
    """Create a PVC through the control plane (reference: Volume create
    via controller). Body: the PVC manifest."""
    manifest = await request.json()
    try:
        await asyncio.to_thread(HUB.driver.apply_volume, manifest, ns)
    except Exception as e:
        return JSONResponse({"error": f"pvc apply failed: {e}"},
                            status_code=500)
    return {"ok": True, "name": manifest.get("metadata", {}).get("name")}


@app.get("/controller/volumes/{ns}")
def list_volumes(ns: str):
    try:
        return {"volumes": HUB.driver.list_volumes(ns)}
    except Exception as e:
        return JSONResponse({"error": str(e)}, status_code=500)


@app.delete("/controller/volumes/{ns}/{name}")
async def delete_volume(ns: str, name: str):
    try:
        await asyncio.to_thread(HUB.driver.delete_volume, name, ns)
    except Exception as e:
        return JSONResponse({"error": str(e)}, status_code=500)
    return {"ok": True}


@app.post("/controller/secrets/{ns}")
async def put_secret(ns: str, request: Request):
    """Create/update a secret through the control plane (reference parity:
    kubernetes_secrets_client.py — the client never needs direct K8s
    credentials). Body: {name, values, as_env, mount_path}."""
    body = await request.json()
    try:
        await asyncio.to_thread(HUB.driver.apply_secret, body, ns)
    except Exception as e:
        return JSONResponse({"error": f"secret apply failed: {e}"},
                            status_code=500)
    return {"ok": True, "name": body.get("name")}


@app.get("/controller/secrets/{ns}")
def list_secrets(ns: str):
    try:
        return {"secrets": HUB.driver.list_secrets(ns)}
    except Exception as e:
        return JSONResponse({"error": str(e)}, status_code=500)


@app.delete("/controller/secrets/{ns}/{name}")
async def delete_secret(ns: str, name: str):
    try:
        await asyncio.to_thread(HUB.driver.delete_secret, name, ns)
    except Exception as e:
        return JSONResponse({"error": str(e)}, status_code=500)
    return {"ok": True}


# -- TCP tunnel for out-of-cluster clients (reference parity:
# -- data_store/websocket_tunnel.py — firewall traversal through the one
# -- public controller port). No WS stack in this image, so the bridge is
# -- two chunked-HTTP streams per connection: POST .../up carries
# -- client->service bytes, GET .../down streams service->client.
TUNNELS = {}


@app.post("/controller/tunnel/open")
async def tunnel_open(request: Request):
    import socket as _socket

    body = await request.json()
    ns = body.get("namespace", "default")
    service = body["service"]
    port = int(body["port"])
    host = None
    if ":" in service:
        # explicit host:port target — allowed only toward loopback on the
        # local driver (dev mode); in-cluster the tunnel must name a
        # service, otherwise the controller becomes a generic pivot into
        # any network it can reach
        host, _, p = service.partition(":")
        port = int(p)
        if HUB.driver_name != "local" or host not in ("127.0.0.1",
                                                      "localhost"):
            return JSONResponse(
                {"error": "explicit host:port targets are restricted to "
                          "loopback on the local driver; name a service "
                          "instead"}, status_code=403)
    elif HUB.driver_name == "local":
        pods = HUB.driver.pods(service, ns)
        if not pods:
            return JSONResponse({"error": f"no pods for {service}"},
                                status_code=404)
        host, _, p = pods[0].partition(":")
        port = int(p or port)
    else:
        host = f"{service}.{ns}.svc.cluster.local"
    # prune stale tunnels (client opened but never drained)
    cutoff = time.time() - 3600
    for tid_, t_ in list(TUNNELS.items()):
        if t_["created"] < cutoff:
            try:
                t_["sock"].close()
            except OSError:
                pass
            TUNNELS.pop(tid_, None)
    try:
        sock = _socket.create_connection((host, port), timeout=10)
    except OSError as e:
        return JSONResponse({"error": f"connect {host}:{port}: {e}"},
                            status_code=502)
    tid = uuid.uuid4().hex
    TUNNELS[tid] = {"sock": sock, "created": time.time()}
    return {"tunnel_id": tid, "target": f"{host}:{port}"}


@app.post("/controller/tunnel/{tid}/up")
async def tunnel_up(tid: str, request: Request):
    t = TUNNELS.get(tid)
    if t is None:
        return JSONResponse({"error": "no such tunnel"}, status_code=404)
    sock = t["sock"]
    try:
        async for chunk in request.stream():
            if chunk:
                await asyncio.to_thread(sock.sendall, chunk)
    except Exception:
        pass
    try:
        sock.shutdown(1)  # half-close: service sees EOF upstream
    except OSError:
        pass
    return {"ok": True}


@app.get("/controller/tunnel/{tid}/down")
async def tunnel_down(tid: str):
    from fastapi.responses import StreamingResponse

    t = TUNNELS.get(tid)
    if t is None:
        return JSONResponse({"error": "no such tunnel"}, status_code=404)
    sock = t["sock"]

    async def gen():
        try:
            while True:
                data = await asyncio.to_thread(sock.recv, 65536)
                if not data:
                    break
                yield data
        finally:
            TUNNELS.pop(tid, None)
            try:
                sock.close()
            except OSError:
                pass

    return StreamingResponse(gen(), media_type="application/octet-stream")


@app.post("/controller/workload")
async def register_workload(request: Request):
    body = await request.json()
    ns = body.get("namespace", "default")
    name = body["name"]
    HUB.workloads[(ns, name)] = {
        "name": name, "namespace": ns,
        "manifest": body.get("manifest") or {},
        "metadata": body.get("metadata") or {},
        "service_config": body.get("service_config") or {},
        "launch_id": body.get("launch_id"),
        "created": time.time(), "updated": time.time(),
    }
    if hasattr(HUB.driver, "persist_workload"):
        try:
            await asyncio.to_thread(
                HUB.driver.persist_workload, HUB.workloads[(ns, name)])
        except Exception:
            pass
    return {"ok": True}


@app.get("/controller/workloads/{ns}")
def list_workloads(ns: str):
    return {"workloads": [w for (n, _), w in HUB.workloads.items() if n == ns]}


@app.get("/controller/workload/{ns}/{name}")
def get_workload(ns: str, name: str):
    w = HUB.workloads.get((ns, name))
    if not w:
        return JSONResponse({"error": "not found"}, status_code=404)
    pods = HUB.driver.pods(name, ns) if hasattr(HUB.driver, "pods") else []
    return {**w, "pods": pods}


@app.delete("/controller/workload/{ns}/{name}")
async def delete_workload(ns: str, name: str):
    HUB.workloads.pop((ns, name), None)
    try:
        await asyncio.to_thread(HUB.driver.delete, name, ns)
    except Exception:
        pass
    if hasattr(HUB.driver, "delete_workload_cr"):
        try:
            await asyncio.to_thread(HUB.driver.delete_workload_cr, name, ns)
        except Exception:
            pass
    return {"ok": True}


@app.get("/controller/podlogs/{ns}/{name}")
def pod_logs(ns: str, name: str, offset: int = 0):
    """Raw pod main-process output (kubectl-logs passthrough; the source
    for App foreground log-follow)."""
    if not hasattr(HUB.driver, "pod_logs"):
        return {"text": "", "offset": offset}
    try:
        text, new_off = HUB.driver.pod_logs(name, ns, offset=offset)
        return {"text": text, "offset": new_off}
    except Exception as e:
        return {"text": "", "offset": offset, "error": str(e)}


@app.get("/controller/events/{ns}/{name}")
def service_events(ns: str, name: str, since: float = 0.0):
    """Pod lifecycle / K8s events for a service, polled by clients during
    `.to()` launches (reference parity: launch-time K8s event streaming)."""
    if not hasattr(HUB.driver, "get_events"):
        return {"events": []}
    try:
        return {"events": HUB.driver.get_events(name, ns, since=since)}
    except Exception as e:
        return {"events": [], "error": str(e)}


@app.post("/controller/pods/stream")
async def pod_stream(request: Request):
    """Pod registration; the response is an endless NDJSON stream of pushes.
    First push is the current module metadata (request_metadata contract)."""
    reg = await request.json()
    pod_name = reg["pod_name"]
    q = asyncio.Queue()
    HUB.pod_queues[pod_name] = q
    HUB.pod_info[pod_name] = reg

    if reg.get("request_metadata"):
        w = HUB.workloads.get((reg.get("namespace", "default"),
                               reg.get("service_name")))
        if w:
            await q.put({"action": "metadata", "metadata": w["metadata"],
                         "launch_id": w.get("launch_id")})

    async def gen():
        try:
            while True:
                try:
                    msg = await asyncio.wait_for(q.get(), timeout=15.0)
                except asyncio.TimeoutError:
                    msg = {"action": "ping"}
                yield json.dumps(msg) + "\n"
        finally:
            HUB.pod_queues.pop(pod_name, None)
            HUB.pod_info.pop(pod_name, None)

    return StreamingResponse(gen(), media_type="application/x-ndjson")


@app.post("/controller/pods/reload_ack")
async def reload_ack(request: Request):
    body = await request.json()
    rec = HUB.acks.get(body.get("launch_id"))
    if rec:
        rec["pending"].discard(body.get("pod_name"))
        if not rec["pending"]:
            rec["event"].set()
    return {"ok": True}


@app.api_route("/controller/k8s/{path:path}",
               methods=["GET", "POST", "PUT", "DELETE", "PATCH"])
async def k8s_proxy(path: str, request: Request):
    """K8s API passthrough for out-of-cluster clients (reference: nginx
    /api|/apis routes + controller proxy). Uses the controller pod's RBAC
    via kubectl --raw; unavailable on the local driver."""
    import subprocess

    from kubetorch_amd.controller.drivers import K8sDriver

    if not K8sDriver().available():
        return JSONResponse({"error": "kubectl unavailable (local driver)"},
                            status_code=501)
    verb_map = {"GET": "get", "POST": "create", "PUT": "replace",
                "DELETE": "delete", "PATCH": "patch"}
    verb = verb_map[request.method]
    args = ["kubectl", verb, "--raw", "/" + path]
    body = await request.body()
    kw = {}
    if body:
        args += ["-f", "-"]
        kw["input"] = body
    res = await asyncio.to_thread(
        subprocess.run, args, capture_output=True, **kw)
    if res.returncode != 0:
        return JSONResponse({"error": res.stderr.decode()[-2000:]},
                            status_code=502)
    import json as _json

    try:
        return _json.loads(res.stdout)
    except ValueError:
        return Response(res.stdout)


@app.get("/controller/debug/connections")
def debug_connections():
    return {"pods": list(HUB.pod_info.values()),
            "workloads": [f"{ns}/{n}" for ns, n in HUB.workloads]}


def main():
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=C.CONTROLLER_PORT)
    ap.add_argument("--host", default="0.0.0.0")
    args = ap.parse_args()
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
