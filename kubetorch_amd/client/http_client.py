"""Client->pod HTTP calls with remote-exception reconstruction and optional
live log streaming. (Reference parity: serving/http_client.py.)"""
import base64
import json
import pickle
import threading
import uuid

import httpx

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import reconstruct_exception

_client = None
_client_lock = threading.Lock()


def shared_client():
    global _client
    with _client_lock:
        if _client is None:
            _client = httpx.Client(
                timeout=httpx.Timeout(C.HTTP_TIMEOUT, connect=30),
                limits=httpx.Limits(max_connections=100),
            )
        return _client


class LogStreamer:
    """Background thread tailing /logs/tail for a request id and printing
    lines as they arrive (poll-based tail; sub-second latency)."""

    def __init__(self, base_url, request_id, printer=print):
        self.base_url = base_url
        self.request_id = request_id
        self.printer = printer
        self._stop = threading.Event()
        self._since = None
        self._thread = threading.Thread(target=self._run, daemon=True)

    def start(self):
        try:
            r = shared_client().get(self.base_url + "/logs/tail",
                                    params={"since": 0, "limit": 1})
            self._since = r.json().get("seq", 0)
        except Exception:
            self._since = 0
        self._thread.start()
        return self

    def stop(self, drain=True):
        if drain:
            self._poll_once()
        self._stop.set()

    def _poll_once(self):
        try:
            r = shared_client().get(
                self.base_url + "/logs/tail",
                params={"since": self._since, "request_id": self.request_id},
                timeout=5,
            )
            for e in r.json().get("entries", []):
                self.printer(f"[remote {e['source']}] {e['line']}")
                self._since = max(self._since, e["seq"] + 1)
        except Exception:
            pass

    def _run(self):
        while not self._stop.wait(0.25):
            self._poll_once()


class MetricsStreamer:
    """Polls the pod's /metrics during a call and prints hardware lines
    (GPU util/VRAM/power from the amd-smi collector). Reference parity:
    per-call metric streaming, http_client.py:758-954."""

    WATCH = ("kt_gpu_utilization_percent", "kt_gpu_vram_used_bytes",
             "kt_gpu_power_watts", "kt_active_requests")

    def __init__(self, base_url, interval=3.0, printer=print):
        self.base_url = base_url
        self.interval = interval
        self.printer = printer
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()

    def _run(self):
        while not self._stop.wait(self.interval):
            try:
                r = shared_client().get(self.base_url + "/metrics", timeout=5)
                vals = {}
                for line in r.text.splitlines():
                    for w in self.WATCH:
                        if line.startswith(w):
                            vals[line.split()[0]] = float(line.split()[-1])
                if vals:
                    pretty = " ".join(f"{k.split('kt_')[-1]}={v:g}"
                                      for k, v in sorted(vals.items()))
                    self.printer(f"[remote metrics] {pretty}")
            except Exception:
                pass


class HTTPClient:
    def __init__(self, base_url, name):
        self.base_url = base_url.rstrip("/")
        self.name = name

    def is_ready(self, launch_id=None, timeout=5):
        try:
            r = shared_client().get(
                self.base_url + "/ready",
                params={"launch_id": launch_id} if launch_id else {},
                timeout=timeout,
            )
            return r.status_code == 200
        except httpx.HTTPError:
            return False

    def call(self, args=(), kwargs=None, method=None, serialization="json",
             stream_logs=False, stream_metrics=False, timeout=None,
             workers=None, restart_procs=False, request_id=None,
             debug=False):
        rid = request_id or uuid.uuid4().hex
        url = f"{self.base_url}/call/{self.name}"
        if method:
            url += f"/{method}"
        params = {}
        if workers is not None:
            params["workers"] = (workers if isinstance(workers, str)
                                 else json.dumps(workers))
        if restart_procs:
            params["restart_procs"] = "true"
        headers = {"X-Request-ID": rid, "X-Serialization": serialization}
        if debug:
            # breakpoints in this call wait for `kt debug <service>`
            headers["X-KT-Debug"] = "1"
            print(f"[kt] debug call: breakpoints in {self.name} will wait "
                  f"for `kt debug` on the pod's KT_DEBUG_PORT")
        if serialization == "pickle":
            body = {"body": base64.b64encode(
                pickle.dumps((tuple(args), kwargs or {}))).decode()}
        else:
            body = {"args": list(args), "kwargs": kwargs or {}}
            try:
                json.dumps(body)
            except (TypeError, ValueError) as e:
                raise TypeError(
                    "arguments are not JSON-serializable; call with "
                    "serialization='pickle' and deploy with "
                    "Compute(allowed_serialization=['json', 'pickle'])"
                ) from e
        streamer = None
        mstreamer = None
        if stream_logs:
            streamer = LogStreamer(self.base_url, rid).start()
        if stream_metrics:
            mstreamer = MetricsStreamer(self.base_url).start()
        try:
            last = None
            for attempt in range(4):
                try:
                    r = shared_client().post(
                        url, json=body, params=params, headers=headers,
                        timeout=timeout or C.HTTP_TIMEOUT,
                    )
                    break
                except (httpx.ConnectError, httpx.ConnectTimeout) as e:
                    # connection never established -> safe to retry (pod may
                    # still be binding its port after a reload/launch)
                    last = e
                    import time

                    time.sleep(0.5 * (attempt + 1))
            else:
                raise last
        finally:
            if streamer:
                streamer.stop()
            if mstreamer:
                mstreamer.stop()
        try:
            data = r.json()
        except json.JSONDecodeError:
            r.raise_for_status()
            raise
        if r.status_code != 200 or "error" in data:
            raise reconstruct_exception(data.get("error", {"message": r.text}))
        if "result_pickle" in data:
            return pickle.loads(base64.b64decode(data["result_pickle"]))
        if serialization == "pickle":
            return pickle.loads(base64.b64decode(data["result"]))
        return data["result"]

    def reload(self, metadata, launch_id):
        r = shared_client().post(
            self.base_url + "/reload",
            json={"metadata": metadata, "launch_id": launch_id},
            timeout=C.LAUNCH_TIMEOUT,
        )
        data = r.json()
        if r.status_code != 200:
            raise reconstruct_exception(data.get("error", {}))
        return data

    def logs(self, since=0, request_id=None, limit=1000):
        r = shared_client().get(
            self.base_url + "/logs/tail",
            params={"since": since, "limit": limit,
                    **({"request_id": request_id} if request_id else {})},
        )
        return r.json().get("entries", [])
