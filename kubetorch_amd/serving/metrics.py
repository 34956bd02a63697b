"""Pod metrics: prometheus-client registry exposed at /metrics and
optionally pushed to the namespace metrics store. Feeds the controller's
inactivity-TTL reaper via kt_last_activity_timestamp.
(Reference parity: serving/metrics_push.py, server_metrics.py.)"""
import os
import threading
import time

from prometheus_client import (
    CollectorRegistry,
    Counter,
    Gauge,
    Histogram,
    generate_latest,
)

from kubetorch_amd import constants as C

REGISTRY = CollectorRegistry()

HTTP_REQUESTS = Counter(
    "kt_http_requests_total", "HTTP requests", ["path", "status"],
    registry=REGISTRY,
)
HTTP_DURATION = Histogram(
    "kt_http_request_duration_seconds", "request latency", ["path"],
    registry=REGISTRY,
)
LAST_ACTIVITY = Gauge(
    "kt_last_activity_timestamp", "unix ts of last user call",
    registry=REGISTRY,
)
ACTIVE_REQUESTS = Gauge(
    "kt_active_requests", "in-flight user calls", registry=REGISTRY,
)
HEARTBEAT = Counter("kt_heartbeat_sent", "TTL heartbeats", registry=REGISTRY)


# custom (optional) collectors — e.g. the amd-smi GPU collector — are
# tracked so a broken one can be evicted instead of 500-ing /metrics
CUSTOM_COLLECTORS = []


def register_custom(collector):
    REGISTRY.register(collector)
    CUSTOM_COLLECTORS.append(collector)


def exposition() -> bytes:
    try:
        return generate_latest(REGISTRY)
    except Exception:
        # a failing optional collector must never starve consumers of the
        # core metrics (autoscaler signal, TTL heartbeat): evict and retry
        while CUSTOM_COLLECTORS:
            c = CUSTOM_COLLECTORS.pop()
            try:
                REGISTRY.unregister(c)
            except Exception:
                pass
        return generate_latest(REGISTRY)


def inactivity_ttl_seconds():
    """Parse the inactivity TTL from env/annotation ('120s', '5m', '2h')."""
    raw = os.environ.get("KT_INACTIVITY_TTL")
    if not raw:
        return None
    raw = raw.strip().lower()
    mult = {"s": 1, "m": 60, "h": 3600, "d": 86400}.get(raw[-1])
    try:
        return int(float(raw[:-1]) * mult) if mult else int(float(raw))
    except ValueError:
        return None


class MetricsPusher:
    """Pushes the registry to the metrics store every interval; sends TTL
    heartbeats at ttl/5 while requests are active."""

    def __init__(self, push_url=None, interval=15.0):
        self.push_url = push_url or os.environ.get("KT_METRICS_PUSH_URL")
        self.interval = interval
        self._stop = threading.Event()
        self._thread = None

    def start(self):
        if not self.push_url:
            return
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()

    def _run(self):
        import httpx

        while not self._stop.wait(self.interval):
            try:
                httpx.post(self.push_url, content=exposition(),
                           headers={"X-Pod": os.environ.get("POD_NAME", "")},
                           timeout=5)
                HEARTBEAT.inc()
            except Exception:
                pass


def touch_activity():
    LAST_ACTIVITY.set(time.time())
