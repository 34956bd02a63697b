"""Worker peer discovery with quorum wait + membership monitoring.

Two sources (reference parity: serving/distributed_supervisor.py:90-339):
  * KT_LOCAL_IPS env — explicit "host:port,host:port" list. Used by the
    local-process driver and tests (fake-cluster mode), and by BYO setups.
  * DNS of the headless service `{svc}-headless.{ns}.svc.cluster.local` —
    the in-cluster path; eventually consistent, hence quorum + monitor.
"""
import os
import socket
import threading
import time

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import QuorumTimeout, WorkerMembershipChanged


def _resolve_dns(hostname, port):
    try:
        infos = socket.getaddrinfo(hostname, None, socket.AF_INET)
        return sorted({f"{i[4][0]}:{port}" for i in infos})
    except socket.gaierror:
        return []


def current_peers(service_name=None, namespace=None, port=C.SERVER_PORT):
    peers_url = os.environ.get("KT_PEERS_URL")
    if peers_url:
        # live peer list from the controller (local driver / BYO): reflects
        # pod death, unlike the static KT_LOCAL_IPS env
        try:
            import httpx

            svc = service_name or os.environ.get(C.ENV_SERVICE_NAME)
            ns = namespace or os.environ.get("POD_NAMESPACE", "default")
            r = httpx.get(f"{peers_url}/controller/workload/{ns}/{svc}",
                          timeout=5)
            if r.status_code == 200:
                return sorted(r.json().get("pods", []))
        except Exception:
            pass
    local = os.environ.get(C.ENV_LOCAL_IPS)
    if local:
        return sorted(p.strip() for p in local.split(",") if p.strip())
    svc = service_name or os.environ.get(C.ENV_SERVICE_NAME)
    ns = namespace or os.environ.get("POD_NAMESPACE", "default")
    if not svc:
        return []
    return _resolve_dns(f"{svc}-headless.{ns}.svc.cluster.local", port)


def pod_ips(num_workers=None, timeout=C.QUORUM_TIMEOUT, service_name=None,
            namespace=None, port=C.SERVER_PORT, interval=1.0):
    """Block until `num_workers` peers are visible (quorum), with backoff.
    User-facing in-pod discovery for custom rendezvous (reference:
    distributed/utils.py:19-129)."""
    if num_workers is None:
        num_workers = int(os.environ.get("KT_NUM_WORKERS", "1"))
    deadline = time.time() + timeout
    peers = []
    sleep = interval
    while time.time() < deadline:
        peers = current_peers(service_name, namespace, port)
        if len(peers) >= num_workers:
            return peers
        time.sleep(sleep)
        sleep = min(sleep * 1.5, 10.0)
    raise QuorumTimeout(
        f"quorum not reached: {len(peers)}/{num_workers} workers after {timeout}s"
    )


class MembershipMonitor:
    """Background thread diffing the peer set every few seconds; on change,
    notifies subscribers with WorkerMembershipChanged."""

    def __init__(self, service_name=None, namespace=None,
                 interval=C.DNS_MONITOR_INTERVAL, port=C.SERVER_PORT):
        self.service_name = service_name
        self.namespace = namespace
        self.interval = interval
        self.port = port
        self._baseline = None
        self._subs = []
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread = None

    def start(self, baseline):
        self._baseline = set(baseline)
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()

    def rebase(self, baseline):
        """Accept the current peer set as the new normal (start of a new
        call): prior joins/leaves no longer count as a change."""
        self._baseline = set(baseline)

    def subscribe(self, callback):
        with self._lock:
            self._subs.append(callback)

    def unsubscribe(self, callback):
        with self._lock:
            if callback in self._subs:
                self._subs.remove(callback)

    def check_now(self):
        cur = set(current_peers(self.service_name, self.namespace, self.port))
        if self._baseline is not None and cur and cur != self._baseline:
            added = cur - self._baseline
            removed = self._baseline - cur
            exc = WorkerMembershipChanged(
                f"worker set changed: +{sorted(added)} -{sorted(removed)}",
                added=added, removed=removed,
            )
            self._baseline = cur
            with self._lock:
                subs = list(self._subs)
            for cb in subs:
                try:
                    cb(exc)
                except Exception:
                    pass
            return exc
        return None

    def _run(self):
        while not self._stop.wait(self.interval):
            self.check_now()
