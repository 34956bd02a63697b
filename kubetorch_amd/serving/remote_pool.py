"""HTTP fan-out to peer worker pods (reference parity: remote_worker_pool.py).

The reference runs a dedicated asyncio subprocess for this; here a
process-wide httpx client + the caller's thread pool is enough (the fan-out
is bounded by TREE_FANOUT=50 concurrent requests per node)."""
import threading

import httpx

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import reconstruct_exception

_client = None
_client_lock = threading.Lock()


def get_client():
    global _client
    with _client_lock:
        if _client is None:
            _client = httpx.Client(
                timeout=httpx.Timeout(C.HTTP_TIMEOUT, connect=10),
                limits=httpx.Limits(max_connections=200),
            )
        return _client


def wait_worker_health(host, timeout=30.0, interval=0.25):
    """Block until the peer pod's server answers /health (a freshly
    re-provisioned pod may still be booting when the fan-out reaches it —
    reference parity: remote_worker_pool health-wait per worker up to the
    quorum timeout). Returns True if healthy, False on timeout."""
    import time

    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            if get_client().get(f"http://{host}/health",
                                timeout=5).status_code == 200:
                return True
        except httpx.HTTPError:
            pass
        time.sleep(interval)
        interval = min(interval * 1.5, 2.0)
    return False


def call_worker_subcall(host, body_b64, method, hosts, subtree, timeout,
                        health_timeout=30.0):
    """POST the serialized call to a peer pod as a distributed subcall.
    Returns the peer's list of per-rank results (already deserialized)."""
    url = f"http://{host}/spmd/subcall"
    payload = {
        "body": body_b64,
        "method": method,
        "hosts": hosts,
        "subtree": subtree,
    }
    try:
        r = get_client().post(url, json=payload,
                              timeout=timeout or C.HTTP_TIMEOUT)
    except (httpx.ConnectError, httpx.ConnectTimeout):
        # pod process exists but its server isn't up yet (respawn/boot):
        # health-wait once, then retry; a truly dead pod surfaces through
        # the membership monitor instead
        if not wait_worker_health(host, timeout=health_timeout):
            raise
        r = get_client().post(url, json=payload,
                              timeout=timeout or C.HTTP_TIMEOUT)
    data = r.json()
    if r.status_code != 200:
        raise reconstruct_exception(data.get("error", data))
    import base64
    import pickle

    return pickle.loads(base64.b64decode(data["result"]))
