"""HTTP fan-out to peer worker pods (reference parity: remote_worker_pool.py).

Two engines:
  * AsyncFanout — a dedicated asyncio loop thread + one AsyncClient with a
    200-subcall concurrency cap (the reference runs the same engine in a
    subprocess). The SPMD coordinator uses this: 100+ pods fan out on one
    thread instead of one thread per host.
  * call_worker_subcall — the sync path, used by tree-relay nodes whose
    fan-out is bounded by TREE_FANOUT anyway.
"""
import asyncio
import threading

import httpx

from kubetorch_amd import constants as C
from kubetorch_amd.exceptions import reconstruct_exception

MAX_CONCURRENT_SUBCALLS = 200  # reference: remote_worker_pool.py cap

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


class AsyncFanout:
    """Singleton asyncio fan-out engine on its own loop thread. submit()
    returns a concurrent.futures.Future, so the supervisor's
    wait(FIRST_EXCEPTION) logic works unchanged alongside local-rank
    futures."""

    _inst = None
    _inst_lock = threading.Lock()

    def __init__(self):
        self._loop = asyncio.new_event_loop()
        t = threading.Thread(target=self._loop.run_forever, daemon=True,
                             name="kt-async-fanout")
        t.start()
        self._client = None
        self._sem = None

    @classmethod
    def instance(cls):
        with cls._inst_lock:
            if cls._inst is None:
                cls._inst = cls()
            return cls._inst

    async def _ensure(self):
        if self._client is None:
            self._client = httpx.AsyncClient(
                timeout=httpx.Timeout(C.HTTP_TIMEOUT, connect=10),
                limits=httpx.Limits(
                    max_connections=MAX_CONCURRENT_SUBCALLS + 56))
            self._sem = asyncio.Semaphore(MAX_CONCURRENT_SUBCALLS)

    async def _health_wait(self, host, timeout=30.0):
        import time

        deadline = time.time() + timeout
        interval = 0.25
        while time.time() < deadline:
            try:
                r = await self._client.get(f"http://{host}/health", timeout=5)
                if r.status_code == 200:
                    return True
            except httpx.HTTPError:
                pass
            await asyncio.sleep(interval)
            interval = min(interval * 1.5, 2.0)
        return False

    async def _subcall(self, host, body_b64, method, hosts, subtree, timeout):
        await self._ensure()
        url = f"http://{host}/spmd/subcall"
        payload = {"body": body_b64, "method": method, "hosts": hosts,
                   "subtree": subtree}
        async with self._sem:
            try:
                r = await self._client.post(
                    url, json=payload, timeout=timeout or C.HTTP_TIMEOUT)
            except (httpx.ConnectError, httpx.ConnectTimeout):
                if not await self._health_wait(host):
                    raise
                r = await self._client.post(
                    url, json=payload, timeout=timeout or C.HTTP_TIMEOUT)
        data = r.json()
        if r.status_code != 200:
            raise reconstruct_exception(data.get("error", data))
        import base64
        import pickle

        return pickle.loads(base64.b64decode(data["result"]))

    def submit(self, host, body_b64, method, hosts, subtree, timeout):
        return asyncio.run_coroutine_threadsafe(
            self._subcall(host, body_b64, method, hosts, subtree, timeout),
            self._loop)


def fanout():
    return AsyncFanout.instance()
