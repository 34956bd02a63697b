"""In-pod log capture: stdout/stderr interception into a ring buffer with
request-id labels, drained to the HTTP/WS tail endpoints and (optionally)
pushed to the namespace log store. Worker subprocess output arrives via a
multiprocessing queue. (Reference parity: serving/log_capture.py — Loki is
replaced by our own log-store service in the data-store pod.)"""
import contextvars
import io
import sys
import threading
import time

request_id_var = contextvars.ContextVar("kt_request_id", default=None)

RING_SIZE = 10000


class RingLog:
    def __init__(self, size=RING_SIZE):
        self.size = size
        self._buf = []
        self._seq = 0
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)

    def append(self, line, source="stdout", request_id=None, level="INFO"):
        entry = {
            "seq": None,
            "ts": time.time(),
            "line": line.rstrip("\n"),
            "source": source,
            "request_id": request_id,
            "level": level,
        }
        with self._cond:
            entry["seq"] = self._seq
            self._seq += 1
            self._buf.append(entry)
            if len(self._buf) > self.size:
                self._buf = self._buf[-self.size:]
            self._cond.notify_all()
        return entry

    def tail(self, since=0, request_id=None, limit=1000):
        with self._lock:
            out = [e for e in self._buf if e["seq"] >= since
                   and (request_id is None or e["request_id"] == request_id)]
        return out[-limit:]

    def wait_for(self, since, timeout=10.0):
        with self._cond:
            if self._seq > since:
                return True
            return self._cond.wait(timeout)

    @property
    def seq(self):
        return self._seq


class _StreamInterceptor(io.TextIOBase):
    def __init__(self, ring, orig, source):
        self.ring = ring
        self.orig = orig
        self.source = source
        self._partial = ""

    def write(self, s):
        self.orig.write(s)
        self._partial += s
        while "\n" in self._partial:
            line, self._partial = self._partial.split("\n", 1)
            if line.strip():
                self.ring.append(line, source=self.source,
                                 request_id=request_id_var.get())
        return len(s)

    def flush(self):
        self.orig.flush()

    def isatty(self):
        return False

    def fileno(self):
        return self.orig.fileno()


RING = RingLog()
_installed = False


def install():
    global _installed
    if _installed:
        return RING
    sys.stdout = _StreamInterceptor(RING, sys.stdout, "stdout")
    sys.stderr = _StreamInterceptor(RING, sys.stderr, "stderr")
    _installed = True
    return RING


def drain_worker_queue(log_q, stop_event):
    """Background thread: worker subprocess log lines -> ring buffer."""
    while not stop_event.is_set():
        try:
            item = log_q.get(timeout=0.5)
        except Exception:
            continue
        if item is None:
            return
        RING.append(item.get("line", ""), source=item.get("source", "worker"),
                    request_id=item.get("request_id"))
