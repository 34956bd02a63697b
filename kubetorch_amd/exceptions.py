"""Framework exceptions + remote exception packaging.

The pod runtime packages exceptions as JSON (type, message, traceback,
optional pickled payload); the client re-raises the real class with the
remote traceback attached (reference parity: serving/http_server.py:1478
and serving/http_client.py:87-196)."""
import base64
import pickle
import traceback


class KubetorchError(Exception):
    pass


class LaunchError(KubetorchError):
    pass


class ImagePullError(LaunchError):
    pass


class ResourceNotAvailableError(LaunchError):
    pass


class PodTerminatedError(KubetorchError):
    def __init__(self, msg, reason=None):
        super().__init__(msg)
        self.reason = reason

    @property
    def evicted(self):
        return self.reason == "Evicted"

    @property
    def oom_killed(self):
        return self.reason == "OOMKilled"


class WorkerMembershipChanged(KubetorchError):
    """Raised into in-flight distributed calls when the worker set changes
    (pod death / scale event). Callers may retry: the next call re-runs
    quorum and re-forms process groups (elastic re-join)."""
    def __init__(self, msg, added=(), removed=()):
        super().__init__(msg)
        self.added = list(added)
        self.removed = list(removed)


class RemoteCallError(KubetorchError):
    """Raised when the remote exception class could not be reconstructed."""
    def __init__(self, error_type, message, remote_traceback):
        super().__init__(f"{error_type}: {message}\n--- remote traceback ---\n{remote_traceback}")
        self.error_type = error_type
        self.remote_traceback = remote_traceback


class QuorumTimeout(KubetorchError):
    pass


REGISTRY = {
    c.__name__: c
    for c in (
        KubetorchError, LaunchError, ImagePullError, ResourceNotAvailableError,
        PodTerminatedError, WorkerMembershipChanged, RemoteCallError,
        QuorumTimeout,
    )
}


def package_exception(exc: BaseException) -> dict:
    """Serialize an exception for the HTTP response body."""
    tb = "".join(traceback.format_exception(type(exc), exc, exc.__traceback__))
    if getattr(exc, "remote_traceback", None):
        # keep the origin traceback when re-packaging across hops
        tb = exc.remote_traceback + "\n--- re-raised at next hop ---\n" + tb
    out = {
        "error_type": type(exc).__name__,
        "message": str(exc),
        "traceback": tb,
    }
    try:
        out["pickled"] = base64.b64encode(pickle.dumps(exc)).decode()
    except Exception:
        pass
    return out


def reconstruct_exception(payload: dict) -> BaseException:
    """Rebuild the remote exception: unpickle if possible, else match a
    builtin/registry class by name, else RemoteCallError."""
    remote_tb = payload.get("traceback", "")
    if "pickled" in payload:
        try:
            exc = pickle.loads(base64.b64decode(payload["pickled"]))
            exc.remote_traceback = remote_tb
            return exc
        except Exception:
            pass
    name = payload.get("error_type", "Exception")
    msg = payload.get("message", "")
    import builtins

    cls = REGISTRY.get(name) or getattr(builtins, name, None)
    if isinstance(cls, type) and issubclass(cls, BaseException):
        try:
            exc = cls(msg)
            exc.remote_traceback = remote_tb
            return exc
        except Exception:
            pass
    return RemoteCallError(name, msg, remote_tb)
