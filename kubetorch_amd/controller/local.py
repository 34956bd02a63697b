"""In-process local controller for local-mode (no Kubernetes).

Starts the controller FastAPI app in a background uvicorn thread on a free
port, once per client process. Used by the LocalDriver dev story and the
end-to-end tests."""
import socket
import threading
import time

import httpx

_local = {"url": None, "server": None}
_lock = threading.Lock()


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def ensure_local_controller(timeout=20.0):
    """Start (once) and return the local controller's base URL."""
    with _lock:
        if _local["url"]:
            return _local["url"]
        import os

        import uvicorn

        os.environ.setdefault("KT_CONTROLLER_DRIVER", "local")
        from kubetorch_amd.controller.app import HUB, app

        port = _free_port()
        url = f"http://127.0.0.1:{port}"
        HUB.set_url(url)
        config = uvicorn.Config(app, host="127.0.0.1", port=port,
                                log_level="error")
        server = uvicorn.Server(config)
        t = threading.Thread(target=server.run, daemon=True)
        t.start()
        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                if httpx.get(url + "/health", timeout=1).status_code == 200:
                    _local["url"] = url
                    _local["server"] = server
                    return url
            except Exception:
                time.sleep(0.05)
        raise RuntimeError("local controller failed to start")


def shutdown_local_controller():
    with _lock:
        if _local["server"] is not None:
            from kubetorch_amd.controller.app import HUB

            if hasattr(HUB.driver, "teardown_all"):
                HUB.driver.teardown_all()
            _local["server"].should_exit = True
            _local["url"] = None
            _local["server"] = None
