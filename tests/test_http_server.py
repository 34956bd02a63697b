"""In-process pod-server tests: the REAL FastAPI app (lifespan included)
driven by fastapi.testclient with KT_* env pointed at local asset modules —
no controller, no subprocess pods. Reference parity model:
python_client/tests/test_http_server.py (runs the real app in-process)."""
import base64
import os
import pickle
import sys

import pytest

# spawns worker subprocesses via the supervisor: retry under CI load spikes
pytestmark = pytest.mark.flaky_retry

ASSETS = os.path.join(os.path.dirname(__file__), "assets", "summer")


@pytest.fixture(scope="module")
def client():
    os.environ["KT_FILE_PATH"] = os.path.join(ASSETS, "summer.py")
    os.environ["KT_PROJECT_ROOT"] = ASSETS
    os.environ["KT_CLS_OR_FN_NAME"] = "summer"
    os.environ["KT_MODULE_NAME"] = "summer"
    os.environ["KT_MODULE_TYPE"] = "fn"
    os.environ["KT_LAUNCH_ID"] = "unit-lid-1"
    os.environ["KT_ALLOWED_SERIALIZATION"] = "json,pickle"
    os.environ["KT_EXEC_TOKEN"] = "unit-exec-token"
    os.environ.pop("KT_CONTROLLER_URL", None)
    os.environ.pop("KT_DISTRIBUTED_CONFIG", None)
    for m in list(sys.modules):
        if m == "kubetorch_amd.serving.http_server":
            del sys.modules[m]
    from fastapi.testclient import TestClient

    from kubetorch_amd.serving import http_server

    with TestClient(http_server.app) as c:
        yield c
    sup = http_server.STATE.get("supervisor")
    if sup is not None:
        sup.cleanup()
        http_server.STATE["supervisor"] = None
    for k in ("KT_FILE_PATH", "KT_PROJECT_ROOT", "KT_CLS_OR_FN_NAME",
              "KT_MODULE_NAME", "KT_MODULE_TYPE", "KT_LAUNCH_ID",
              "KT_ALLOWED_SERIALIZATION", "KT_EXEC_TOKEN"):
        os.environ.pop(k, None)


def test_health_and_ready_gating(client):
    assert client.get("/health").json()["status"] == "ok"
    r = client.get("/ready")
    assert r.status_code == 200 and r.json()["ready"] is True
    # launch_id gating: mismatched id -> 503 until the reload lands
    r = client.get("/ready", params={"launch_id": "other-lid"})
    assert r.status_code == 503
    assert "unit-lid-1" in r.json()["reason"]
    r = client.get("/ready", params={"launch_id": "unit-lid-1"})
    assert r.status_code == 200


def test_call_json_and_pickle(client):
    r = client.post("/call/summer", json={"args": [2, 3], "kwargs": {}})
    assert r.status_code == 200, r.text
    assert r.json()["result"] == 5
    body = {"body": base64.b64encode(pickle.dumps(((7, 8), {}))).decode()}
    r = client.post("/call/summer", json=body,
                    headers={"X-Serialization": "pickle"})
    assert r.status_code == 200
    assert pickle.loads(base64.b64decode(r.json()["result"])) == 15


def test_call_wrong_name_404(client):
    r = client.post("/call/not_deployed", json={"args": [], "kwargs": {}})
    assert r.status_code == 404
    err = r.json()["error"]
    assert err["error_type"] == "KeyError"
    assert "summer" in err["message"]


def test_remote_exception_packaging(client):
    r = client.post("/call/summer", json={"args": ["x", 1], "kwargs": {}})
    assert r.status_code == 500
    err = r.json()["error"]
    assert err["error_type"] == "TypeError"
    assert "Traceback" in err["traceback"] or err["traceback"]


def test_exec_endpoint(client):
    r = client.post("/exec", json={"command": "echo hi && exit 0"},
                    headers={"X-KT-Exec-Token": "unit-exec-token"})
    assert r.status_code == 200
    assert r.json()["returncode"] == 0
    assert "hi" in r.json()["stdout"]


def test_exec_requires_token(client):
    # no token -> 403, wrong token -> 403 (route is RCE if left open)
    r = client.post("/exec", json={"command": "id"})
    assert r.status_code == 403
    r = client.post("/exec", json={"command": "id"},
                    headers={"X-KT-Exec-Token": "wrong"})
    assert r.status_code == 403


def test_exec_disabled_without_configured_token(client):
    tok = os.environ.pop("KT_EXEC_TOKEN")
    try:
        r = client.post("/exec", json={"command": "id"},
                        headers={"X-KT-Exec-Token": ""})
        assert r.status_code == 403
    finally:
        os.environ["KT_EXEC_TOKEN"] = tok


def test_pickle_serialization_default_denied(client):
    # with no allowlist env, only json is accepted (reference parity:
    # KT_ALLOWED_SERIALIZATION default "json")
    allowed = os.environ.pop("KT_ALLOWED_SERIALIZATION")
    try:
        body = {"body": base64.b64encode(pickle.dumps(((7, 8), {}))).decode()}
        r = client.post("/call/summer", json=body,
                        headers={"X-Serialization": "pickle"})
        assert r.status_code == 400
        assert "not allowed" in r.json()["error"]["message"]
        # json still works
        r = client.post("/call/summer", json={"args": [1, 2], "kwargs": {}})
        assert r.status_code == 200 and r.json()["result"] == 3
    finally:
        os.environ["KT_ALLOWED_SERIALIZATION"] = allowed


def test_metrics_and_logs_endpoints(client):
    m = client.get("/metrics")
    assert m.status_code == 200
    assert "kt_last_activity_timestamp" in m.text
    logs = client.get("/logs/tail", params={"limit": 10})
    assert logs.status_code == 200
    assert isinstance(logs.json()["entries"], list)


def test_metadata_echo_keeps_supervisor(client):
    """A registration-time metadata push that changes nothing must NOT
    recreate the supervisor (it would terminate the pool under an
    in-flight first call); a changed config or a launch_id must."""
    from kubetorch_amd.serving import http_server

    sup0 = http_server.get_supervisor()
    md = {"callable_name": "summer", "module_name": "summer",
          "file_path": os.environ["KT_FILE_PATH"],
          "project_root": os.environ["KT_PROJECT_ROOT"],
          "module_type": "fn"}
    http_server.do_reload(md, launch_id=None)       # metadata echo
    assert http_server.STATE["supervisor"] is sup0, \
        "unchanged metadata push recreated the supervisor"

    http_server.do_reload(md, launch_id="unit-lid-2")  # genuine reload
    sup1 = http_server.STATE["supervisor"]
    assert sup1 is not sup0
    assert http_server.STATE["launch_id"] == "unit-lid-2"

    md2 = dict(md, callable_name="slow_echo", module_name="slow_echo")
    http_server.do_reload(md2, launch_id=None)      # changed config
    assert http_server.STATE["supervisor"] is not sup1
    # restore for any later tests in this module
    http_server.do_reload(md, launch_id="unit-lid-1")
