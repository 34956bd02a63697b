"""Monarch supervisor launch contract (reference:
serving/monarch_supervisor.py): allocator service per pod, single
controller process, KT_MONARCH_HOSTS env with every pod's allocator
address. The monarch wheel has no ROCm build, so a stub allocator binary
stands in; absence raises with guidance."""
import os
import stat
import sys

import pytest

from kubetorch_amd.serving.supervisors import (MonarchSupervisor,
                                               supervisor_factory)

ASSETS = os.path.join(os.path.dirname(__file__), "assets", "summer")

pytestmark = pytest.mark.flaky_retry

STUB = """#!{python}
import socket, sys, time
port = int([a for a in sys.argv if a.startswith("--port=")][0].split("=")[1])
assert any(a == "--program=monarch_bootstrap" for a in sys.argv), sys.argv
s = socket.socket()
s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
s.bind(("127.0.0.1", port))
s.listen(1)
while True:
    time.sleep(1)
"""


@pytest.fixture()
def stub_allocator(tmp_path, monkeypatch):
    stub = tmp_path / "process_allocator"
    stub.write_text(STUB.format(python=sys.executable))
    stub.chmod(stub.stat().st_mode | stat.S_IEXEC)
    monkeypatch.setenv("KT_MONARCH_ALLOCATOR", str(stub))
    monkeypatch.setenv("KT_MONARCH_PORT", "26610")
    monkeypatch.setenv("KT_LOCAL_IPS", "127.0.0.1:41001,127.0.0.2:41001")
    monkeypatch.setenv("KT_NUM_WORKERS", "2")
    monkeypatch.setenv("KT_FILE_PATH", os.path.join(ASSETS, "summer.py"))
    monkeypatch.setenv("KT_PROJECT_ROOT", ASSETS)
    monkeypatch.setenv("KT_CLS_OR_FN_NAME", "read_env")
    monkeypatch.setenv("KT_MODULE_TYPE", "fn")
    return stub


def test_missing_allocator_raises_with_guidance(monkeypatch):
    monkeypatch.delenv("KT_MONARCH_ALLOCATOR", raising=False)
    monkeypatch.setenv("PATH", "/nonexistent")
    with pytest.raises(NotImplementedError, match="process_allocator"):
        supervisor_factory("monarch")


@pytest.mark.timeout(180)
def test_allocator_service_and_controller_env(stub_allocator):
    sup = supervisor_factory("monarch", num_workers=2)
    try:
        assert isinstance(sup, MonarchSupervisor)
        assert sup._alloc_proc.poll() is None  # allocator service live
        # the controller worker sees every pod's allocator address
        result = sup.call(args=("KT_MONARCH_HOSTS",), kwargs={})
        assert result == "127.0.0.1:26610,127.0.0.2:26610"
    finally:
        sup.cleanup()
    assert sup._alloc_proc.poll() is not None  # torn down


GCS_STUB = """#!{python}
import socket, time
s = socket.socket()
s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
s.bind(("127.0.0.1", {port}))
s.listen(1)
while True:
    time.sleep(1)
"""


@pytest.mark.timeout(180)
def test_ray_supervisor_launch_contract(tmp_path, monkeypatch):
    """RaySupervisor launch contract with a stub head command (no ray
    wheel on this image): starts KUBERAY_GEN_RAY_START_CMD, waits for GCS
    liveness, serves calls from one worker, tears the head down."""
    import socket as _socket

    from kubetorch_amd.serving.supervisors import RaySupervisor

    s = _socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    stub = tmp_path / "ray_head.py"
    stub.write_text(GCS_STUB.format(python=sys.executable, port=port))
    monkeypatch.setenv("KT_RAY_GCS_PORT", str(port))
    monkeypatch.setenv("KUBERAY_GEN_RAY_START_CMD",
                       f"{sys.executable} {stub}")
    monkeypatch.setenv("KT_FILE_PATH", os.path.join(ASSETS, "summer.py"))
    monkeypatch.setenv("KT_PROJECT_ROOT", ASSETS)
    monkeypatch.setenv("KT_CLS_OR_FN_NAME", "summer")
    monkeypatch.setenv("KT_MODULE_TYPE", "fn")
    sup = RaySupervisor()
    try:
        assert sup._ray_proc.poll() is None  # head process live
        assert sup.call(args=(3, 4)) == 7
    finally:
        sup.cleanup()
    assert sup._ray_proc.poll() is not None


@pytest.mark.timeout(120)
def test_ray_supervisor_head_crash_raises(tmp_path, monkeypatch):
    from kubetorch_amd.serving.supervisors import RaySupervisor

    monkeypatch.setenv("KT_RAY_GCS_PORT", "1")  # nothing will listen
    monkeypatch.setenv("KUBERAY_GEN_RAY_START_CMD", "exit 3")
    with pytest.raises(RuntimeError, match="exited during startup"):
        RaySupervisor()
