"""CLI tests (typer runner) against the local driver."""
import os
import sys

from typer.testing import CliRunner

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "assets", "summer"))
os.environ["KT_LOCAL_MODE"] = "true"
os.environ["KT_USERNAME"] = "clitest"

import pytest  # noqa: E402
import kubetorch_amd as kt  # noqa: E402

pytestmark = [pytest.mark.flaky_retry, pytest.mark.minimal]
from kubetorch_amd.cli import app  # noqa: E402
from tests.assets.summer import summer as summer_mod  # noqa: E402

runner = CliRunner()


def test_check():
    r = runner.invoke(app, ["check"])
    assert r.exit_code == 0, r.output
    assert "controller ok" in r.output


def test_config_show():
    r = runner.invoke(app, ["config"])
    assert r.exit_code == 0
    assert "namespace" in r.output


def test_list_describe_call_teardown():
    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        r = runner.invoke(app, ["list"])
        assert f.name in r.output.replace("\n", "")
        r = runner.invoke(app, ["describe", f.name])
        assert r.exit_code == 0
        r = runner.invoke(app, ["call", f.name, "--args", "[2, 3]"])
        assert r.exit_code == 0, r.output
        assert "5" in r.output
    finally:
        r = runner.invoke(app, ["teardown", f.name])
        assert "deleted" in r.output


def test_secrets_volumes_manifests():
    r = runner.invoke(app, ["secrets", "create", "hf", "--values",
                            '{"HF_TOKEN": "x"}'])
    assert r.exit_code == 0 and "kt-secret-hf" in r.output
    r = runner.invoke(app, ["volumes", "create", "cache", "--size", "5Gi"])
    assert r.exit_code == 0 and "pvc cache created" in r.output
    r = runner.invoke(app, ["volumes", "list"])
    assert r.exit_code == 0 and "cache" in r.output
    r = runner.invoke(app, ["volumes", "delete", "cache"])
    assert r.exit_code == 0
    r = runner.invoke(app, ["secrets", "delete", "hf"])
    assert r.exit_code == 0


def test_run_bash_and_pip_helpers():
    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        out = f.run_bash("echo hello-from-pod")
        assert out["returncode"] == 0
        assert "hello-from-pod" in out["stdout"]
    finally:
        f.teardown()


def test_run_follow_streams_and_exits():
    r = runner.invoke(app, ["run", "echo from-kt-run && sleep 1",
                            "--name", "clirun", "--follow",
                            "--timeout", "120"])
    assert r.exit_code == 0, r.output[-500:]
    assert "from-kt-run" in r.output
    assert "app finished" in r.output


def test_apply_manifest_file(tmp_path):
    """kt apply: a raw Deployment manifest file becomes running pods
    through the controller."""
    import yaml

    manifest = kt.Compute(cpus=1).to_manifest("cli-applied")
    path = tmp_path / "dep.yaml"
    path.write_text(yaml.safe_dump(manifest))
    r = runner.invoke(app, ["apply", str(path)])
    assert r.exit_code == 0, r.output[-400:]
    from kubetorch_amd.controller.app import HUB

    try:
        assert len(HUB.driver.pods("cli-applied", "default")) == 1
    finally:
        runner.invoke(app, ["teardown", "cli-applied"])


def test_data_verbs_roundtrip(tmp_path):
    """kt put/ls/get/rm through the CLI against the local store."""
    src = tmp_path / "payload.txt"
    src.write_text("cli data plane")
    r = runner.invoke(app, ["put", "clitest/data1", str(src)])
    assert r.exit_code == 0, r.output
    r = runner.invoke(app, ["ls", "clitest"])
    assert r.exit_code == 0 and "data1" in r.output
    dest = tmp_path / "out"
    r = runner.invoke(app, ["get", "clitest/data1", str(dest)])
    assert r.exit_code == 0, r.output
    got = dest / "payload.txt"
    assert (got.read_text() if got.exists() else dest.read_text()) \
        == "cli data plane"
    r = runner.invoke(app, ["rm", "clitest/data1"])
    assert r.exit_code == 0
    r = runner.invoke(app, ["ls", "clitest"])
    assert "data1" not in r.output


def test_logs_and_workload_verbs():
    """kt logs tails a deployed fn's captured stdout; kt workload shows the
    registered module metadata."""
    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        f(2, 3)  # prints "summing 2+3" inside the pod
        r = runner.invoke(app, ["workload", f.name])
        assert r.exit_code == 0, r.output
        assert f.name in r.output
        # LogCapture batches (1 s flush): poll until the line lands
        import time
        deadline = time.time() + 15
        out = ""
        while time.time() < deadline and "summing" not in out:
            r = runner.invoke(app, ["logs", f.name])
            assert r.exit_code == 0, r.output
            out = r.output
            time.sleep(0.5)
        assert "summing" in out
    finally:
        f.teardown()


def test_bench_verb_through_dispatch_stack():
    """kt bench --model tiny on CPU: the flagship bench runs THROUGH the
    deploy -> SPMD fan-out path and prints the tokens/s JSON line."""
    import json
    import re

    r = runner.invoke(app, ["bench", "--model", "tiny", "--gpus", "0",
                            "--steps", "2", "--warmup", "1",
                            "--batch", "1", "--seq", "256"])
    assert r.exit_code == 0, r.output
    m = re.search(r"\{.*\}", r.output.replace("\n", ""))
    assert m, r.output
    result = json.loads(m.group(0))
    assert result["steps"] == 2
    assert result["value"] > 0
    assert result["metric"] == "tiny_ddp_tokens_per_sec"


def test_dashboard_local_lists_metrics_endpoints():
    f = kt.fn(summer_mod.summer).to(kt.Compute(cpus=1))
    try:
        r = runner.invoke(app, ["dashboard"])
        assert r.exit_code == 0, r.output
        assert "/metrics" in r.output
        assert f.name in r.output
    finally:
        f.teardown()
