"""Shared pytest config for kubetorch_amd.

Registers the `gpu` marker (tests needing a real MI355X) and skips those
tests automatically when no GPU is present, mirroring the reference's
level system (reference: python_client/tests/conftest.py:27-41).
"""
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_addoption(parser):
    parser.addoption(
        "--level", default=None,
        choices=["unit", "minimal", "release", "gpu"],
        help="reference-parity test levels: unit = fast in-process tests "
             "only; minimal = + full-stack local-driver e2e; release = all "
             "CPU tests; gpu = only GPU-marked tests. Default: no level "
             "filtering (marker expressions like -m 'not gpu' still apply).")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")
    config.addinivalue_line(
        "markers",
        "minimal: full-stack e2e through the local driver (subprocess "
        "pods) — the reference's 'minimal' integration tier")
    config.addinivalue_line(
        "markers",
        "flaky_retry: rerun once on failure (multi-process e2e tests are "
        "sensitive to CI host load spikes)")


def pytest_runtest_protocol(item, nextitem):
    """Retry (twice) for tests marked flaky_retry: the local-driver e2e and
    spawn-based gloo tests trip timeouts when the shared CI host stalls; a
    failure there is re-run before being reported."""
    if "flaky_retry" not in item.keywords:
        return None
    from _pytest.runner import runtestprotocol

    reports = runtestprotocol(item, nextitem=nextitem, log=False)
    for _ in range(2):
        if not any(r.failed for r in reports):
            break
        reports_retry = runtestprotocol(item, nextitem=nextitem, log=False)
        if not any(r.failed for r in reports_retry):
            reports = reports_retry
            break
        reports = reports_retry
    for r in reports:
        item.ihook.pytest_runtest_logreport(report=r)
    return True


def pytest_collection_modifyitems(config, items):
    level = config.getoption("--level")
    if level:
        keep = []
        for item in items:
            is_gpu = "gpu" in item.keywords
            is_minimal = "minimal" in item.keywords
            if level == "gpu":
                ok = is_gpu
            elif level == "unit":
                ok = not is_gpu and not is_minimal
            elif level == "minimal":
                ok = not is_gpu
            else:  # release: everything CPU (gpu still gated below)
                ok = not is_gpu
            if ok:
                keep.append(item)
        items[:] = keep
    # GPU tests keep the old 600 s budget (first-touch kernel compiles,
    # pg-timeout waits in the elastic drills); the 300 s pytest.ini default
    # is sized for the CPU suite, whose slowest test is ~31 s
    for item in items:
        if "gpu" in item.keywords and item.get_closest_marker(
                "timeout") is None:
            item.add_marker(pytest.mark.timeout(600))
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
