"""Property-based tests (hypothesis) for the pure-logic helpers that sit on
security- or correctness-critical paths: k8s name sanitization, safe tar
extraction, manifest replica derivation, and decode-graph bucketing.

These complement the example-based tests: hypothesis drives each helper with
adversarial generated inputs (traversal paths, unicode, giant values) and
checks the *invariant*, not a sample.
"""
import io
import os
import tarfile

import pytest
from hypothesis import given, settings, strategies as st

from kubetorch_amd.client.module import sanitize_name
from kubetorch_amd.controller.drivers import desired_replicas
from kubetorch_amd.utils.tar import safe_extractall

SETTINGS = settings(max_examples=150, deadline=None)


# -- sanitize_name -----------------------------------------------------------

@SETTINGS
@given(st.text(min_size=0, max_size=200))
def test_sanitize_name_always_k8s_safe(name):
    out = sanitize_name(name)
    assert len(out) <= 63
    assert all(c.islower() or c.isdigit() or c == "-" for c in out)
    assert not out.startswith("-") and not out.endswith("-") or out == ""


@SETTINGS
@given(st.text(min_size=0, max_size=200))
def test_sanitize_name_idempotent(name):
    once = sanitize_name(name)
    assert sanitize_name(once) == once


# -- safe_extractall ---------------------------------------------------------

def _tar_with(name, data=b"x", typ=tarfile.REGTYPE, linkname=""):
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as t:
        info = tarfile.TarInfo(name)
        info.type = typ
        info.linkname = linkname
        if typ == tarfile.REGTYPE:
            info.size = len(data)
            t.addfile(info, io.BytesIO(data))
        else:
            t.addfile(info)
    buf.seek(0)
    return tarfile.open(fileobj=buf)


# member names built from path segments, some adversarial
_SEGMENT = st.one_of(
    st.text(alphabet="abcdefghij0123456789_-.", min_size=1, max_size=8),
    st.just(".."), st.just("."),
)


@SETTINGS
@given(st.lists(_SEGMENT, min_size=1, max_size=6))
def test_safe_extractall_never_escapes(tmp_path_factory, segments):
    dest = str(tmp_path_factory.mktemp("x"))
    name = "/".join(segments)
    tar = _tar_with(name)
    try:
        safe_extractall(tar, dest)
    except ValueError:
        return  # rejected: fine — the property is "never writes outside"
    # accepted: every file that landed must be inside dest
    root = os.path.realpath(dest)
    for dirpath, _dirs, files in os.walk(root):
        for f in files:
            p = os.path.realpath(os.path.join(dirpath, f))
            assert p.startswith(root + os.sep)


@SETTINGS
@given(st.lists(st.sampled_from(["..", "a", "b"]), min_size=1, max_size=5))
def test_safe_extractall_symlink_never_escapes(tmp_path_factory, segs):
    dest = str(tmp_path_factory.mktemp("x"))
    link = "/".join(segs)
    tar = _tar_with("lnk", typ=tarfile.SYMTYPE, linkname=link)
    try:
        safe_extractall(tar, dest)
    except ValueError:
        return
    target = os.path.realpath(os.path.join(dest, link))
    root = os.path.realpath(dest)
    assert target == root or target.startswith(root + os.sep)


def test_safe_extractall_rejects_absolute_and_device(tmp_path):
    with pytest.raises(ValueError):
        safe_extractall(_tar_with("/etc/evil"), str(tmp_path))
    with pytest.raises(ValueError):
        safe_extractall(_tar_with("dev", typ=tarfile.CHRTYPE), str(tmp_path))


@SETTINGS
@given(st.integers(min_value=0, max_value=0o7777))
def test_safe_extractall_strips_setuid(tmp_path_factory, mode):
    dest = str(tmp_path_factory.mktemp("x"))
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w") as t:
        info = tarfile.TarInfo("f")
        info.size = 1
        info.mode = mode
        t.addfile(info, io.BytesIO(b"x"))
    buf.seek(0)
    safe_extractall(tarfile.open(fileobj=buf), dest)
    got = os.stat(os.path.join(dest, "f")).st_mode & 0o7777
    assert got & 0o7000 == 0  # no setuid/setgid/sticky survives


# -- desired_replicas --------------------------------------------------------

@SETTINGS
@given(st.integers(min_value=0, max_value=512))
def test_desired_replicas_deployment(n):
    m = {"kind": "Deployment", "spec": {"replicas": n}}
    assert desired_replicas(m) == (n or 1)


@SETTINGS
@given(st.dictionaries(st.sampled_from(["Master", "Worker", "Chief"]),
                       st.integers(min_value=0, max_value=64),
                       min_size=1, max_size=3))
def test_desired_replicas_pytorchjob_sums(specs):
    m = {"kind": "PyTorchJob", "spec": {"pytorchReplicaSpecs": {
        k: {"replicas": v} for k, v in specs.items()}}}
    assert desired_replicas(m) == (sum(specs.values()) or 1)


@SETTINGS
@given(st.lists(st.integers(min_value=0, max_value=32), max_size=4))
def test_desired_replicas_raycluster(worker_groups):
    m = {"kind": "RayCluster", "spec": {
        "headGroupSpec": {},
        "workerGroupSpecs": [{"replicas": n} for n in worker_groups]}}
    assert desired_replicas(m) == 1 + sum(worker_groups)


def test_desired_replicas_defaults_to_one():
    assert desired_replicas({"kind": "Whatever", "spec": {}}) == 1
    assert desired_replicas({}) == 1


# -- decode-graph bucketing --------------------------------------------------

class _Bucketer:
    """Standalone shim exposing the engine's bucketing math (the method has
    no state beyond max_len)."""
    from kubetorch_amd.models.serving import BatchedGenerator as _G  # noqa
    _bucket_for = _G._bucket_for

    def __init__(self, max_len):
        self.max_len = max_len


@SETTINGS
@given(st.integers(min_value=1, max_value=100_000),
       st.integers(min_value=128, max_value=131_072))
def test_bucket_for_invariants(need, max_len):
    b = _Bucketer(max_len)
    L = b._bucket_for(need)
    assert L >= min(need, max_len)     # covers the need (up to cap)
    assert L <= max_len                # never exceeds the cache
    assert L >= 128
    if L < max_len:
        assert L & (L - 1) == 0        # power of two below the cap
        if L > 128:
            assert L // 2 < need       # minimal: next smaller bucket too small


@SETTINGS
@given(st.integers(min_value=1, max_value=4096))
def test_bucket_for_monotone(need):
    b = _Bucketer(8192)
    assert b._bucket_for(need) <= b._bucket_for(need + 1)


# -- autoscaling annotation round-trip ---------------------------------------

@SETTINGS
@given(st.integers(min_value=1, max_value=10_000),
       st.integers(min_value=0, max_value=64),
       st.integers(min_value=0, max_value=128),
       st.integers(min_value=1, max_value=3600))
def test_autoscale_spec_roundtrip(target, min_scale, max_scale, window_s):
    """AutoscalingConfig -> knative manifest annotations -> the controller's
    _autoscale_spec must agree on every scaling parameter."""
    from kubetorch_amd.controller.app import _autoscale_spec
    from kubetorch_amd.provisioning.manifests import build_knative_manifest
    from kubetorch_amd.resources.autoscaling import AutoscalingConfig

    cfg = AutoscalingConfig(target=target, min_scale=min_scale,
                            max_scale=max_scale, window=f"{window_s}s")
    m = build_knative_manifest("svc", "ns", "img", autoscaling=cfg)
    spec = _autoscale_spec(m)
    assert spec is not None
    assert spec["target"] == target
    assert spec["min"] == max(1, min_scale)   # controller floors at 1
    assert spec["max"] == max_scale           # 0 = unlimited
    assert spec["window"] == float(window_s)


@SETTINGS
@given(st.floats(min_value=0, max_value=10_000,
                 allow_nan=False, allow_infinity=False),
       st.sampled_from(["s", "m", "h", ""]))
def test_parse_duration(value, unit):
    from kubetorch_amd.controller.app import _parse_duration
    mult = {"s": 1, "m": 60, "h": 3600, "": 1}[unit]
    got = _parse_duration(f"{value}{unit}")
    assert abs(got - value * mult) < 1e-6


def test_parse_duration_garbage_defaults():
    from kubetorch_amd.controller.app import _parse_duration
    assert _parse_duration("", 7.0) == 7.0
    assert _parse_duration(None, 7.0) == 7.0
    assert _parse_duration("not-a-time", 7.0) == 7.0


def test_autoscale_spec_none_for_plain_deployment():
    from kubetorch_amd.controller.app import _autoscale_spec
    from kubetorch_amd.provisioning.manifests import build_deployment_manifest
    m = build_deployment_manifest("svc", "ns", "img")
    assert _autoscale_spec(m) is None


# -- Image dockerfile round-trip ---------------------------------------------

_PAYLOAD = st.text(
    alphabet=st.characters(blacklist_categories=("Cs", "Cc"),
                           blacklist_characters="\n\r"),
    min_size=1, max_size=60,
).map(str.strip).filter(lambda s: s and not s.startswith("#"))


@SETTINGS
@given(st.lists(st.tuples(st.sampled_from(["RUN", "ENV", "COPY", "CMD",
                                           "SYNC"]),
                          _PAYLOAD), max_size=8))
def test_image_dockerfile_roundtrip(steps):
    """Image.contents() -> from_dockerfile() preserves base image and the
    ordered step list (the reload differ depends on this being stable)."""
    from kubetorch_amd.resources.image import Image

    img = Image("rocm/pytorch:latest")
    img.steps = list(steps)
    back = Image.from_dockerfile(img.contents())
    assert back.image_id == "rocm/pytorch:latest"
    assert back.steps == list(steps)


def test_image_dockerfile_skips_comments_and_blanks():
    from kubetorch_amd.resources.image import Image
    img = Image.from_dockerfile(
        "# a comment\n\nFROM base\nRUN echo hi\n  \n# more\nENV A=1\n")
    assert img.image_id == "base"
    assert img.steps == [("RUN", "echo hi"), ("ENV", "A=1")]


# -- config layering ---------------------------------------------------------

@SETTINGS
@given(st.sampled_from(["namespace", "username", "image"]),
       st.text(alphabet="abcdef012", min_size=1, max_size=8),
       st.text(alphabet="abcdef012", min_size=1, max_size=8),
       st.booleans())
def test_config_env_beats_file(tmp_path_factory, key, envval, fileval,
                               set_env):
    """Layering invariant: env KT_* always wins over the config file; the
    file wins over defaults."""
    import yaml

    from kubetorch_amd.config import KTConfig

    path = str(tmp_path_factory.mktemp("cfg") / "config")
    with open(path, "w") as f:
        yaml.safe_dump({key: fileval}, f)
    envkey = "KT_" + key.upper()
    old = os.environ.pop(envkey, None)
    try:
        if set_env:
            os.environ[envkey] = envval
        cfg = KTConfig(path=path)
        cfg._cluster = {}  # no cluster fetch in a unit test
        assert cfg.get(key) == (envval if set_env else fileval)
    finally:
        if old is None:
            os.environ.pop(envkey, None)
        else:
            os.environ[envkey] = old


def test_config_cluster_below_file_above_defaults(tmp_path):
    from kubetorch_amd.config import KTConfig

    cfg = KTConfig(path=str(tmp_path / "missing"))
    cfg._cluster = {"namespace": "from-cluster"}
    old = os.environ.pop("KT_NAMESPACE", None)
    try:
        assert cfg.get("namespace") == "from-cluster"   # cluster > defaults
        cfg.set("namespace", "from-file")
        assert cfg.get("namespace") == "from-file"      # file > cluster
    finally:
        if old is not None:
            os.environ["KT_NAMESPACE"] = old
