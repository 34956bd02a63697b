"""Pod-side image-setup interpreter: executes Image dockerfile steps (RUN /
ENV / COPY / SYNC / CMD) inside the running pod, re-running only the steps
that changed since the last setup — this is what keeps the hot loop at
seconds instead of an image rebuild. (Reference parity:
serving/http_server.py:510-832 cached_image_setup.)"""
import os
import shlex
import subprocess
import sys

_CACHED_STEPS = []
PIP_CMD_VAR = "KT_PIP_INSTALL_CMD"


def _default_pip():
    return f"{shlex.quote(sys.executable)} -m pip install"


def parse_steps(contents: str):
    steps = []
    for raw in (contents or "").splitlines():
        line = raw.strip()
        if not line or line.startswith("#"):
            continue
        kind, _, payload = line.partition(" ")
        steps.append((kind.upper(), payload.strip()))
    return steps


def _pip_requirements_satisfied(cmd):
    """True if `cmd` is a plain `pip install` whose requirements are all
    already importable at a matching version (reference parity: the
    pip-freeze diff that keeps warm reloads from re-resolving installed
    deps). Anything unparseable (urls, -r files, editable installs,
    options with values) conservatively returns False."""
    try:
        toks = shlex.split(cmd)
    except ValueError:
        return False
    if "install" not in toks or not any("pip" in t for t in toks):
        return False
    specs = []
    for t in toks[toks.index("install") + 1:]:
        if t.startswith("-"):
            return False  # options may change resolution; just run it
        if any(c in t for c in ":/@"):
            return False  # url / path / vcs spec
        specs.append(t)
    if not specs:
        return False
    from importlib import metadata

    for spec in specs:
        for op in ("==", ">=", "<=", "~=", "!=", ">", "<"):
            if op in spec:
                name, _, want = spec.partition(op)
                break
        else:
            name, op, want = spec, None, None
        name = name.strip().split("[")[0]
        try:
            have = metadata.version(name)
        except metadata.PackageNotFoundError:
            return False
        if op == "==" and have != want:
            return False
        if op in (">=", "<=", "~=", "!=", ">", "<") and op != "==":
            # only exact pins are verified offline; ranges -> run pip
            return False
    return True


def run_step(kind, payload, app_state=None):
    if kind == "FROM":
        return  # base image is fixed at pod creation
    if kind == "ENV":
        key, _, val = payload.partition("=")
        os.environ[key.strip()] = os.path.expandvars(val.strip())
        return
    if kind == "RUN":
        cmd = payload.replace(f"${PIP_CMD_VAR}",
                              os.environ.get(PIP_CMD_VAR, _default_pip()))
        cmd = os.path.expandvars(cmd)
        if _pip_requirements_satisfied(cmd):
            print(f"[image-setup] skip (already installed): {cmd}")
            return
        res = subprocess.run(["bash", "-lc", cmd], capture_output=True,
                             text=True)
        if res.returncode != 0:
            raise RuntimeError(
                f"image step failed ({cmd!r}): {res.stderr[-2000:]}")
        if res.stdout:
            print(res.stdout, end="")
        return
    if kind in ("COPY", "SYNC"):
        # content arrives via the data-store workdir sync; verify presence
        target = payload.split()[-1]
        if target and not os.path.exists(os.path.expandvars(target)):
            print(f"[image-setup] note: {kind} target {target} not present")
        return
    if kind == "CMD":
        proc = subprocess.Popen(["bash", "-lc", payload])
        if app_state is not None:
            old = app_state.get("app_proc")
            if old is not None and old.poll() is None:
                old.terminate()
            app_state["app_proc"] = proc
        return
    raise ValueError(f"unknown image step {kind!r}")


def cached_image_setup(contents: str, app_state=None):
    """Execute only the changed suffix of the step list vs the last setup.
    Returns the number of steps executed."""
    global _CACHED_STEPS
    steps = parse_steps(contents)
    common = 0
    for a, b in zip(_CACHED_STEPS, steps):
        if a != b:
            break
        common += 1
    to_run = steps[common:]
    for kind, payload in to_run:
        run_step(kind, payload, app_state=app_state)
    _CACHED_STEPS = steps
    return len(to_run)
