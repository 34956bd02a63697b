"""Pointer-based callable loading inside worker pods/processes.

A callable is addressed by (project_root, file_path, name) extracted on the
client (see client/pointers.py) and shipped as env vars — the code itself
arrives via the data-store sync, never pickled. (Reference parity:
serving/http_server.py:878-1137 load_callable / load_callable_from_env.)
"""
import base64
import importlib
import importlib.util
import os
import pickle
import sys
import threading

from kubetorch_amd import constants as C

_LOAD_LOCK = threading.Lock()
_CACHE = {}


def patch_sys_path(file_path, project_root):
    for p in (project_root, os.path.dirname(file_path) if file_path else None):
        if p and p not in sys.path:
            sys.path.insert(0, p)


def load_from_pointers(file_path, name, project_root=None, init_args=None,
                       module_type="fn", fresh=False):
    """Import `name` from the module at file_path. For cls, instantiate with
    init_args (b64-pickled (args, kwargs) or dict)."""
    key = (file_path, name, module_type)
    with _LOAD_LOCK:
        if not fresh and key in _CACHE:
            return _CACHE[key]
        patch_sys_path(file_path, project_root)
        mod_name = os.path.splitext(os.path.basename(file_path))[0]
        if fresh and mod_name in sys.modules:
            del sys.modules[mod_name]
        spec = importlib.util.spec_from_file_location(mod_name, file_path)
        mod = importlib.util.module_from_spec(spec)
        sys.modules[mod_name] = mod
        spec.loader.exec_module(mod)
        obj = getattr(mod, name)
        if module_type == "cls":
            args, kwargs = (), {}
            if init_args:
                if isinstance(init_args, str):
                    args, kwargs = pickle.loads(base64.b64decode(init_args))
                elif isinstance(init_args, dict):
                    kwargs = init_args
            obj = obj(*args, **kwargs)
        _CACHE[key] = obj
        return obj


def load_from_env(fresh=False):
    file_path = os.environ.get(C.ENV_FILE_PATH)
    name = os.environ.get(C.ENV_CALLABLE_NAME)
    if not file_path or not name:
        raise RuntimeError(
            f"no callable configured ({C.ENV_FILE_PATH}/{C.ENV_CALLABLE_NAME} unset)"
        )
    return load_from_pointers(
        file_path,
        name,
        project_root=os.environ.get(C.ENV_PROJECT_ROOT),
        init_args=os.environ.get(C.ENV_INIT_ARGS),
        module_type=os.environ.get(C.ENV_MODULE_TYPE, "fn"),
        fresh=fresh,
    )


def clear_cache():
    with _LOAD_LOCK:
        _CACHE.clear()
