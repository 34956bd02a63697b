"""Remote class proxy: any method becomes a remote call (reference parity:
resources/callables/cls/cls.py)."""
import functools

from kubetorch_amd.client.module import Module
from kubetorch_amd.client.pointers import extract_pointers


class Cls(Module):
    module_type = "cls"

    def __getattr__(self, item):
        if item.startswith("_") or item in self.__dict__:
            raise AttributeError(item)

        @functools.wraps(lambda: None)
        def remote_method(*args, **kwargs):
            opts = {}
            if "kt_config" in kwargs:
                from kubetorch_amd.workload_configs import expand_config

                opts.update(expand_config(kwargs.pop("kt_config")))
            for key in ("workers", "restart_procs", "stream_logs",
                        "stream_metrics", "timeout", "serialization",
                        "debug"):
                if f"kt_{key}" in kwargs:
                    opts[key] = kwargs.pop(f"kt_{key}")
            return self._call(args, kwargs, method=item, **opts)

        remote_method.__name__ = item
        return remote_method


def _norm_init_args(init_args):
    if init_args is None:
        return None
    is_pair = (isinstance(init_args, tuple) and len(init_args) == 2
               and isinstance(init_args[0], (tuple, list))
               and isinstance(init_args[1], dict))
    if is_pair:
        return init_args
    if isinstance(init_args, (list, tuple)):
        return (tuple(init_args), {})
    return ((), dict(init_args))


def cls(class_obj=None, init_args=None, name=None, remote_dir=None,
        remote_import_path=None):
    """kt.cls(MyClass, init_args=((...), {...})) -> Cls proxy. The class is
    instantiated once per worker process with init_args at load time.
    remote_dir mode mirrors kt.fn: serve a class already baked into the
    image via remote_import_path="pkg.module:Class"."""
    if remote_dir is not None:
        if not remote_import_path or ":" not in remote_import_path:
            raise ValueError(
                'remote_dir needs remote_import_path="pkg.module:Class"')
        import os

        module_path, _, symbol = remote_import_path.partition(":")
        rel = module_path.replace(".", "/") + ".py"
        pointers = {
            "name": symbol,
            "file_path": os.path.join(remote_dir, rel),
            "rel_path": rel,
            "project_root": remote_dir,
            "remote": True,
        }
        return Cls(pointers, name=name or symbol.lower(),
                   init_args=_norm_init_args(init_args))
    if class_obj is None:
        raise TypeError("kt.cls() needs a class (or remote_dir=...)")
    return Cls(extract_pointers(class_obj), name=name,
               init_args=_norm_init_args(init_args))
