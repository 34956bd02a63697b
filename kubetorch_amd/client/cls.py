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


def cls(class_obj, init_args=None, name=None):
    """kt.cls(MyClass, init_args=((...), {...})) -> Cls proxy. The class is
    instantiated once per worker process with init_args at load time."""
    if init_args is not None:
        is_pair = (isinstance(init_args, tuple) and len(init_args) == 2
                   and isinstance(init_args[0], (tuple, list))
                   and isinstance(init_args[1], dict))
        if not is_pair:
            if isinstance(init_args, (list, tuple)):
                init_args = (tuple(init_args), {})
            else:
                init_args = ((), dict(init_args))
    return Cls(extract_pointers(class_obj), name=name, init_args=init_args)
