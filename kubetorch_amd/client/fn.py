"""Remote function proxy (reference parity: resources/callables/fn/fn.py)."""
from kubetorch_amd.client.module import Module
from kubetorch_amd.client.pointers import extract_pointers


class Fn(Module):
    module_type = "fn"

    def __call__(self, *args, **kwargs):
        opts = {}
        if "kt_config" in kwargs:
            from kubetorch_amd.workload_configs import expand_config

            opts.update(expand_config(kwargs.pop("kt_config")))
        for key in ("workers", "restart_procs", "stream_logs",
                    "stream_metrics", "timeout", "serialization", "debug"):
            if f"kt_{key}" in kwargs:
                opts[key] = kwargs.pop(f"kt_{key}")
        return self._call(args, kwargs, **opts)

    async def call_async(self, *args, **kwargs):
        import asyncio

        return await asyncio.to_thread(self.__call__, *args, **kwargs)


def fn(callable_obj, name=None):
    """kt.fn(train) -> Fn proxy; deploy with .to(kt.Compute(...))."""
    return Fn(extract_pointers(callable_obj), name=name)
