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


def fn(callable_obj=None, name=None, sync_dir=None, remote_dir=None,
       remote_import_path=None):
    """kt.fn(train) -> Fn proxy; deploy with .to(kt.Compute(...)).

    remote_dir mode (reference parity: fn(sync_dir/remote_dir/
    remote_import_path)): dispatch to a function whose code is already on
    the pod image — no client-side code sync. Pass
    ``remote_import_path="pkg.module:func"`` and ``remote_dir="/app"``;
    the pod imports /app/pkg/module.py and serves ``func``."""
    if sync_dir and remote_dir:
        raise ValueError("sync_dir and remote_dir are mutually exclusive")
    if remote_dir is not None:
        if not remote_import_path or ":" not in remote_import_path:
            raise ValueError(
                'remote_dir needs remote_import_path="pkg.module:func"')
        module_path, _, func = remote_import_path.partition(":")
        import os

        rel = module_path.replace(".", "/") + ".py"
        pointers = {
            "name": func,
            "file_path": os.path.join(remote_dir, rel),
            "rel_path": rel,
            "project_root": remote_dir,
            "remote": True,  # code lives on the image: skip workdir sync
        }
        return Fn(pointers, name=name or func)
    if callable_obj is None:
        raise TypeError("kt.fn() needs a callable (or remote_dir=...)")
    pointers = extract_pointers(callable_obj)
    if sync_dir:
        # sync THIS directory instead of the auto-detected project root
        # (reference: Module sync_dir — the fn's file must live under it)
        import os

        root = os.path.abspath(sync_dir)
        fp = os.path.abspath(pointers["file_path"])
        if not fp.startswith(root + os.sep):
            raise ValueError(
                f"{pointers['name']} at {fp} is not under sync_dir {root}")
        pointers["project_root"] = root
        pointers["rel_path"] = os.path.relpath(fp, root)
    return Fn(pointers, name=name)
