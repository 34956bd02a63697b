"""Decorators: @kt.compute / @kt.distribute / @kt.autoscale / @kt.async_ —
chainable config captured for `kt deploy` (reference parity:
resources/compute/decorators.py)."""
import functools


class PartialModule:
    """A function/class tagged with deployment config; `kt deploy` turns it
    into a deployed Fn/Cls. Calling it locally still runs the original."""

    def __init__(self, obj, compute_kwargs=None, distribute_args=None,
                 autoscale_kwargs=None, is_async=False):
        functools.update_wrapper(self, obj) if callable(obj) else None
        self.obj = obj
        self.compute_kwargs = compute_kwargs or {}
        self.distribute_args = distribute_args
        self.autoscale_kwargs = autoscale_kwargs
        self.is_async = is_async

    def __call__(self, *args, **kwargs):
        return self.obj(*args, **kwargs)

    def build_module(self):
        import inspect

        from kubetorch_amd.client.cls import cls as cls_factory
        from kubetorch_amd.client.fn import fn as fn_factory
        from kubetorch_amd.resources.compute import Compute

        compute = Compute(**self.compute_kwargs)
        if self.distribute_args:
            compute.distribute(*self.distribute_args[0], **self.distribute_args[1])
        if self.autoscale_kwargs is not None:
            compute.autoscale(**self.autoscale_kwargs)
        factory = cls_factory if inspect.isclass(self.obj) else fn_factory
        mod = factory(self.obj)
        mod.compute = compute
        return mod

    def deploy(self):
        mod = self.build_module()
        return mod.to(mod.compute)


def _wrap(obj, **updates):
    pm = obj if isinstance(obj, PartialModule) else PartialModule(obj)
    for k, v in updates.items():
        setattr(pm, k, v)
    return pm


def compute(**compute_kwargs):
    def deco(obj):
        return _wrap(obj, compute_kwargs=compute_kwargs)
    return deco


def distribute(*args, **kwargs):
    def deco(obj):
        return _wrap(obj, distribute_args=(args, kwargs))
    return deco


def autoscale(**kwargs):
    def deco(obj):
        return _wrap(obj, autoscale_kwargs=kwargs)
    return deco


def async_(obj):
    return _wrap(obj, is_async=True)
