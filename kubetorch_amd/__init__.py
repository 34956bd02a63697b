"""kubetorch_amd — an MI355X-native serverless ML dispatch framework.

A from-scratch framework with the capabilities of run-house/kubetorch,
re-designed for AMD Instinct MI355X (gfx950): PyTorch-ROCm + hand-written
HIP/CDNA4 kernels for the compute path, RCCL over xGMI for collectives,
amd.com/gpu scheduling for the control plane.

Public API mirrors the reference's `kt` namespace (kt.fn, kt.cls,
kt.Compute, ...; reference: python_client/kubetorch/__init__.py).
Heavy client-side modules are imported lazily so that the training path
(ops/models/parallel) does not pull in the control-plane stack.
"""

__version__ = "0.1.0"

_LAZY = {
    "fn": ("kubetorch_amd.client.fn", "fn"),
    "cls": ("kubetorch_amd.client.cls", "cls"),
    "app": ("kubetorch_amd.client.app", "app"),
    "Fn": ("kubetorch_amd.client.fn", "Fn"),
    "Cls": ("kubetorch_amd.client.cls", "Cls"),
    "App": ("kubetorch_amd.client.app", "App"),
    "config": ("kubetorch_amd.config", "config"),
    "secret": ("kubetorch_amd.resources.secret", "secret_factory"),
    "Compute": ("kubetorch_amd.resources.compute", "Compute"),
    "Image": ("kubetorch_amd.resources.image", "Image"),
    "Volume": ("kubetorch_amd.resources.volume", "Volume"),
    "Secret": ("kubetorch_amd.resources.secret", "Secret"),
    "Endpoint": ("kubetorch_amd.resources.endpoint", "Endpoint"),
    "AutoscalingConfig": ("kubetorch_amd.resources.autoscaling", "AutoscalingConfig"),
    "LoggingConfig": ("kubetorch_amd.workload_configs", "LoggingConfig"),
    "MetricsConfig": ("kubetorch_amd.workload_configs", "MetricsConfig"),
    "DebugConfig": ("kubetorch_amd.workload_configs", "DebugConfig"),
    "images": ("kubetorch_amd.resources.images", None),
    "compute": ("kubetorch_amd.resources.decorators", "compute"),
    "distribute": ("kubetorch_amd.resources.decorators", "distribute"),
    "autoscale": ("kubetorch_amd.resources.decorators", "autoscale"),
    "async_": ("kubetorch_amd.resources.decorators", "async_"),
    "put": ("kubetorch_amd.data_store.commands", "put"),
    "get": ("kubetorch_amd.data_store.commands", "get"),
    "ls": ("kubetorch_amd.data_store.commands", "ls"),
    "rm": ("kubetorch_amd.data_store.commands", "rm"),
    "get_broadcast": ("kubetorch_amd.data_store.commands", "get_broadcast"),
    "BroadcastWindow": ("kubetorch_amd.data_store.types", "BroadcastWindow"),
    "pod_ips": ("kubetorch_amd.serving.discovery", "pod_ips"),
    "distributed": ("kubetorch_amd.distributed", None),
    # exception types users catch (reference exports these at top level)
    "KubetorchError": ("kubetorch_amd.exceptions", "KubetorchError"),
    "LaunchError": ("kubetorch_amd.exceptions", "LaunchError"),
    "ImagePullError": ("kubetorch_amd.exceptions", "ImagePullError"),
    "ResourceNotAvailableError": (
        "kubetorch_amd.exceptions", "ResourceNotAvailableError"),
    "PodTerminatedError": ("kubetorch_amd.exceptions", "PodTerminatedError"),
    "WorkerMembershipChanged": (
        "kubetorch_amd.exceptions", "WorkerMembershipChanged"),
    "RemoteCallError": ("kubetorch_amd.exceptions", "RemoteCallError"),
    "QuorumTimeout": ("kubetorch_amd.exceptions", "QuorumTimeout"),
    "deep_breakpoint": ("kubetorch_amd.serving.pdb_ws", "deep_breakpoint"),
    "secret_factory": ("kubetorch_amd.resources.secret", "secret_factory"),
    "globals": ("kubetorch_amd.globals", None),
    "ops": ("kubetorch_amd.ops", None),
    "models": ("kubetorch_amd.models", None),
    "parallel": ("kubetorch_amd.parallel", None),
}


def __getattr__(name):
    if name in _LAZY:
        import importlib
        import sys

        mod_name, attr = _LAZY[name]
        mod = importlib.import_module(mod_name)
        obj = mod if attr is None else getattr(mod, attr)
        # NB: use the module dict directly — importing the kubetorch_amd.globals
        # submodule shadows the globals() builtin in this namespace.
        sys.modules[__name__].__dict__[name] = obj
        return obj
    raise AttributeError(f"module 'kubetorch_amd' has no attribute {name!r}")


def __dir__():
    import sys

    return sorted(list(sys.modules[__name__].__dict__) + list(_LAZY))
