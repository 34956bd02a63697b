"""Per-call workload configs: bundled knobs for log streaming, hardware
metric streaming, and remote debugging (reference parity: the
LoggingConfig / MetricsConfig / DebugConfig dataclasses of globals.py,
documented as docs/api/python/workload_configs.rst).

Usage — pass one or more to any remote call via ``kt_config=``::

    f(x, kt_config=kt.LoggingConfig(stream_logs=True))
    f(x, kt_config=[kt.MetricsConfig(stream_metrics=True),
                    kt.DebugConfig(debug=True)])

Each config contributes the same per-call options that the flat
``kt_stream_logs`` / ``kt_stream_metrics`` / ``kt_debug`` keywords set;
flat keywords win over a bundle when both are given.
"""
from dataclasses import dataclass


@dataclass
class LoggingConfig:
    """Per-call log streaming (client tails the pod's captured stdout for
    this request_id while the call runs).

    Fields mirror the reference constructor (globals.py LoggingConfig):
    `level` filters streamed lines client-side; `include_events` controls
    K8s launch-event streaming during `.to()`."""
    stream_logs: bool = True
    level: str = "info"
    include_events: bool = True

    def call_opts(self):
        return {"stream_logs": self.stream_logs}


@dataclass
class MetricsConfig:
    """Per-call hardware metric streaming (CPU/mem + amd-smi GPU util/VRAM
    sampled from the pod's /metrics while the call runs). `interval` is the
    seconds between metric samples (reference: MetricsConfig.interval)."""
    stream_metrics: bool = True
    interval: int = 30

    def call_opts(self):
        return {"stream_metrics": self.stream_metrics}


@dataclass
class DebugConfig:
    """Remote debugging: breakpoints hit during this call wait for
    ``kt debug <service>`` to attach over the pod's debug port
    (reference: DebugConfig mode/port; mode is always the pdb-over-socket
    transport here)."""
    debug: bool = True
    mode: str = "pdb"
    port: int = 5678

    def call_opts(self):
        return {"debug": self.debug}


def expand_config(cfg):
    """One config or a list/tuple of configs -> merged call-opts dict."""
    if cfg is None:
        return {}
    items = cfg if isinstance(cfg, (list, tuple)) else [cfg]
    opts = {}
    for c in items:
        if not hasattr(c, "call_opts"):
            raise TypeError(
                f"kt_config takes LoggingConfig/MetricsConfig/DebugConfig "
                f"(or a list of them), got {type(c).__name__}")
        opts.update(c.call_opts())
    return opts
