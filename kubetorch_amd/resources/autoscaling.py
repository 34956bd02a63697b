"""Knative autoscaling annotation bundle (reference parity:
provisioning/autoscaling.py:13-106)."""
from dataclasses import dataclass


@dataclass
class AutoscalingConfig:
    target: int = 100
    metric: str = "concurrency"      # concurrency | rps | cpu | memory
    window: str = "60s"
    min_scale: int = 0
    max_scale: int = 0               # 0 = unlimited
    initial_scale: int = 1
    concurrency: int = 0             # container concurrency hard limit
    scale_down_delay: str = "0s"
    progress_deadline: str = "600s"

    def to_annotations(self):
        cls_ = ("hpa.autoscaling.knative.dev"
                if self.metric in ("cpu", "memory")
                else "kpa.autoscaling.knative.dev")
        ann = {
            "autoscaling.knative.dev/class": cls_,
            "autoscaling.knative.dev/metric": self.metric,
            "autoscaling.knative.dev/target": str(self.target),
            "autoscaling.knative.dev/window": self.window,
            "autoscaling.knative.dev/min-scale": str(self.min_scale),
            "autoscaling.knative.dev/initial-scale": str(self.initial_scale),
            "autoscaling.knative.dev/scale-down-delay": self.scale_down_delay,
            "serving.knative.dev/progress-deadline": self.progress_deadline,
        }
        if self.max_scale:
            ann["autoscaling.knative.dev/max-scale"] = str(self.max_scale)
        return ann
