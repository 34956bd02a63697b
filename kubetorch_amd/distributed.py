"""User-facing in-pod distribution helpers (`kt.distributed`).

Reference parity: kubetorch.distributed (pod_ips + quorum wait). The
implementations live in serving/discovery.py; this module is the public
namespace.
"""
from kubetorch_amd.serving.discovery import (  # noqa: F401
    MembershipMonitor,
    pod_ips,
)
