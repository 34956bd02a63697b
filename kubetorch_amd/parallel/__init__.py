from kubetorch_amd.parallel.ddp import FlatDDP, init_distributed  # noqa: F401
