from kubetorch_amd.parallel.ddp import FlatDDP, init_distributed  # noqa: F401
from kubetorch_amd.parallel.schedules import (  # noqa: F401
    constant_with_warmup,
    warmup_cosine,
    warmup_linear,
)
