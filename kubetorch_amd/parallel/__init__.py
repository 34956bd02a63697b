from kubetorch_amd.parallel.ddp import FlatDDP, init_distributed  # noqa: F401
from kubetorch_amd.parallel.elastic import (  # noqa: F401
    ElasticStepper,
    FileRendezvous,
    PeersRendezvous,
    reform_process_group,
)
from kubetorch_amd.parallel.schedules import (  # noqa: F401
    constant_with_warmup,
    warmup_cosine,
    warmup_linear,
)
