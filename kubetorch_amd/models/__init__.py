from kubetorch_amd.models.llama import (  # noqa: F401
    KVCache,
    Llama,
    LlamaConfig,
    llama2_7b,
    llama3_8b,
    llama3_70b,
    llama_tiny,
)
from kubetorch_amd.models.moe import MoEMLP, convert_to_moe  # noqa: F401
from kubetorch_amd.models.serving import BatchedGenerator  # noqa: F401
