from kubetorch_amd.models.llama import (  # noqa: F401
    KVCache,
    Llama,
    LlamaConfig,
    llama3_8b,
    llama_tiny,
)
from kubetorch_amd.models.serving import BatchedGenerator  # noqa: F401
