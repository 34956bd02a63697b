from kubetorch_amd.models.llama import (  # noqa: F401
    Llama,
    LlamaConfig,
    llama3_8b,
    llama_tiny,
)
