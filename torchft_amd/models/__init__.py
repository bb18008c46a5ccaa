from torchft_amd.models.llama import (
    LLAMA3_8B,
    LLAMA3_70B,
    LLAMA_DEBUG,
    Llama,
    LlamaConfig,
)

__all__ = ["Llama", "LlamaConfig", "LLAMA3_8B", "LLAMA3_70B", "LLAMA_DEBUG"]
