from .llama import (LlamaConfig, LlamaForCausalLM, llama_2_7b,  # noqa: F401
                    llama_2_70b, llama_3_8b, llama_3_70b,
                    llama_tiny)
from .qwen2 import (Qwen2Config, Qwen2ForCausalLM, qwen2_7b,  # noqa: F401
                    qwen2_tiny)
