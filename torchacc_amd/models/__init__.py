from .llama import (LlamaConfig, LlamaForCausalLM, llama_2_7b,  # noqa: F401
                    llama_2_70b, llama_tiny)
