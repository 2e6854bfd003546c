"""hetu_amd.models — GPT + Llama families (reference python/hetu/models)."""
from .gpt import (GPT_CONFIGS, GPTConfig, GPTLMHeadModel,  # noqa: F401
                  build_gpt_pipeline_stage, build_gpt_train_graph)
from .llama import (LLAMA_CONFIGS, LlamaConfig,  # noqa: F401
                    build_llama_pipeline_stage, build_llama_train_graph)
