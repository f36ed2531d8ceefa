"""Model zoo: Llama-3 (flagship), Qwen2, Phi-3, GPT-2, OPT, Falcon, BERT, Mixtral MoE."""

from .llama import (LlamaConfig, LlamaForCausalLM, llama3_8b, llama3_1_8b, llama3_70b,
                    llama_tiny, llama_mini, phi3_mini, qwen2_7b, qwen2_mini,
                    mistral_7b)
from .gpt2 import GPT2Config, GPT2ForCausalLM, gpt2_small, gpt2_tiny, opt_125m, opt_mini
from .bert import (BertConfig, BertForPreTraining, BertModel, bert_base,
                   bert_large, bert_tiny)
from .falcon import (FalconConfig, FalconForCausalLM, falcon_7b,
                     falcon_mini, falcon_mini_gqa)
from .mixtral import (MixtralConfig, MixtralForCausalLM, mixtral_8x7b,
                      mixtral_tiny, mixtral_mini)

__all__ = [
    "LlamaConfig", "LlamaForCausalLM", "llama3_8b", "llama3_1_8b", "llama3_70b",
    "llama_tiny", "llama_mini", "phi3_mini", "qwen2_7b", "qwen2_mini", "mistral_7b", "GPT2Config", "GPT2ForCausalLM",
    "gpt2_small", "gpt2_tiny", "opt_125m", "opt_mini", "BertConfig", "BertModel",
    "BertForPreTraining", "bert_base", "bert_large", "bert_tiny",
    "FalconConfig", "FalconForCausalLM", "falcon_7b", "falcon_mini",
    "falcon_mini_gqa",
    "MixtralConfig", "MixtralForCausalLM",
    "mixtral_8x7b", "mixtral_tiny", "mixtral_mini",
]
