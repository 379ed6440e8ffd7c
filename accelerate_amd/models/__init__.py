from .bert import BertConfig, BertForSequenceClassification
from .gpt2 import GPT2Config, GPT2LMHeadModel
from .llama import LlamaConfig, LlamaForCausalLM, build_llama_pipeline_stages, causal_lm_loss
from .llama_moe import LlamaMoEConfig, LlamaMoEForCausalLM
from .t5 import T5Config, T5ForConditionalGeneration

__all__ = [
    "BertConfig",
    "BertForSequenceClassification",
    "GPT2Config",
    "GPT2LMHeadModel",
    "LlamaConfig",
    "LlamaForCausalLM",
    "build_llama_pipeline_stages",
    "causal_lm_loss",
    "LlamaMoEConfig",
    "LlamaMoEForCausalLM",
    "T5Config",
    "T5ForConditionalGeneration",
]
