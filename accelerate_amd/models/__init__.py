from .bert import BertConfig, BertForSequenceClassification
from .gpt2 import GPT2Config, GPT2LMHeadModel
from .llama import LlamaConfig, LlamaForCausalLM
from .llama_moe import LlamaMoEConfig, LlamaMoEForCausalLM
from .t5 import T5Config, T5ForConditionalGeneration

__all__ = [
    "BertConfig",
    "BertForSequenceClassification",
    "GPT2Config",
    "GPT2LMHeadModel",
    "LlamaConfig",
    "LlamaForCausalLM",
    "LlamaMoEConfig",
    "LlamaMoEForCausalLM",
    "T5Config",
    "T5ForConditionalGeneration",
]
