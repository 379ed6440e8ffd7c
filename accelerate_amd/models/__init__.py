from .bert import BertConfig, BertForSequenceClassification
from .gpt2 import GPT2Config, GPT2LMHeadModel
from .llama import LlamaConfig, LlamaForCausalLM

__all__ = [
    "BertConfig",
    "BertForSequenceClassification",
    "GPT2Config",
    "GPT2LMHeadModel",
    "LlamaConfig",
    "LlamaForCausalLM",
]
