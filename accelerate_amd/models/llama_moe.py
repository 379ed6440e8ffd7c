"""Mixtral-style sparse-MoE Llama: decoder layers whose MLP is an
expert-parallel top-k mixture (parallel/ep.py ExpertParallelMoE).

The model the EP dimension exists FOR: under DP+EP each rank holds
n_experts/ep_world experts, tokens travel by variable-split all-to-all,
the gate and attention stay data-parallel (`_no_ddp_sync` on expert
params keeps our DDP engine from averaging distinct experts).
``LlamaForCausalLM.forward`` semantics are preserved (dict with
loss/logits) so the sharded engine, bench and tests reuse the Llama
machinery unchanged.
"""

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..parallel.ep import ExpertParallelMoE, balance_loss
from .llama import LlamaAttention, LlamaConfig, RMSNorm, build_rope_cache


@dataclass
class LlamaMoEConfig(LlamaConfig):
    n_experts: int = 8
    top_k: int = 2
    aux_loss_coef: float = 0.01

    @classmethod
    def mixtral_8x7b_shape(cls, **overrides):
        """The Mixtral-8x7B geometry (random init; shapes only)."""
        d = dict(hidden_size=4096, intermediate_size=14336, num_hidden_layers=32,
                 num_attention_heads=32, num_key_value_heads=8, vocab_size=32000,
                 n_experts=8, top_k=2)
        d.update(overrides)
        return cls(**d)

    @classmethod
    def tiny_moe(cls, **overrides):
        d = dict(vocab_size=1024, hidden_size=64, intermediate_size=128,
                 num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
                 max_position_embeddings=128, n_experts=4, top_k=2)
        d.update(overrides)
        return cls(**d)


class LlamaMoEDecoderLayer(nn.Module):
    def __init__(self, config: LlamaMoEConfig, attn_impl: str = "chunked", ep_group=None):
        super().__init__()
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps)
        self.self_attn = LlamaAttention(config, attn_impl)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps)
        self.mlp = ExpertParallelMoE(
            config.hidden_size,
            config.intermediate_size,
            n_experts=config.n_experts,
            top_k=config.top_k,
            group=ep_group,
            aux_loss_coef=config.aux_loss_coef,
        )

    def forward(self, x, cos, sin, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, kv_cache)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaMoEForCausalLM(nn.Module):
    def __init__(self, config: LlamaMoEConfig = None, attn_impl: str = "chunked", ep_group=None):
        super().__init__()
        self.config = c = config or LlamaMoEConfig()
        self.embed_tokens = nn.Embedding(c.vocab_size, c.hidden_size)
        self.layers = nn.ModuleList(
            LlamaMoEDecoderLayer(c, attn_impl, ep_group) for _ in range(c.num_hidden_layers)
        )
        self.norm = RMSNorm(c.hidden_size, c.rms_norm_eps)
        self.lm_head = nn.Linear(c.hidden_size, c.vocab_size, bias=False)
        head_dim = c.hidden_size // c.num_attention_heads
        cos, sin = build_rope_cache(c.max_position_embeddings, head_dim, c.rope_theta, "cpu")
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

    def forward(self, input_ids, labels: Optional[torch.Tensor] = None):
        x = self.embed_tokens(input_ids)
        cos = self.rope_cos.to(x.device)
        sin = self.rope_sin.to(x.device)
        for layer in self.layers:
            x = layer(x, cos, sin)
        logits = self.lm_head(self.norm(x))
        out = {"logits": logits}
        if labels is not None:
            lm_loss = F.cross_entropy(
                logits[:, :-1].reshape(-1, logits.shape[-1]).float(), labels[:, 1:].reshape(-1)
            )
            out["aux_loss"] = balance_loss(self)
            out["loss"] = lm_loss + out["aux_loss"]
        return out
