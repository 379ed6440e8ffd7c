"""GPT-2 family (decoder-only, learned positions, pre-LN) — MI355X-first.

Third model family beside BERT and Llama (the reference's Megatron plugin
parses bert/gpt2/t5/llama configs — dataclasses.py:2842-3056 — and its fp8
benchmarks quote GPT-2-large; this is the native training model for that
config). head_dim is 64 across the family, so attention rides the fused
CDNA4 flash kernel (`ops/attention.py`); LayerNorms take the fused bf16
kernel path on GPU.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedLayerNorm


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    max_position_embeddings: int = 1024
    layer_norm_eps: float = 1e-5
    initializer_range: float = 0.02

    @classmethod
    def gpt2_large(cls, **overrides):
        """774M: 36 layers x 1280 hidden x 20 heads (head_dim 64)."""
        return cls(hidden_size=1280, num_hidden_layers=36, num_attention_heads=20, **overrides)

    @classmethod
    def tiny(cls, **overrides):
        d = dict(vocab_size=1024, hidden_size=64, num_hidden_layers=2,
                 num_attention_heads=2, max_position_embeddings=128)
        d.update(overrides)
        return cls(**d)


class GPT2Attention(nn.Module):
    def __init__(self, config: GPT2Config):
        super().__init__()
        self.n_heads = config.num_attention_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        self.c_attn = nn.Linear(config.hidden_size, 3 * config.hidden_size)
        self.c_proj = nn.Linear(config.hidden_size, config.hidden_size)

    def forward(self, x, kv_cache=None):
        B, S, H = x.shape
        y = self.c_attn(x)
        # split by the PROJECTION's width, not the input width: under tensor
        # parallelism c_attn is column-sharded and yields [q_loc|k_loc|v_loc]
        q, k, v = y.split(y.shape[-1] // 3, dim=2)
        q = q.view(B, S, self.n_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.n_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.n_heads, self.head_dim).transpose(1, 2)
        from ..ops.attention import dispatch_attention, sequence_parallel_info

        sp_mode, _, _, _ = sequence_parallel_info()
        if sp_mode is not None and kv_cache is None and S > 1:
            # sequence shard path: positions were offset in the model forward
            ctx = dispatch_attention(q, k, v, causal=True)
            ctx = ctx.transpose(1, 2).reshape(B, S, -1)
            return self.c_proj(ctx)
        past = 0
        if kv_cache is not None:
            past = kv_cache["len"]
            kv_cache["k"][:, :, past : past + S] = k
            kv_cache["v"][:, :, past : past + S] = v
            kv_cache["len"] = past + S
            k = kv_cache["k"][:, :, : past + S]
            v = kv_cache["v"][:, :, : past + S]
        total = k.shape[2]
        if x.is_cuda and S > 1:
            from ..ops.attention import flash_attention

            ctx = flash_attention(q, k, v, causal=True)
        else:
            scale = 1.0 / math.sqrt(self.head_dim)
            scores = torch.matmul(q, k.transpose(-1, -2)) * scale
            if S > 1:
                mask = torch.ones(S, total, dtype=torch.bool, device=x.device).tril(past)
                scores = scores.masked_fill(~mask, torch.finfo(scores.dtype).min)
            probs = F.softmax(scores.float(), dim=-1).to(v.dtype)
            ctx = torch.matmul(probs, v)
        ctx = ctx.transpose(1, 2).reshape(B, S, -1)
        return self.c_proj(ctx)


class GPT2MLP(nn.Module):
    def __init__(self, config: GPT2Config):
        super().__init__()
        self.c_fc = nn.Linear(config.hidden_size, 4 * config.hidden_size)
        self.c_proj = nn.Linear(4 * config.hidden_size, config.hidden_size)

    def forward(self, x):
        return self.c_proj(F.gelu(self.c_fc(x), approximate="tanh"))


class GPT2Block(nn.Module):
    def __init__(self, config: GPT2Config):
        super().__init__()
        self.ln_1 = FusedLayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.attn = GPT2Attention(config)
        self.ln_2 = FusedLayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.mlp = GPT2MLP(config)

    def forward(self, x, kv_cache=None):
        x = x + self.attn(self.ln_1(x), kv_cache)
        x = x + self.mlp(self.ln_2(x))
        return x


class GPT2LMHeadModel(nn.Module):
    # model-generic TP contract consumed by parallel.tp.apply_tp_plan via
    # Accelerator.prepare; c_attn is a fused [q|k|v] projection
    tp_plan = {
        "h.*.attn.c_attn": "colwise_fused3",
        "h.*.attn.c_proj": "rowwise",
        "h.*.mlp.c_fc": "colwise",
        "h.*.mlp.c_proj": "rowwise",
    }
    tp_shard_attrs = {"h.*.attn": ("n_heads",)}

    def __init__(self, config: GPT2Config = None):
        super().__init__()
        c = self.config = config or GPT2Config()
        self.wte = nn.Embedding(c.vocab_size, c.hidden_size)
        self.wpe = nn.Embedding(c.max_position_embeddings, c.hidden_size)
        self.h = nn.ModuleList(GPT2Block(c) for _ in range(c.num_hidden_layers))
        self.ln_f = FusedLayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.lm_head = nn.Linear(c.hidden_size, c.vocab_size, bias=False)
        self.lm_head.weight = self.wte.weight  # tied (GPT-2 convention)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=self.config.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, labels=None, kv_caches=None):
        B, S = input_ids.shape
        past = kv_caches[0]["len"] if kv_caches is not None else 0
        from ..ops.attention import sequence_parallel_info

        sp_mode, _, sp_rank, _ = sequence_parallel_info()
        pos0 = past + (sp_rank * S if sp_mode is not None and kv_caches is None else 0)
        pos = torch.arange(pos0, pos0 + S, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos)[None]
        for i, block in enumerate(self.h):
            x = block(x, kv_caches[i] if kv_caches is not None else None)
        x = self.ln_f(x)
        logits = self.lm_head(x)
        out = {"logits": logits}
        if labels is not None:
            out["loss"] = F.cross_entropy(
                logits[:, :-1].reshape(-1, logits.shape[-1]).float(), labels[:, 1:].reshape(-1)
            )
        return out

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 32):
        """Greedy decode with per-layer KV caches (prefill once, then one
        position per step — each decode step reads the weights once)."""
        c = self.config
        B = input_ids.shape[0]
        max_len = min(c.max_position_embeddings, input_ids.shape[1] + max_new_tokens)
        dtype = self.wte.weight.dtype
        head_dim = c.hidden_size // c.num_attention_heads
        caches = [
            {
                "k": torch.zeros(B, c.num_attention_heads, max_len, head_dim,
                                 device=input_ids.device, dtype=dtype),
                "v": torch.zeros(B, c.num_attention_heads, max_len, head_dim,
                                 device=input_ids.device, dtype=dtype),
                "len": 0,
            }
            for _ in range(c.num_hidden_layers)
        ]
        ids = input_ids
        logits = self.forward(ids[:, -max_len:], kv_caches=caches)["logits"]
        for _ in range(max_new_tokens):
            nxt = logits[:, -1].argmax(-1, keepdim=True)
            ids = torch.cat([ids, nxt], dim=1)
            if caches[0]["len"] >= max_len:
                break
            logits = self.forward(nxt, kv_caches=caches)["logits"]
        return ids
