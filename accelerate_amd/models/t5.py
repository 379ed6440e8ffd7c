"""T5 family (encoder-decoder, relative position bias, RMS norms).

Completes the bert/gpt2/t5/llama quartet the reference's Megatron plugin
parses (reference dataclasses.py:2842-3056); the reference's big-model
inference table features T0pp-11B (T5 family) — `T5Config.t5_11b()` is
that geometry for dispatch/memory planning.

Attention runs the math path: T5's additive relative-position bias is not
expressible in the fused flash kernel (bias operand) — a known kernel
extension, not silently dropped. Norms take the fused RMSNorm kernel.
"""

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedRMSNorm


@dataclass
class T5Config:
    vocab_size: int = 32128
    d_model: int = 512
    d_kv: int = 64
    d_ff: int = 2048
    num_layers: int = 6           # per stack (encoder AND decoder)
    num_heads: int = 8
    relative_attention_num_buckets: int = 32
    relative_attention_max_distance: int = 128
    layer_norm_epsilon: float = 1e-6
    decoder_start_token_id: int = 0

    @classmethod
    def t5_11b(cls, **overrides):
        """T0pp/T5-11B geometry (reference big-model table entry)."""
        d = dict(d_model=1024, d_kv=128, d_ff=65536, num_layers=24, num_heads=128)
        d.update(overrides)
        return cls(**d)

    @classmethod
    def tiny(cls, **overrides):
        d = dict(vocab_size=512, d_model=32, d_kv=8, d_ff=64, num_layers=2, num_heads=4)
        d.update(overrides)
        return cls(**d)


def relative_position_bucket(relative_position, bidirectional, num_buckets, max_distance):
    """T5's log-spaced distance bucketing (same contract as the published
    formulation: exact buckets near 0, log buckets out to max_distance)."""
    ret = torch.zeros_like(relative_position)
    n = -relative_position
    if bidirectional:
        num_buckets //= 2
        ret = ret + (n < 0).long() * num_buckets
        n = n.abs()
    else:
        n = torch.clamp(n, min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    large = max_exact + (
        torch.log(n.float().clamp(min=1) / max_exact)
        / math.log(max_distance / max_exact)
        * (num_buckets - max_exact)
    ).long()
    large = torch.clamp(large, max=num_buckets - 1)
    return ret + torch.where(is_small, n, large)


class T5Attention(nn.Module):
    def __init__(self, config: T5Config, has_relative_bias=False, causal=False):
        super().__init__()
        c = config
        self.n_heads, self.d_kv, self.causal = c.num_heads, c.d_kv, causal
        inner = c.num_heads * c.d_kv
        self.q = nn.Linear(c.d_model, inner, bias=False)
        self.k = nn.Linear(c.d_model, inner, bias=False)
        self.v = nn.Linear(c.d_model, inner, bias=False)
        self.o = nn.Linear(inner, c.d_model, bias=False)
        self.config = c
        self.relative_attention_bias = (
            nn.Embedding(c.relative_attention_num_buckets, c.num_heads) if has_relative_bias else None
        )

    def compute_bias(self, q_len, k_len, device):
        ctx = torch.arange(q_len, device=device)[:, None]
        mem = torch.arange(k_len, device=device)[None, :]
        buckets = relative_position_bucket(
            mem - ctx,
            bidirectional=not self.causal,
            num_buckets=self.config.relative_attention_num_buckets,
            max_distance=self.config.relative_attention_max_distance,
        )
        return self.relative_attention_bias(buckets).permute(2, 0, 1)[None]  # [1,H,q,k]

    def forward(self, x, kv=None, position_bias=None):
        B, S, _ = x.shape
        kv = x if kv is None else kv
        Sk = kv.shape[1]
        q = self.q(x).view(B, S, self.n_heads, self.d_kv).transpose(1, 2)
        k = self.k(kv).view(B, Sk, self.n_heads, self.d_kv).transpose(1, 2)
        v = self.v(kv).view(B, Sk, self.n_heads, self.d_kv).transpose(1, 2)
        # T5 convention: NO 1/sqrt(d) scaling (folded into init)
        scores = torch.matmul(q, k.transpose(-1, -2))
        if position_bias is not None:
            scores = scores + position_bias
        if self.causal and S > 1:
            mask = torch.ones(S, Sk, dtype=torch.bool, device=x.device).tril(Sk - S)
            scores = scores.masked_fill(~mask, torch.finfo(scores.dtype).min)
        probs = F.softmax(scores.float(), dim=-1).to(v.dtype)
        ctx = torch.matmul(probs, v).transpose(1, 2).reshape(B, S, -1)
        return self.o(ctx)


class T5FF(nn.Module):
    def __init__(self, config: T5Config):
        super().__init__()
        self.wi = nn.Linear(config.d_model, config.d_ff, bias=False)
        self.wo = nn.Linear(config.d_ff, config.d_model, bias=False)

    def forward(self, x):
        return self.wo(F.relu(self.wi(x)))


class T5Block(nn.Module):
    def __init__(self, config: T5Config, is_decoder, has_relative_bias):
        super().__init__()
        c = config
        self.is_decoder = is_decoder
        self.ln1 = FusedRMSNorm(c.d_model, c.layer_norm_epsilon)
        self.self_attn = T5Attention(c, has_relative_bias, causal=is_decoder)
        if is_decoder:
            self.ln_cross = FusedRMSNorm(c.d_model, c.layer_norm_epsilon)
            self.cross_attn = T5Attention(c, has_relative_bias=False, causal=False)
        self.ln2 = FusedRMSNorm(c.d_model, c.layer_norm_epsilon)
        self.ff = T5FF(c)

    def forward(self, x, enc=None, position_bias=None):
        x = x + self.self_attn(self.ln1(x), position_bias=position_bias)
        if self.is_decoder:
            x = x + self.cross_attn(self.ln_cross(x), kv=enc)
        return x + self.ff(self.ln2(x))


class T5Stack(nn.Module):
    def __init__(self, config: T5Config, is_decoder):
        super().__init__()
        self.blocks = nn.ModuleList(
            T5Block(config, is_decoder, has_relative_bias=(i == 0))
            for i in range(config.num_layers)
        )
        self.final_norm = FusedRMSNorm(config.d_model, config.layer_norm_epsilon)

    def forward(self, x, enc=None):
        # layer-0 owns the relative bias; later layers REUSE it (T5 sharing)
        bias = self.blocks[0].self_attn.compute_bias(x.shape[1], x.shape[1], x.device)
        for block in self.blocks:
            x = block(x, enc=enc, position_bias=bias)
        return self.final_norm(x)


class T5ForConditionalGeneration(nn.Module):
    def __init__(self, config: T5Config = None):
        super().__init__()
        self.config = c = config or T5Config()
        self.shared = nn.Embedding(c.vocab_size, c.d_model)
        self.encoder = T5Stack(c, is_decoder=False)
        self.decoder = T5Stack(c, is_decoder=True)
        self.lm_head = nn.Linear(c.d_model, c.vocab_size, bias=False)
        self.lm_head.weight = self.shared.weight
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)

    def _shift_right(self, labels):
        shifted = labels.new_full(labels.shape, self.config.decoder_start_token_id)
        shifted[:, 1:] = labels[:, :-1]
        return shifted

    def forward(self, input_ids, decoder_input_ids=None, labels: Optional[torch.Tensor] = None):
        if decoder_input_ids is None:
            if labels is None:
                raise ValueError("need decoder_input_ids or labels")
            decoder_input_ids = self._shift_right(labels)
        enc = self.encoder(self.shared(input_ids))
        dec = self.decoder(self.shared(decoder_input_ids), enc=enc)
        logits = self.lm_head(dec)
        out = {"logits": logits}
        if labels is not None:
            out["loss"] = F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]).float(), labels.reshape(-1)
            )
        return out

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 16):
        enc = self.encoder(self.shared(input_ids))
        dec_ids = input_ids.new_full((input_ids.shape[0], 1), self.config.decoder_start_token_id)
        for _ in range(max_new_tokens):
            dec = self.decoder(self.shared(dec_ids), enc=enc)
            nxt = self.lm_head(dec[:, -1:]).argmax(-1)
            dec_ids = torch.cat([dec_ids, nxt], dim=1)
        return dec_ids
