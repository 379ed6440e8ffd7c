"""BERT-base for sequence classification — the headline-benchmark model
(BASELINE.json: nlp_example BERT-base DDP bf16; the reference drives
transformers' bert-base-cased, examples/nlp_example.py:58).

Written MI355X-first: GEMMs go through rocBLAS/hipBLASLt (torch.matmul /
nn.Linear); attention is explicit batched-GEMM + softmax shaped so autocast
bf16 keeps everything on MFMA-backed library GEMMs. Hot elementwise ops are
left to torch-ROCm's HIP kernels; fused CDNA4 kernels (attention, LayerNorm)
slot in via accelerate_amd.ops as they land.
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedDropoutAddLayerNorm


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    hidden_dropout_prob: float = 0.1
    attention_probs_dropout_prob: float = 0.1
    layer_norm_eps: float = 1e-12
    num_labels: int = 2
    initializer_range: float = 0.02

    @classmethod
    def bert_base(cls, **overrides):
        return cls(**overrides)

    @classmethod
    def bert_large(cls, **overrides):
        base = dict(hidden_size=1024, num_hidden_layers=24, num_attention_heads=16, intermediate_size=4096)
        base.update(overrides)
        return cls(**base)


class BertEmbeddings(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.word_embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, config.hidden_size)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, config.hidden_size)
        self.LayerNorm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.register_buffer(
            "position_ids", torch.arange(config.max_position_embeddings).unsqueeze(0), persistent=False
        )

    def forward(self, input_ids, token_type_ids=None):
        seq_len = input_ids.shape[1]
        pos_ids = self.position_ids[:, :seq_len]
        emb = self.word_embeddings(input_ids) + self.position_embeddings(pos_ids)
        if token_type_ids is not None:
            emb = emb + self.token_type_embeddings(token_type_ids)
        else:
            emb = emb + self.token_type_embeddings.weight[0]
        return self.dropout(self.LayerNorm(emb))


class BertSelfAttention(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        self.qkv = nn.Linear(config.hidden_size, 3 * config.hidden_size)
        self.out = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout_p = config.attention_probs_dropout_prob

    def forward(self, hidden, attention_mask=None):
        B, S, H = hidden.shape
        qkv = self.qkv(hidden).view(B, S, 3, self.num_heads, self.head_dim).permute(2, 0, 3, 1, 4)
        q, k, v = qkv[0], qkv[1], qkv[2]
        # batched rocBLAS GEMMs + softmax; S is small for the headline config
        scores = torch.matmul(q, k.transpose(-1, -2)) * (1.0 / math.sqrt(self.head_dim))
        if attention_mask is not None:
            scores = scores + attention_mask.to(scores.dtype)
        probs = F.softmax(scores.float(), dim=-1).to(v.dtype)
        if self.training and self.dropout_p > 0:
            probs = F.dropout(probs, p=self.dropout_p)
        ctx = torch.matmul(probs, v)
        ctx = ctx.transpose(1, 2).reshape(B, S, H)
        return self.out(ctx)


class BertLayer(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.attention = BertSelfAttention(config)
        # both residual junctions run dropout+add+LayerNorm as ONE fused
        # kernel pair on GPU (ops/norms.py); params match nn.LayerNorm
        self.attn_norm = FusedDropoutAddLayerNorm(
            config.hidden_size, eps=config.layer_norm_eps, p=config.hidden_dropout_prob
        )
        self.ffn_in = nn.Linear(config.hidden_size, config.intermediate_size)
        self.ffn_out = nn.Linear(config.intermediate_size, config.hidden_size)
        self.ffn_norm = FusedDropoutAddLayerNorm(
            config.hidden_size, eps=config.layer_norm_eps, p=config.hidden_dropout_prob
        )

    def forward(self, hidden, attention_mask=None):
        attn = self.attention(hidden, attention_mask)
        hidden = self.attn_norm(hidden, attn)
        ffn = self.ffn_out(F.gelu(self.ffn_in(hidden), approximate="tanh"))
        hidden = self.ffn_norm(hidden, ffn)
        return hidden


class BertModel(nn.Module):
    def __init__(self, config: BertConfig):
        super().__init__()
        self.config = config
        self.embeddings = BertEmbeddings(config)
        self.layers = nn.ModuleList([BertLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None):
        extended_mask = None
        if attention_mask is not None:
            # [B, S] 1/0 mask -> additive [B, 1, 1, S] in the compute dtype
            dtype = self.embeddings.word_embeddings.weight.dtype
            if dtype not in (torch.float32, torch.bfloat16, torch.float16):
                dtype = torch.float32
            extended_mask = (1.0 - attention_mask[:, None, None, :].to(device=input_ids.device, dtype=dtype)) * torch.finfo(
                dtype
            ).min
        hidden = self.embeddings(input_ids, token_type_ids)
        for layer in self.layers:
            hidden = layer(hidden, extended_mask)
        pooled = torch.tanh(self.pooler(hidden[:, 0]))
        return hidden, pooled


class BertForSequenceClassification(nn.Module):
    def __init__(self, config: BertConfig = None):
        super().__init__()
        self.config = config or BertConfig()
        self.bert = BertModel(self.config)
        self.dropout = nn.Dropout(self.config.hidden_dropout_prob)
        self.classifier = nn.Linear(self.config.hidden_size, self.config.num_labels)
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, nn.LayerNorm):
            module.bias.data.zero_()
            module.weight.data.fill_(1.0)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None, labels=None):
        _, pooled = self.bert(input_ids, attention_mask, token_type_ids)
        logits = self.classifier(self.dropout(pooled))
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float(), labels)
        return {"loss": loss, "logits": logits}
