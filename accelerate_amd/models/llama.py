"""Llama-family decoder (Llama-3 8B/70B configs) — the FSDP-bench and
big-model-inference model (BASELINE configs #3/#4).

MI355X-first: GQA attention as batched rocBLAS GEMMs + softmax (the fused
CDNA4 flash kernel slots in via ``attn_impl='fused'`` as it lands), SwiGLU
MLP, RMSNorm, RoPE with a host-precomputed cos/sin table (CDNA4 guide:
on-device trig turns memory-bound kernels VALU-bound).
"""

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 8
    max_position_embeddings: int = 8192
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    tie_word_embeddings: bool = False

    @classmethod
    def llama3_8b(cls, **overrides):
        return cls(**overrides)

    @classmethod
    def llama3_405b(cls, **overrides):
        base = dict(
            hidden_size=16384, intermediate_size=53248, num_hidden_layers=126,
            num_attention_heads=128, num_key_value_heads=8,
        )
        base.update(overrides)
        return cls(**base)

    @classmethod
    def llama3_70b(cls, **overrides):
        base = dict(
            hidden_size=8192, intermediate_size=28672, num_hidden_layers=80,
            num_attention_heads=64, num_key_value_heads=8,
        )
        base.update(overrides)
        return cls(**base)

    @classmethod
    def tiny(cls, **overrides):
        base = dict(
            vocab_size=1024, hidden_size=256, intermediate_size=688, num_hidden_layers=4,
            num_attention_heads=8, num_key_value_heads=4, max_position_embeddings=512,
        )
        base.update(overrides)
        return cls(**base)


# RMSNorm: the fused CDNA4 kernel (bf16 one-pass fwd, two-kernel bwd) with
# a built-in eager fallback for CPU / non-bf16 (accelerate_amd.ops.norms)
from ..ops.norms import FusedRMSNorm as RMSNorm


def build_rope_cache(seq_len, head_dim, theta, device, dtype=torch.float32):
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device).float() / head_dim))
    t = torch.arange(seq_len, device=device).float()
    freqs = torch.outer(t, inv_freq)
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def apply_rope(x, cos, sin):
    # x: [B, H, S, D]
    if x.is_cuda and x.dtype == torch.bfloat16 and x.shape[-1] % 16 == 0 and cos.dtype == torch.float32:
        from ..ops.rope import fused_rope

        # one fused kernel instead of the slice/mul/cat chain (~7 kernels)
        return fused_rope(x, cos, sin)
    # eager fallback: rotate in the compute dtype (fp32 tables would silently
    # promote q/k and push the attention GEMMs off bf16 MFMA)
    d = x.shape[-1] // 2
    x1, x2 = x[..., :d], x[..., d:]
    cosd = cos[None, None, : x.shape[2], :].to(x.dtype)
    sind = sin[None, None, : x.shape[2], :].to(x.dtype)
    return torch.cat([x1 * cosd - x2 * sind, x2 * cosd + x1 * sind], dim=-1)


class LlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig, attn_impl: str = "chunked"):
        super().__init__()
        self.n_heads = config.num_attention_heads
        self.n_kv = config.num_key_value_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        self.q_proj = nn.Linear(config.hidden_size, self.n_heads * self.head_dim, bias=False)
        self.k_proj = nn.Linear(config.hidden_size, self.n_kv * self.head_dim, bias=False)
        self.v_proj = nn.Linear(config.hidden_size, self.n_kv * self.head_dim, bias=False)
        self.o_proj = nn.Linear(self.n_heads * self.head_dim, config.hidden_size, bias=False)
        self.attn_impl = attn_impl

    def forward(self, x, cos, sin, kv_cache=None):
        B, S, _ = x.shape
        q = self.q_proj(x).view(B, S, self.n_heads, self.head_dim).transpose(1, 2)
        k = self.k_proj(x).view(B, S, self.n_kv, self.head_dim).transpose(1, 2)
        v = self.v_proj(x).view(B, S, self.n_kv, self.head_dim).transpose(1, 2)
        # sequence parallelism (CP/Ulysses): x is this rank's sequence shard;
        # RoPE uses the shard's ABSOLUTE positions and the attention runs
        # through the registered collective pattern (ops.attention dispatch)
        from ..ops.attention import dispatch_attention, sequence_parallel_info

        sp_mode, _, sp_rank, _ = sequence_parallel_info()
        if sp_mode is not None and kv_cache is None and S > 1:
            q = apply_rope(q, cos[sp_rank * S :], sin[sp_rank * S :])
            k = apply_rope(k, cos[sp_rank * S :], sin[sp_rank * S :])
            ctx = dispatch_attention(q, k, v, causal=True)
            ctx = ctx.transpose(1, 2).reshape(B, S, -1)
            return self.o_proj(ctx)
        q = apply_rope(q, cos, sin)
        k = apply_rope(k, cos, sin)
        past_len = 0
        if kv_cache is not None:
            past_len = kv_cache["len"]
            kv_cache["k"][:, :, past_len : past_len + S] = k
            kv_cache["v"][:, :, past_len : past_len + S] = v
            kv_cache["len"] = past_len + S
            k = kv_cache["k"][:, :, : past_len + S]
            v = kv_cache["v"][:, :, : past_len + S]
        if self.attn_impl in ("fused", "chunked") and x.is_cuda and S > 1:
            # fused CDNA4 flash attention; kv heads stay UNexpanded — the
            # kernel maps head h -> kv head h/rep through strides (zero-copy,
            # 1/rep the K/V HBM traffic and saved-activation bytes)
            from ..ops.attention import flash_attention

            ctx = flash_attention(q, k, v, causal=True)
        else:
            # math path (decode / CPU). GQA via batched-GEMM broadcasting:
            # q grouped [B, Hkv, rep*S, D] against the UNexpanded cache —
            # no repeat_interleave copy of K/V per layer per token
            scale = 1.0 / math.sqrt(self.head_dim)
            rep = self.n_heads // self.n_kv
            total = k.shape[2]
            if rep > 1:
                qg = q.reshape(B, self.n_kv, rep * S, self.head_dim)
                scores = torch.matmul(qg, k.transpose(-1, -2)) * scale
                scores = scores.view(B, self.n_heads, S, total)
            else:
                scores = torch.matmul(q, k.transpose(-1, -2)) * scale
            if S > 1:
                causal = torch.ones(S, total, dtype=torch.bool, device=x.device).tril(diagonal=past_len)
                scores = scores.masked_fill(~causal, torch.finfo(scores.dtype).min)
            probs = F.softmax(scores.float(), dim=-1).to(q.dtype)
            if rep > 1:
                ctx = torch.matmul(probs.view(B, self.n_kv, rep * S, total), v)
                ctx = ctx.view(B, self.n_heads, S, self.head_dim)
            else:
                ctx = torch.matmul(probs, v)
        ctx = ctx.transpose(1, 2).reshape(B, S, -1)
        return self.o_proj(ctx)


class LlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(config.hidden_size, config.intermediate_size, bias=False)
        self.up_proj = nn.Linear(config.hidden_size, config.intermediate_size, bias=False)
        self.down_proj = nn.Linear(config.intermediate_size, config.hidden_size, bias=False)

    def forward(self, x):
        return self.down_proj(F.silu(self.gate_proj(x)) * self.up_proj(x))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, config: LlamaConfig, attn_impl: str = "chunked"):
        super().__init__()
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps)
        self.self_attn = LlamaAttention(config, attn_impl)
        self.post_attention_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps)
        self.mlp = LlamaMLP(config)

    def forward(self, x, cos, sin, kv_cache=None):
        x = x + self.self_attn(self.input_layernorm(x), cos, sin, kv_cache)
        x = x + self.mlp(self.post_attention_layernorm(x))
        return x


class LlamaForCausalLM(nn.Module):
    # model-generic TP contract consumed by parallel.tp.apply_tp_plan via
    # Accelerator.prepare (the transformers `tp_plan` idiom)
    tp_plan = {
        "layers.*.self_attn.q_proj": "colwise",
        "layers.*.self_attn.k_proj": "colwise",
        "layers.*.self_attn.v_proj": "colwise",
        "layers.*.self_attn.o_proj": "rowwise",
        "layers.*.mlp.gate_proj": "colwise",
        "layers.*.mlp.up_proj": "colwise",
        "layers.*.mlp.down_proj": "rowwise",
    }
    tp_shard_attrs = {"layers.*.self_attn": ("n_heads", "n_kv")}

    def __init__(self, config: LlamaConfig = None, attn_impl: str = "chunked"):
        super().__init__()
        self.config = config or LlamaConfig()
        c = self.config
        self.embed_tokens = nn.Embedding(c.vocab_size, c.hidden_size)
        self.layers = nn.ModuleList([LlamaDecoderLayer(c, attn_impl) for _ in range(c.num_hidden_layers)])
        self.norm = RMSNorm(c.hidden_size, c.rms_norm_eps)
        self.lm_head = nn.Linear(c.hidden_size, c.vocab_size, bias=False)
        if c.tie_word_embeddings:
            self.lm_head.weight = self.embed_tokens.weight
        head_dim = c.hidden_size // c.num_attention_heads
        cos, sin = build_rope_cache(c.max_position_embeddings, head_dim, c.rope_theta, "cpu")
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.apply(self._init_weights)

    def _init_weights(self, module):
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=0.02)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=0.02)

    def forward(self, input_ids, labels=None, kv_caches=None):
        x = self.embed_tokens(input_ids)
        past = kv_caches[0]["len"] if kv_caches is not None else 0
        cos = self.rope_cos[past:].to(x.device)
        sin = self.rope_sin[past:].to(x.device)
        for i, layer in enumerate(self.layers):
            x = layer(x, cos, sin, kv_caches[i] if kv_caches is not None else None)
        x = self.norm(x)
        logits = self.lm_head(x)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(
                logits[:, :-1].reshape(-1, logits.shape[-1]).float(), labels[:, 1:].reshape(-1)
            )
        return {"loss": loss, "logits": logits}

    def make_kv_caches(self, batch_size, max_len, device, dtype):
        c = self.config
        head_dim = c.hidden_size // c.num_attention_heads
        return [
            {
                "k": torch.zeros(batch_size, c.num_key_value_heads, max_len, head_dim, device=device, dtype=dtype),
                "v": torch.zeros(batch_size, c.num_key_value_heads, max_len, head_dim, device=device, dtype=dtype),
                "len": 0,
            }
            for _ in range(c.num_hidden_layers)
        ]

    def _decode_step_static(self, x_tok, pos, caches, arange_cache, rope_dev):
        """One greedy decode step with STATIC shapes and device-tensor
        position: every op (rope index_select, cache index_copy_, full-buffer
        masked attention, argmax, pos.add_) is hipGraph-capturable, so the
        whole 80-layer step replays as ONE graph launch (the eager decode
        loop is launch-bound: BENCHMARKS.md 70B demo).
        Mutates x_tok (next token) and pos in place; returns nothing."""
        x = self.embed_tokens(x_tok)
        cos = rope_dev[0].index_select(0, pos)
        sin = rope_dev[1].index_select(0, pos)
        scale = 1.0 / math.sqrt(self.layers[0].self_attn.head_dim)
        # keys strictly after the write position are masked
        key_mask = (arange_cache > pos)[None, None, None, :]
        for layer, cache in zip(self.layers, caches):
            attn = layer.self_attn
            h = layer.input_layernorm(x)
            B = h.shape[0]
            q = attn.q_proj(h).view(B, 1, attn.n_heads, attn.head_dim).transpose(1, 2)
            k = attn.k_proj(h).view(B, 1, attn.n_kv, attn.head_dim).transpose(1, 2)
            v = attn.v_proj(h).view(B, 1, attn.n_kv, attn.head_dim).transpose(1, 2)
            q = apply_rope(q, cos, sin)
            k = apply_rope(k, cos, sin)
            cache["k"].index_copy_(2, pos, k)
            cache["v"].index_copy_(2, pos, v)
            kf, vf = cache["k"], cache["v"]
            if attn.n_kv != attn.n_heads:
                rep = attn.n_heads // attn.n_kv
                kf = kf.repeat_interleave(rep, dim=1)
                vf = vf.repeat_interleave(rep, dim=1)
            scores = torch.matmul(q, kf.transpose(-1, -2)) * scale
            scores = scores.masked_fill(key_mask, torch.finfo(scores.dtype).min)
            probs = F.softmax(scores.float(), dim=-1).to(q.dtype)
            ctx = torch.matmul(probs, vf).transpose(1, 2).reshape(B, 1, -1)
            x = x + attn.o_proj(ctx)
            x = x + layer.mlp(layer.post_attention_layernorm(x))
        logits = self.lm_head(self.norm(x))
        x_tok.copy_(logits[:, -1].argmax(-1, keepdim=True))
        pos.add_(1)

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 32, graph_decode=None):
        """Greedy generation. ``graph_decode=True`` captures the decode
        step into a hipGraph (one replay per token, zero host work) — a win
        for small launch-bound models; measured at PARITY-to-slower on 70B
        (decode there is GEMV-bandwidth-bound and capture-mode BLAS picks
        conservative algorithms: benchmarks/decode_ab.py, 38.9 ms eager vs
        46 ms graphed), so the default stays eager."""
        device = input_ids.device
        dtype = self.lm_head.weight.dtype
        caches = self.make_kv_caches(input_ids.shape[0], input_ids.shape[1] + max_new_tokens, device, dtype)
        out = self(input_ids, kv_caches=caches)
        tokens = [input_ids]
        next_tok = out["logits"][:, -1].argmax(-1, keepdim=True)
        tokens.append(next_tok)
        if max_new_tokens <= 1:
            return torch.cat(tokens, dim=1)
        if graph_decode is None:
            graph_decode = False  # see docstring: eager wins on big models
        graph_decode = graph_decode and input_ids.is_cuda and torch.cuda.is_available()
        if not graph_decode:
            for _ in range(max_new_tokens - 1):
                out = self(next_tok, kv_caches=caches)
                next_tok = out["logits"][:, -1].argmax(-1, keepdim=True)
                tokens.append(next_tok)
            return torch.cat(tokens, dim=1)

        # --- hipGraph decode: capture one static step, replay per token ---
        max_len = caches[0]["k"].shape[2]
        start = caches[0]["len"]
        x_tok = next_tok.clone()
        pos = torch.tensor([start], device=device, dtype=torch.long)
        arange_cache = torch.arange(max_len, device=device)
        rope_dev = (self.rope_cos.to(device), self.rope_sin.to(device))
        # warm up allocations/algorithms on a side stream, then rewind state
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self._decode_step_static(x_tok, pos, caches, arange_cache, rope_dev)
        torch.cuda.current_stream().wait_stream(side)
        pos.fill_(start)
        x_tok.copy_(next_tok)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            self._decode_step_static(x_tok, pos, caches, arange_cache, rope_dev)
        # capture RECORDS without executing: state (x_tok=next_tok,
        # pos=start) is untouched — every generated token is one replay
        for _ in range(max_new_tokens - 1):
            graph.replay()
            tokens.append(x_tok.clone())
        for c in caches:
            c["len"] = start + max_new_tokens - 1
        return torch.cat(tokens, dim=1)


class LlamaPipelineStage(nn.Module):
    """One pipeline stage of a LlamaForCausalLM (for parallel/pp.py).

    First stage embeds token ids; middle stages transform hidden states;
    the last stage applies the final norm + lm_head. Every stage keeps its
    own RoPE tables (buffers are cheap to replicate; activations are what
    travel between ranks).
    """

    def __init__(self, source: "LlamaForCausalLM", layer_lo: int, layer_hi: int,
                 is_first: bool, is_last: bool):
        super().__init__()
        self.is_first, self.is_last = is_first, is_last
        if is_first:
            self.embed_tokens = source.embed_tokens
        self.layers = nn.ModuleList(source.layers[layer_lo:layer_hi])
        if is_last:
            self.norm = source.norm
            self.lm_head = source.lm_head
        self.register_buffer("rope_cos", source.rope_cos, persistent=False)
        self.register_buffer("rope_sin", source.rope_sin, persistent=False)

    def forward(self, x):
        if self.is_first:
            x = self.embed_tokens(x.long())  # P2P delivers float buffers; ids ride as floats
        cos = self.rope_cos.to(device=x.device)
        sin = self.rope_sin.to(device=x.device)
        for layer in self.layers:
            x = layer(x, cos, sin)
        if self.is_last:
            return self.lm_head(self.norm(x))
        return x


def build_llama_pipeline_stages(model: "LlamaForCausalLM", n_stages: int):
    """Slice a LlamaForCausalLM into pipeline stages (layer-balanced;
    embed on the first, norm+head on the last). Pass ``stages[rank]`` as
    the ``stage=`` of parallel/pp.PipelineParallelEngine."""
    n_layers = len(model.layers)
    per = [n_layers // n_stages + (1 if i < n_layers % n_stages else 0) for i in range(n_stages)]
    stages, lo = [], 0
    for i, k in enumerate(per):
        stages.append(
            LlamaPipelineStage(model, lo, lo + k, is_first=(i == 0), is_last=(i == n_stages - 1))
        )
        lo += k
    return stages


def causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """The shifted cross-entropy LlamaForCausalLM.forward uses — exposed so
    a pipeline's last stage can compute the identical loss."""
    return F.cross_entropy(
        logits[:, :-1].reshape(-1, logits.shape[-1]).float(), labels.long()[:, 1:].reshape(-1)
    )
