"""GPU tests for the sharded engine (single GPU, world 1 semantics: bf16
compute params + fp32 master) and the Llama family."""

import pytest
import torch

gpu = pytest.mark.gpu


@gpu
def test_sharded_bf16_compute_fp32_master():
    from accelerate_amd.parallel.fsdp import ShardedModel
    from accelerate_amd.state import PartialState

    PartialState()
    import torch.nn as nn

    torch.manual_seed(0)
    base = nn.Sequential(nn.Linear(64, 256), nn.ReLU(), nn.Linear(256, 64))
    ref = nn.Sequential(nn.Linear(64, 256), nn.ReLU(), nn.Linear(256, 64)).cuda()
    ref.load_state_dict(base.state_dict())
    model = ShardedModel(base, min_num_params=100, compute_dtype=torch.bfloat16)
    x = torch.randn(8, 64, device="cuda")
    out = model(x.to(torch.bfloat16))
    ref_out = ref(x)
    assert (out.float() - ref_out).abs().max() < 0.1  # bf16 compute tolerance
    # master shards are fp32
    assert all(u.shard.dtype == torch.float32 for u in model.units)
    # params are bf16 views
    assert all(p.dtype == torch.bfloat16 for u in model.units for p in u.params)
    out.float().sum().backward()
    model.finalize_backward()
    assert all(u.shard.grad is not None and u.shard.grad.dtype == torch.float32 for u in model.units)


@gpu
def test_llama_tiny_train_step():
    from accelerate_amd import Accelerator
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
    from accelerate_amd.ops.optim import FusedAdamW

    torch.manual_seed(0)
    acc = Accelerator(mixed_precision="bf16")
    model = LlamaForCausalLM(LlamaConfig.tiny())
    opt = FusedAdamW(model.parameters(), lr=1e-4)
    model, opt = acc.prepare(model, opt)
    ids = torch.randint(0, 1024, (2, 64), device="cuda")
    out = model(ids, labels=ids)
    assert torch.isfinite(out["loss"])
    acc.backward(out["loss"])
    opt.step()
    torch.cuda.synchronize()


@gpu
def test_llama_generate_kv_cache():
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny()).cuda().to(torch.bfloat16).eval()
    ids = torch.randint(0, 1024, (1, 16), device="cuda")
    out = model.generate(ids, max_new_tokens=8)
    assert out.shape == (1, 24)
    # KV-cached generation must match full-context forward
    with torch.no_grad():
        full = model(out[:, :-1])["logits"][:, -1].argmax(-1)
    assert (full == out[:, -1]).all()
