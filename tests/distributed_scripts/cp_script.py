"""2-process gloo context-parallel oracle: sequence-sharded attention
(all-gather KV + causal q_start offset) must match full-sequence attention,
forward AND input gradients; Llama CP logits match the unsharded model."""

import torch

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.ops.attention import flash_attention
from accelerate_amd.parallel.cp import apply_context_parallel_llama, context_parallel_attention, shard_sequence
from accelerate_amd.utils.operations import gather


def test_attention_parity(acc):
    n, r = acc.num_processes, acc.process_index
    set_seed(0)
    B, H, S, D = 1, 2, 32, 16
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    ref = flash_attention(q, k, v, causal=True, q_block=8, k_block=8)

    s = S // n
    ql = q[:, :, r * s : (r + 1) * s].clone().requires_grad_(True)
    kl = k[:, :, r * s : (r + 1) * s].clone().requires_grad_(True)
    vl = v[:, :, r * s : (r + 1) * s].clone().requires_grad_(True)
    out = context_parallel_attention(ql, kl, vl, causal=True)
    assert torch.allclose(out, ref[:, :, r * s : (r + 1) * s], atol=1e-5), (
        out - ref[:, :, r * s : (r + 1) * s]
    ).abs().max()

    # gradient parity vs full-sequence reference
    q2 = q.clone().requires_grad_(True)
    k2 = k.clone().requires_grad_(True)
    v2 = v.clone().requires_grad_(True)
    full = flash_attention(q2, k2, v2, causal=True, q_block=8, k_block=8)
    dout = torch.randn(B, H, S, D, generator=torch.Generator().manual_seed(5))
    full.backward(dout)
    out.backward(dout[:, :, r * s : (r + 1) * s])
    assert torch.allclose(ql.grad, q2.grad[:, :, r * s : (r + 1) * s], atol=1e-5)
    assert torch.allclose(kl.grad, k2.grad[:, :, r * s : (r + 1) * s], atol=1e-5), (
        kl.grad - k2.grad[:, :, r * s : (r + 1) * s]
    ).abs().max()
    assert torch.allclose(vl.grad, v2.grad[:, :, r * s : (r + 1) * s], atol=1e-5)
    if acc.is_main_process:
        print("CP_ATTN_PASS")


def test_llama_cp(acc):
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM

    n, r = acc.num_processes, acc.process_index
    set_seed(0)
    config = LlamaConfig.tiny(num_hidden_layers=2)
    ref = LlamaForCausalLM(config, attn_impl="chunked").eval()
    model = LlamaForCausalLM(config).eval()
    model.load_state_dict(ref.state_dict())
    apply_context_parallel_llama(model)

    ids = torch.randint(0, 1024, (1, 32), generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        full_logits = ref(ids)["logits"]
        local = model(shard_sequence(ids))["logits"]
    mine = full_logits[:, r * 16 : (r + 1) * 16]
    assert torch.allclose(local, mine, atol=1e-4), (local - mine).abs().max()
    if acc.is_main_process:
        print("CP_LLAMA_PASS")


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes == 2
    test_attention_parity(acc)
    test_llama_cp(acc)
    acc.end_training()


if __name__ == "__main__":
    main()
