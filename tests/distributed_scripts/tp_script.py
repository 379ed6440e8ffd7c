"""2-process gloo TP oracle: column→row MLP and a TP-sharded Llama layer
must match the unsharded reference (forward AND gradients)."""

import torch
import torch.distributed as dist
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.parallel.tp import ColumnParallelLinear, RowParallelLinear, tp_parallelize_llama


def test_mlp_parity(acc):
    set_seed(0)
    ref = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 16))
    col = ColumnParallelLinear.from_linear(ref[0])
    row = RowParallelLinear.from_linear(ref[2])
    x = torch.randn(4, 16, generator=torch.Generator().manual_seed(1), requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)

    y_ref = ref(x2)
    y = row(torch.relu(col(x)))
    assert torch.allclose(y, y_ref, atol=1e-5), (y - y_ref).abs().max()

    dout = torch.randn(4, 16, generator=torch.Generator().manual_seed(2))
    y.backward(dout)
    y_ref.backward(dout)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    # weight grads: column shard grad == the rank's slice of the full grad
    n, r = acc.num_processes, acc.process_index
    full_g = ref[0].weight.grad
    lo = r * col.out_per_rank
    assert torch.allclose(col.weight.grad, full_g[lo : lo + col.out_per_rank], atol=1e-5)
    full_g2 = ref[2].weight.grad
    lo2 = r * row.in_per_rank
    assert torch.allclose(row.weight.grad, full_g2[:, lo2 : lo2 + row.in_per_rank], atol=1e-5)
    if acc.is_main_process:
        print("TP_MLP_PASS")


def test_llama_parity(acc):
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM

    set_seed(0)
    config = LlamaConfig.tiny(num_hidden_layers=2)
    ref = LlamaForCausalLM(config).eval()
    set_seed(0)
    model = LlamaForCausalLM(config).eval()
    model.load_state_dict(ref.state_dict())
    tp_parallelize_llama(model)
    ids = torch.randint(0, 1024, (1, 12), generator=torch.Generator().manual_seed(3))
    with torch.no_grad():
        out_ref = ref(ids)["logits"]
        out = model(ids)["logits"]
    assert torch.allclose(out, out_ref, atol=1e-4), (out - out_ref).abs().max()
    if acc.is_main_process:
        print("TP_LLAMA_PASS")


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes == 2
    test_mlp_parity(acc)
    test_llama_parity(acc)
    acc.end_training()


if __name__ == "__main__":
    main()
