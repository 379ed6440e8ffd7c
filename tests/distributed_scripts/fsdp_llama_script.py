"""2-process gloo regression: the sharded engine must train a DICT-returning
model (LlamaForCausalLM) — backward unsharding is triggered by output-tensor
grad hooks, not module backward hooks (which never fire for dict outputs)."""

import os

import torch

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM
from accelerate_amd.parallel.fsdp import ShardedModel


def main():
    os.environ["ACCELERATE_USE_FSDP"] = "1"
    os.environ["FSDP_TRANSFORMER_CLS_TO_WRAP"] = "LlamaDecoderLayer"
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index
    set_seed(0)
    config = LlamaConfig.tiny(num_hidden_layers=2)
    model = LlamaForCausalLM(config)
    ref = LlamaForCausalLM(config)
    ref.load_state_dict(model.state_dict())
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    model, opt = acc.prepare(model, opt)
    assert isinstance(model, ShardedModel)

    g = torch.Generator().manual_seed(3)
    ids = torch.randint(0, 1024, (4, 32), generator=g)
    for step in range(3):
        opt.zero_grad()
        out = model(ids[r::n], labels=ids[r::n])
        acc.backward(out["loss"])
        opt.step()
        ref_opt.zero_grad()
        # reference: average of per-shard losses == our AVG grad reduction
        (0.5 * (ref(ids[0::n], labels=ids[0::n])["loss"] + ref(ids[1::n], labels=ids[1::n])["loss"])).backward()
        ref_opt.step()

    full = model.full_state_dict()
    bad = []
    for k, v in ref.state_dict().items():
        if not torch.allclose(full[k], v, atol=2e-4):
            bad.append((k, (full[k] - v).abs().max().item()))
    assert not bad, f"param mismatches: {bad[:5]}"
    if acc.is_main_process:
        print("FSDP_DICT_MODEL_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
