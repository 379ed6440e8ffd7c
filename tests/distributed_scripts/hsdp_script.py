"""4-process gloo HSDP oracle (2 shard groups × 2 replicas): hybrid-sharded
training must equal single-process training on the concatenated batch."""

import os

import torch
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.parallel.fsdp import ShardedModel


def make_model():
    set_seed(42)
    return nn.Sequential(nn.Linear(8, 64), nn.ReLU(), nn.Linear(64, 64), nn.ReLU(), nn.Linear(64, 1))


def main():
    os.environ["ACCELERATE_USE_FSDP"] = "1"
    os.environ["FSDP_MIN_NUM_PARAMS"] = "100"
    os.environ["FSDP_SHARDING_STRATEGY"] = "hybrid_shard"
    os.environ["FSDP_SHARD_GROUP_SIZE"] = "2"
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index
    assert n == 4, f"oracle needs 4 ranks, got {n}"

    model = make_model()
    ref = make_model()
    ref.load_state_dict(model.state_dict())
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    model, opt = acc.prepare(model, opt)
    assert isinstance(model, ShardedModel)
    assert model.world == 2, f"shard group should be 2, got {model.world}"
    assert model.replica_group is not None

    g = torch.Generator().manual_seed(3)
    X = torch.randn(64, 8, generator=g)
    Y = torch.randn(64, 1, generator=g)
    for step in range(3):
        xb = X[step * 16 : (step + 1) * 16]
        yb = Y[step * 16 : (step + 1) * 16]
        opt.zero_grad()
        loss = ((model(xb[r::n]) - yb[r::n]) ** 2).mean()
        acc.backward(loss)
        opt.step()
        ref_opt.zero_grad()
        ((ref(xb) - yb) ** 2).mean().backward()
        ref_opt.step()

    full = model.full_state_dict()
    for k, v in ref.state_dict().items():
        assert torch.allclose(full[k], v, atol=1e-5), f"HSDP mismatch {k}: {(full[k]-v).abs().max()}"
    if acc.is_main_process:
        print("HSDP_PARITY_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
