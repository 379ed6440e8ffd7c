"""4-process gloo oracle: COMPOSED tensor parallel x data parallel through
ParallelismConfig groups — tp=2 shards the linears, dp_replicate=2 averages
grads across replicas via our DDP engine on the dp group. Updated weights
must match a single-process reference run exactly."""

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.parallel.ddp import DistributedDataParallelEngine
from accelerate_amd.parallel.tp import ColumnParallelLinear, RowParallelLinear


class RefNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.up = nn.Linear(8, 16)
        self.down = nn.Linear(16, 8)

    def forward(self, x):
        return self.down(F.gelu(self.up(x)))


class TPNet(nn.Module):
    def __init__(self, group):
        super().__init__()
        self.up = ColumnParallelLinear(8, 16, group=group)
        self.down = RowParallelLinear(16, 8, group=group)

    def forward(self, x):
        return self.down(F.gelu(self.up(x)))


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes == 4
    pc = ParallelismConfig(dp_replicate_size=2, tp_size=2)
    groups = pc.build_groups()
    me = pc.coords(acc.process_index)
    tp, dp = me["tp"], me["dp_replicate"]

    set_seed(0)
    ref = RefNet()
    model = TPNet(groups["tp"])
    with torch.no_grad():
        model.up.weight.copy_(ref.up.weight[tp * 8 : (tp + 1) * 8])
        model.up.bias.copy_(ref.up.bias[tp * 8 : (tp + 1) * 8])
        model.down.weight.copy_(ref.down.weight[:, tp * 8 : (tp + 1) * 8])
        if model.down.bias is not None:
            model.down.bias.copy_(ref.down.bias)

    engine = DistributedDataParallelEngine(model, process_group=groups["dp_replicate"])
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)

    g = torch.Generator().manual_seed(5)
    X = torch.randn(8, 8, generator=g)  # dp replica r trains on X[r::2]
    for _ in range(2):
        opt.zero_grad()
        loss = engine(X[dp::2]).pow(2).mean()
        loss.backward()
        engine.finalize()
        opt.step()
        ref_opt.zero_grad()
        (0.5 * (ref(X[0::2]).pow(2).mean() + ref(X[1::2]).pow(2).mean())).backward()
        ref_opt.step()

    assert torch.allclose(model.up.weight, ref.up.weight[tp * 8 : (tp + 1) * 8], atol=1e-6)
    assert torch.allclose(model.down.weight, ref.down.weight[:, tp * 8 : (tp + 1) * 8], atol=1e-6)
    # replicas of the same tp shard agree bitwise across the dp group
    w = model.up.weight.detach()
    ws = [torch.empty_like(w) for _ in range(2)]
    dist.all_gather(ws, w, group=groups["dp_replicate"])
    assert torch.equal(ws[0], ws[1])
    if acc.is_main_process:
        print("TP_DP_COMPOSE_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
