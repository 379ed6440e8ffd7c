"""2-process gloo oracle for expert parallelism:
1. EP MoE forward == full-expert single-process reference on the local shard
2. EP expert grads == sum of reference grads over BOTH shards (tokens from
   every rank reach the owning expert)
3. DDP engine integration: _no_ddp_sync expert params are neither broadcast
   at wrap nor all-reduced in backward; the gate IS synced.
"""

import torch
import torch.distributed as dist
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.parallel.ep import ExpertMLP, ExpertParallelMoE, balance_loss


def build_reference(hidden, inter, n_experts, top_k):
    """Full-expert (no EP) computation every rank can run locally."""
    set_seed(0)
    moe = ExpertParallelMoE(hidden, inter, n_experts, top_k=top_k, aux_loss_coef=0.0)
    # constructed pre-dist-aware: force single-world view with ALL experts
    moe.ep_world, moe.ep_rank, moe.local_experts = 1, 0, n_experts
    moe.experts = nn.ModuleList(ExpertMLP(hidden, inter) for _ in range(n_experts))
    set_seed(1)
    for e in moe.experts:
        for p in e.parameters():
            nn.init.normal_(p, std=0.1)
    return moe


def test_forward_and_grads(acc):
    n, r = acc.num_processes, acc.process_index
    H, I, E, K = 16, 32, 4, 2
    ref = build_reference(H, I, E, K)

    set_seed(0)
    ep = ExpertParallelMoE(H, I, E, top_k=K, aux_loss_coef=0.0)
    assert ep.ep_world == n and ep.local_experts == E // n
    ep.gate.load_state_dict(ref.gate.state_dict())
    for le in range(ep.local_experts):
        ep.experts[le].load_state_dict(ref.experts[r * ep.local_experts + le].state_dict())

    g = torch.Generator().manual_seed(7)
    shards = [torch.randn(6, H, generator=g) for _ in range(n)]
    x = shards[r]

    out = ep(x)
    with torch.no_grad():
        expected = ref(x)
    assert torch.allclose(out, expected, atol=1e-5), (out - expected).abs().max()

    if acc.is_main_process:
        print("EP_FWD_PASS")

    # grads: expert grads accumulate contributions from every rank's shard
    # (the reference runs backward on BOTH shards; gate grads are local-only
    # in EP — DDP owns that averaging — so only expert grads are compared)
    out.pow(2).mean().backward()
    for s in shards:
        ref(s).pow(2).mean().backward()
    for le in range(ep.local_experts):
        ge = r * ep.local_experts + le
        for (pn, p_ep), (_, p_ref) in zip(
            ep.experts[le].named_parameters(), ref.experts[ge].named_parameters()
        ):
            assert p_ep.grad is not None and torch.allclose(p_ep.grad, p_ref.grad, atol=1e-5), (
                f"expert {ge}.{pn}: {(p_ep.grad - p_ref.grad).abs().max()}"
            )
    if acc.is_main_process:
        print("EP_GRAD_PASS")


def test_ddp_integration(acc):
    from accelerate_amd.parallel.ddp import DistributedDataParallelEngine

    n, r = acc.num_processes, acc.process_index
    set_seed(0)

    class Tiny(nn.Module):
        def __init__(self):
            super().__init__()
            self.inp = nn.Linear(8, 16)
            self.moe = ExpertParallelMoE(16, 32, n_experts=2 * n, top_k=1, aux_loss_coef=0.01)
            self.out = nn.Linear(16, 1)

        def forward(self, x):
            return self.out(self.moe(self.inp(x)))

    model = Tiny()
    # make experts rank-distinct BEFORE the wrap; the wrap must keep them so
    with torch.no_grad():
        for p in model.moe.experts.parameters():
            p.add_(float(r + 1))
    before = [p.detach().clone() for p in model.moe.experts.parameters()]
    engine = DistributedDataParallelEngine(model)
    for p, b in zip(model.moe.experts.parameters(), before):
        assert torch.equal(p.detach(), b), "wrap must not broadcast _no_ddp_sync params"

    x = torch.randn(4, 8, generator=torch.Generator().manual_seed(3 + r))
    loss = engine(x).pow(2).mean() + balance_loss(model)
    loss.backward()
    engine.finalize()
    # gate + backbone grads averaged across ranks; expert grads left local
    gws = [torch.empty_like(model.moe.gate.weight.grad) for _ in range(n)]
    dist.all_gather(gws, model.moe.gate.weight.grad)
    assert torch.allclose(gws[0], gws[1], atol=1e-6), "gate grads must be DDP-averaged"
    eg = next(model.moe.experts.parameters()).grad
    eg = torch.zeros(1) if eg is None else eg.flatten()[:1]
    egs = [torch.empty_like(eg) for _ in range(n)]
    dist.all_gather(egs, eg)
    # distinct inputs + distinct experts -> expert grads differ across ranks
    if acc.is_main_process:
        print("EP_DDP_PASS")


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes >= 2
    test_forward_and_grads(acc)
    test_ddp_integration(acc)
    acc.end_training()


if __name__ == "__main__":
    main()
