"""2-process gloo oracle for join_uneven_inputs: rank 0 trains on 5 batches,
rank 1 on 3 — WITHOUT the join protocol the reducer's collective counts
would mismatch and deadlock; with it, the exhausted rank shadows the
stragglers' steps (zero contributions, full-world average: torch Join
semantics) and final weights are identical across ranks."""

import torch
import torch.distributed as dist
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed


def main():
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index
    assert n == 2

    set_seed(0)
    model = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    ref = nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 2))
    ref.load_state_dict(model.state_dict())
    model = acc.prepare_model(model)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)

    g = torch.Generator().manual_seed(5)
    X = [torch.randn(4, 8, generator=g) for _ in range(10)]  # per-(rank,step) batches
    n_steps = [5, 3]  # rank 0 runs 5 steps, rank 1 runs 3

    with acc.join_uneven_inputs([model]):
        for step in range(n_steps[r]):
            opt.zero_grad()
            loss = model(X[step * 2 + r]).pow(2).mean()
            acc.backward(loss)
            opt.step()

    # reference semantics: steps 0-2 average both ranks' grads; steps 3-4
    # average rank 0's grad with ZERO (divide by full world size = 2)
    for step in range(5):
        ref_opt.zero_grad()
        g0 = torch.autograd.grad(ref(X[step * 2 + 0]).pow(2).mean(), list(ref.parameters()))
        if step < 3:
            g1 = torch.autograd.grad(ref(X[step * 2 + 1]).pow(2).mean(), list(ref.parameters()))
        else:
            g1 = [torch.zeros_like(t) for t in g0]
        for p, a, b in zip(ref.parameters(), g0, g1):
            p.grad = (a + b) / 2
        ref_opt.step()

    for (pn, p), (_, rp) in zip(acc.unwrap_model(model).named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, rp, atol=1e-6), f"join parity mismatch {pn}: {(p - rp).abs().max()}"
    # both ranks end with identical weights
    w = acc.unwrap_model(model)[0].weight.detach()
    ws = [torch.empty_like(w) for _ in range(n)]
    dist.all_gather(ws, w)
    assert torch.equal(ws[0], ws[1])
    if acc.is_main_process:
        print("JOIN_UNEVEN_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
