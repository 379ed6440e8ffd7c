"""Distributed oracle (runs under torch.distributed.run, gloo, world 2):
- DDP training == single-process training on the concatenated batch
- grads are UNSYNCED inside no_sync/accumulate windows, synced at boundary
(the reference's test_sync.py oracle)."""

import torch
import torch.nn as nn
import torch.distributed as dist

from accelerate_amd import Accelerator, set_seed


def fresh_models():
    set_seed(42)
    model = nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 1))
    ref = nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 1))
    ref.load_state_dict(model.state_dict())
    return model, ref


def test_parity(acc):
    model, ref = fresh_models()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    model, opt = acc.prepare(model, opt)
    n, r = acc.num_processes, acc.process_index
    g = torch.Generator().manual_seed(7)
    X = torch.randn(32, 4, generator=g)
    Y = torch.randn(32, 1, generator=g)
    for step in range(4):
        xb, yb = X[step * 8 : (step + 1) * 8], Y[step * 8 : (step + 1) * 8]
        opt.zero_grad()
        loss = ((model(xb[r::n]) - yb[r::n]) ** 2).mean()
        acc.backward(loss)
        opt.step()
        ref_opt.zero_grad()
        ((ref(xb) - yb) ** 2).mean().backward()
        ref_opt.step()
    for (pn, p), (_, rp) in zip(acc.unwrap_model(model).named_parameters(), ref.named_parameters()):
        assert torch.allclose(p, rp, atol=1e-6), f"parity mismatch {pn}"
    if acc.is_main_process:
        print("PARITY_PASS")


def test_no_sync(acc):
    model, _ = fresh_models()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    model, opt = acc.prepare(model, opt)
    n, r = acc.num_processes, acc.process_index
    g = torch.Generator().manual_seed(9)
    X = torch.randn(16, 4, generator=g)
    Y = torch.randn(16, 1, generator=g)
    opt.zero_grad()
    with acc.no_sync(model):
        loss = ((model(X[r::n]) - Y[r::n]) ** 2).mean()
        acc.backward(loss)
    grad = acc.unwrap_model(model)[0].weight.grad.clone()
    gathered = [torch.empty_like(grad) for _ in range(n)]
    dist.all_gather(gathered, grad)
    assert not torch.allclose(gathered[0], gathered[1]), "grads must differ under no_sync"
    # boundary step syncs: second backward outside no_sync reduces ACCUMULATED grads
    loss = ((model(X[r::n]) - Y[r::n]) ** 2).mean()
    acc.backward(loss)
    grad = acc.unwrap_model(model)[0].weight.grad.clone()
    gathered = [torch.empty_like(grad) for _ in range(n)]
    dist.all_gather(gathered, grad)
    assert torch.allclose(gathered[0], gathered[1], atol=1e-7), "grads must match after sync step"
    if acc.is_main_process:
        print("NOSYNC_PASS")


def test_accumulate_gating(acc):
    acc.gradient_accumulation_steps = 2
    acc.step = 0
    model, _ = fresh_models()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    model, opt = acc.prepare(model, opt)
    n, r = acc.num_processes, acc.process_index
    X = torch.randn(16, 4)
    seen = []
    for i in range(4):
        with acc.accumulate(model):
            loss = model(X[r::n]).mean()
            acc.backward(loss)
            seen.append(acc.sync_gradients)
    assert seen == [False, True, False, True], f"accumulate gating wrong: {seen}"
    if acc.is_main_process:
        print("ACCUM_PASS")


def test_collectives(acc):
    n, r = acc.num_processes, acc.process_index
    t = torch.full((2,), float(r + 1))
    g = acc.gather(t)
    assert g.shape == (2 * n,)
    assert g[0].item() == 1.0 and g[-1].item() == float(n)
    red = acc.reduce(torch.full((1,), float(r + 1)), reduction="sum")
    assert red.item() == sum(range(1, n + 1))
    # ragged gather via pad
    ragged = torch.ones(r + 1)
    padded = acc.pad_across_processes(ragged)
    assert padded.shape[0] == n
    objs = acc.state
    from accelerate_amd.utils.operations import gather_object

    objects = gather_object([{"rank": r}])  # list input -> flat concat
    assert [o["rank"] for o in objects] == list(range(n))
    if acc.is_main_process:
        print("COLLECTIVES_PASS")


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes == 2, f"expected world 2, got {acc.num_processes}"
    test_parity(acc)
    test_no_sync(acc)
    test_collectives(acc)
    test_accumulate_gating(acc)
    acc.end_training()


if __name__ == "__main__":
    main()
