"""Bundled distributed sanity script run by `accelerate-amd test`
(reference: test_utils/scripts/test_script.py — same checks, compact):
RNG sync, dataloader sharding coverage, collectives, distributed-vs-single
training parity, checkpoint round-trip. Asserts inside every worker.
"""

import tempfile

import torch
from torch.utils.data import DataLoader

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.state import PartialState
from accelerate_amd.test_utils.training import RegressionDataset, RegressionModel
from accelerate_amd.utils.operations import gather, reduce
from accelerate_amd.utils.random_utils import synchronize_rng_states


def rng_sync_check():
    state = PartialState()
    if state.use_distributed:
        synchronize_rng_states(["torch"])
        t = torch.randn(4)
        gathered = gather(t.to(state.device)).reshape(state.num_processes, -1).cpu()
        for r in range(1, state.num_processes):
            assert torch.equal(gathered[0], gathered[r]), "torch RNG not synchronized"
    state.print("RNG sync check: OK")


def dl_coverage_check(accelerator):
    state = PartialState()
    ds = RegressionDataset(length=64, seed=5)
    dl = DataLoader(ds, batch_size=4)
    dl = accelerator.prepare_data_loader(dl)
    seen = []
    for batch in dl:
        seen.append(batch["x"])
    seen = torch.cat(seen)
    all_seen = gather(seen.to(state.device)).cpu()
    expected = torch.tensor(ds.x)
    assert torch.isin(expected, all_seen).all(), "dataloader shards do not cover the dataset"
    state.print("DataLoader sharding check: OK")


def collectives_check(accelerator):
    state = PartialState()
    t = torch.full((2,), float(state.process_index + 1), device=state.device)
    g = gather(t)
    assert g.numel() == 2 * state.num_processes
    s = reduce(t, reduction="sum")
    expected = sum(range(1, state.num_processes + 1))
    assert s[0].item() == expected, (s, expected)
    state.print("Collectives check: OK")


def training_check(accelerator):
    """Distributed training must equal single-process training on the same
    global batches (reference oracle: test_script.py:449-620, ATOL 1e-6)."""
    state = PartialState()
    set_seed(42)
    ds = RegressionDataset(length=64, seed=6)

    # every rank trains an identical reference model on the GATHERED global
    # batch; the distributed model trains on its shard. DDP grad averaging
    # must make both end up with the same weights.
    set_seed(42)
    model = RegressionModel()
    ref_model = RegressionModel()
    ref_model.load_state_dict(model.state_dict())
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref_model.parameters(), lr=0.05)
    dl = accelerator.prepare_data_loader(DataLoader(ds, batch_size=8))
    model, opt = accelerator.prepare(model, opt)
    ref_model.to(state.device)
    for _ in range(2):
        for batch in dl:
            opt.zero_grad()
            loss = ((model(batch["x"]) - batch["y"]) ** 2).mean()
            accelerator.backward(loss)
            opt.step()

            gx, gy = gather(batch["x"]), gather(batch["y"])
            ref_opt.zero_grad()
            ref_loss = ((ref_model(gx) - gy) ** 2).mean()
            ref_loss.backward()
            ref_opt.step()

    unwrapped = accelerator.unwrap_model(model)
    assert torch.allclose(unwrapped.a, ref_model.a, atol=1e-5), (unwrapped.a, ref_model.a)
    assert torch.allclose(unwrapped.b, ref_model.b, atol=1e-5), (unwrapped.b, ref_model.b)
    state.print("Training parity check: OK")


def checkpoint_check(accelerator):
    from accelerate_amd.utils.operations import broadcast_object_list

    state = PartialState()
    model = RegressionModel(a=1, b=1)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    model, opt = accelerator.prepare(model, opt)
    # every rank must use the SAME directory (rank 0 creates it)
    d = [tempfile.mkdtemp() if state.is_main_process else None]
    broadcast_object_list(d)
    d = d[0]
    accelerator.save_state(d)
    with torch.no_grad():
        accelerator.unwrap_model(model).a.fill_(99.0)
    accelerator.wait_for_everyone()
    accelerator.load_state(d)
    assert accelerator.unwrap_model(model).a.item() == 1.0
    accelerator.free_memory()
    if state.is_main_process:
        import shutil

        shutil.rmtree(d, ignore_errors=True)
    state.print("Checkpoint round-trip check: OK")


def main():
    accelerator = Accelerator()
    state = PartialState()
    state.print(f"** Testing accelerate_amd on {state.num_processes} process(es), device {state.device} **")
    rng_sync_check()
    dl_coverage_check(accelerator)
    collectives_check(accelerator)
    training_check(accelerator)
    checkpoint_check(accelerator)
    state.print("All checks passed!")
    accelerator.end_training()


if __name__ == "__main__":
    main()
