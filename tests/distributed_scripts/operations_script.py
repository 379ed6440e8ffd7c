"""3-process gloo oracle for the collective utilities with RANK-UNEVEN
shapes: pad_across_processes must equalize to the max length before a
gather can concatenate, nested structures recurse, and reduce averages."""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator
from accelerate_amd.utils.operations import (
    broadcast,
    gather,
    gather_object,
    pad_across_processes,
    reduce,
)


def main():
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index

    # uneven first-dim: rank r holds r+1 rows
    t = torch.full((r + 1, 3), float(r))
    padded = pad_across_processes(t, dim=0, pad_index=-1.0)
    assert padded.shape == (n, 3), padded.shape
    assert torch.equal(padded[: r + 1], t)
    assert (padded[r + 1 :] == -1.0).all()

    g = gather(padded)
    assert g.shape == (n * n, 3)
    for rr in range(n):
        blk = g[rr * n : rr * n + rr + 1]
        assert (blk == float(rr)).all()

    # nested structure recursion
    nested = {"a": t, "b": [torch.tensor([float(r)])]}
    pn = pad_across_processes(nested, dim=0, pad_index=0.0)
    assert pn["a"].shape == (n, 3)
    gn = gather(pn)
    assert gn["b"][0].shape == (n,)
    assert sorted(gn["b"][0].tolist()) == [float(i) for i in range(n)]

    # reduce mean and broadcast
    red = reduce(torch.tensor([float(r)]), reduction="mean")
    assert torch.allclose(red, torch.tensor([sum(range(n)) / n])), red
    b = broadcast(torch.tensor([float(r)]), from_process=1)
    assert torch.equal(b, torch.tensor([1.0])), b

    objs = gather_object([f"rank{r}"])
    assert objs == [f"rank{i}" for i in range(n)], objs

    dist.barrier()
    if acc.is_main_process:
        print("OPERATIONS_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
