"""Multi-process gloo oracle for pipeline-parallel TRAINING: an n-stage MLP
trained with GPipe microbatching must match a single-process run of the
same full model (same grads, same updated weights, same loss).
The reference raises NotImplementedError for PP training — this is a
capability beyond it."""

import torch
import torch.distributed as dist
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.parallel.pp import PipelineParallelEngine, split_into_stages


def run(acc, schedule):
    n, r = acc.num_processes, acc.process_index
    set_seed(0)
    full = nn.Sequential(
        nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 32), nn.Tanh(), nn.Linear(32, 4)
    )
    ref = nn.Sequential(*[nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 32), nn.Tanh(), nn.Linear(32, 4)])
    ref.load_state_dict(full.state_dict())

    engine = PipelineParallelEngine(model=full, num_microbatches=4, schedule=schedule)
    opt = torch.optim.SGD(engine.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    loss_fn = nn.MSELoss()

    g = torch.Generator().manual_seed(4)
    for step in range(3):
        X = torch.randn(8, 8, generator=g)
        T = torch.randn(8, 4, generator=g)
        opt.zero_grad()
        loss = engine.train_step(
            inputs=X if engine.is_first else None,
            targets=T if engine.is_last else None,
            loss_fn=loss_fn if engine.is_last else None,
        )
        opt.step()
        ref_opt.zero_grad()
        ref_loss = loss_fn(ref(X), T)
        ref_loss.backward()
        ref_opt.step()
        if engine.is_last:
            assert torch.allclose(loss, ref_loss, atol=1e-6), (loss, ref_loss)

    # each rank's stage params must equal the reference's matching slice
    stages = split_into_stages(ref, n)
    for p_eng, p_ref in zip(engine.stage.parameters(), stages[r].parameters()):
        assert torch.allclose(p_eng, p_ref, atol=1e-6), (p_eng - p_ref).abs().max()


def run_device(acc, schedule):
    """GPU variant: stage + P2P buffers on this rank's device over RCCL
    (the ADVICE round-1 device-placement fix); ref replicated per rank."""
    set_seed(0)
    dev = acc.device
    full = nn.Sequential(
        nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 32), nn.Tanh(), nn.Linear(32, 4)
    )
    ref = nn.Sequential(nn.Linear(8, 32), nn.Tanh(), nn.Linear(32, 32), nn.Tanh(), nn.Linear(32, 4))
    ref.load_state_dict(full.state_dict())
    ref = ref.to(dev)

    engine = PipelineParallelEngine(model=full, num_microbatches=4, schedule=schedule)
    assert next(engine.stage.parameters(), torch.empty(0, device=dev)).device.type == dev.type
    opt = torch.optim.SGD(engine.parameters(), lr=0.1)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    loss_fn = nn.MSELoss()
    g = torch.Generator().manual_seed(4)
    for step in range(3):
        X = torch.randn(8, 8, generator=g).to(dev)
        T = torch.randn(8, 4, generator=g).to(dev)
        opt.zero_grad()
        loss = engine.train_step(
            inputs=X if engine.is_first else None,
            targets=T if engine.is_last else None,
            loss_fn=loss_fn if engine.is_last else None,
        )
        opt.step()
        ref_opt.zero_grad()
        ref_loss = loss_fn(ref(X), T)
        ref_loss.backward()
        ref_opt.step()
        if engine.is_last:
            assert torch.allclose(loss, ref_loss, atol=1e-4), (loss, ref_loss)
    stages = split_into_stages(ref, acc.num_processes)
    for p_eng, p_ref in zip(engine.stage.parameters(), stages[acc.process_index].parameters()):
        assert torch.allclose(p_eng, p_ref, atol=1e-4), (p_eng - p_ref).abs().max()


def main():
    import os

    on_gpu = os.environ.get("PP_GPU", "0") == "1"
    acc = Accelerator(cpu=not on_gpu)
    assert acc.num_processes >= 2
    if on_gpu:
        run_device(acc, "gpipe")
        run_device(acc, "1f1b")
    else:
        run(acc, "gpipe")
        run(acc, "1f1b")
    if acc.is_main_process:
        print("PP_TRAIN_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
