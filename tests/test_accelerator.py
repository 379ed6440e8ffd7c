import os
import pickle
import tempfile

import pytest
import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from accelerate_amd import Accelerator, GradientAccumulationPlugin, set_seed
from accelerate_amd.optimizer import AcceleratedOptimizer
from accelerate_amd.scheduler import AcceleratedScheduler


def create_components(seed=42):
    set_seed(seed)
    model = nn.Sequential(nn.Linear(4, 8), nn.ReLU(), nn.Linear(8, 1))
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-3)
    scheduler = torch.optim.lr_scheduler.StepLR(optimizer, step_size=10)
    ds = TensorDataset(torch.randn(32, 4), torch.randn(32, 1))
    dl = DataLoader(ds, batch_size=4)
    return model, optimizer, scheduler, dl


def train_epoch(accelerator, model, optimizer, scheduler, dl):
    total = 0.0
    for x, y in dl:
        optimizer.zero_grad()
        loss = ((model(x) - y) ** 2).mean()
        accelerator.backward(loss)
        optimizer.step()
        scheduler.step()
        total += loss.item()
    return total


def test_prepare_and_train():
    accelerator = Accelerator()
    model, optimizer, scheduler, dl = create_components()
    model, optimizer, dl, scheduler = accelerator.prepare(model, optimizer, dl, scheduler)
    assert isinstance(optimizer, AcceleratedOptimizer)
    assert isinstance(scheduler, AcceleratedScheduler)
    loss0 = train_epoch(accelerator, model, optimizer, scheduler, dl)
    loss1 = train_epoch(accelerator, model, optimizer, scheduler, dl)
    assert loss1 < loss0


def test_gradient_accumulation_gating():
    accelerator = Accelerator(gradient_accumulation_steps=2)
    model, optimizer, scheduler, dl = create_components()
    model, optimizer, dl = accelerator.prepare(model, optimizer, dl)
    steps_with_sync = []
    for i, (x, y) in enumerate(dl):
        with accelerator.accumulate(model):
            loss = ((model(x) - y) ** 2).mean()
            accelerator.backward(loss)
            steps_with_sync.append(accelerator.sync_gradients)
            optimizer.step()
            optimizer.zero_grad()
    # every second step syncs; end_of_dataloader forces a final sync
    assert steps_with_sync[0] is False
    assert steps_with_sync[1] is True
    assert steps_with_sync[-1] is True


def test_optimizer_step_skipped_when_accumulating():
    accelerator = Accelerator(gradient_accumulation_steps=4)
    model, optimizer, scheduler, dl = create_components()
    model, optimizer, dl = accelerator.prepare(model, optimizer, dl)
    x, y = next(iter(dl))
    with accelerator.accumulate(model):
        loss = ((model(x) - y) ** 2).mean()
        accelerator.backward(loss)
        before = [p.clone() for p in model.parameters()]
        optimizer.step()  # gated: no sync on step 1 of 4
        after = list(model.parameters())
        for b, a in zip(before, after):
            assert torch.equal(b, a)


def test_loss_scaled_by_accumulation_steps():
    accelerator = Accelerator(gradient_accumulation_steps=2)
    model, optimizer, scheduler, dl = create_components()
    model, optimizer = accelerator.prepare(model, optimizer)
    x = torch.randn(4, 4)
    y = torch.randn(4, 1)
    loss = ((model(x) - y) ** 2).mean()
    accelerator.backward(loss)
    g_accum = [p.grad.clone() for p in model.parameters()]
    # reference grads at half scale
    model2, _, _, _ = create_components()
    model2.load_state_dict(accelerator.unwrap_model(model).state_dict())
    loss2 = ((model2(x) - y) ** 2).mean() / 2
    loss2.backward()
    for g1, (n, p2) in zip(g_accum, model2.named_parameters()):
        assert torch.allclose(g1, p2.grad, atol=1e-7), n


def test_clip_grad_norm_cpu():
    accelerator = Accelerator()
    model, optimizer, scheduler, dl = create_components()
    model, optimizer = accelerator.prepare(model, optimizer)
    x, y = torch.randn(8, 4), torch.randn(8, 1)
    loss = ((model(x) - y) ** 2).mean()
    accelerator.backward(loss)
    norm = accelerator.clip_grad_norm_(model.parameters(), max_norm=0.01)
    total = torch.sqrt(sum((p.grad**2).sum() for p in model.parameters()))
    assert total <= 0.011
    assert norm > 0


def test_save_load_state_roundtrip():
    accelerator = Accelerator()
    model, optimizer, scheduler, dl = create_components()
    model, optimizer, dl, scheduler = accelerator.prepare(model, optimizer, dl, scheduler)
    train_epoch(accelerator, model, optimizer, scheduler, dl)
    with tempfile.TemporaryDirectory() as d:
        accelerator.save_state(d)
        saved = {k: v.clone() for k, v in accelerator.unwrap_model(model).state_dict().items()}
        # perturb
        train_epoch(accelerator, model, optimizer, scheduler, dl)
        accelerator.load_state(d)
        for k, v in accelerator.unwrap_model(model).state_dict().items():
            assert torch.equal(v, saved[k]), k


def test_register_for_checkpointing():
    class Counter:
        def __init__(self):
            self.n = 0

        def state_dict(self):
            return {"n": self.n}

        def load_state_dict(self, sd):
            self.n = sd["n"]

    accelerator = Accelerator()
    c = Counter()
    accelerator.register_for_checkpointing(c)
    c.n = 7
    with tempfile.TemporaryDirectory() as d:
        accelerator.save_state(d)
        c.n = 0
        accelerator.load_state(d)
    assert c.n == 7


def test_save_model_sharded_index():
    accelerator = Accelerator()
    model = nn.Sequential(nn.Linear(64, 64), nn.Linear(64, 64))
    with tempfile.TemporaryDirectory() as d:
        accelerator.save_model(model, d, max_shard_size=20000)  # force sharding
        files = os.listdir(d)
        assert any("index" in f for f in files)
        # reload all shards and compare
        import safetensors.torch

        loaded = {}
        for f in files:
            if f.endswith(".safetensors"):
                loaded.update(safetensors.torch.load_file(os.path.join(d, f)))
        for k, v in model.state_dict().items():
            assert torch.equal(loaded[k], v)


def test_free_memory():
    accelerator = Accelerator()
    model, optimizer, scheduler, dl = create_components()
    model, optimizer = accelerator.prepare(model, optimizer)
    assert len(accelerator._models) == 1
    accelerator.free_memory()
    assert accelerator._models == []
    assert accelerator._optimizers == []


def test_accelerator_reinstantiation():
    a1 = Accelerator()
    a2 = Accelerator()
    assert a1.state.__dict__ is a2.state.__dict__


def test_autocast_context():
    accelerator = Accelerator(mixed_precision="bf16", cpu=True)
    with accelerator.autocast():
        x = torch.randn(2, 2) @ torch.randn(2, 2)
    assert x.dtype == torch.bfloat16


def test_unwrap_model_removes_fp32_wrapper():
    accelerator = Accelerator(mixed_precision="bf16", cpu=True)
    model, optimizer, scheduler, dl = create_components()
    model = accelerator.prepare_model(model)
    assert hasattr(model, "_original_forward")
    unwrapped = accelerator.unwrap_model(model, keep_fp32_wrapper=False)
    out = unwrapped(torch.randn(2, 4))
    assert out.dtype == torch.float32


def test_prepare_model_twice_returns_same_wrapper():
    """Double-wrap protection (reference test_accelerator.py:469): preparing
    an already-prepared model must not re-wrap it."""
    import torch.nn as nn

    acc = Accelerator(cpu=True)
    model = nn.Linear(4, 2)
    m1 = acc.prepare_model(model)
    m2 = acc.prepare_model(m1)
    assert m2 is m1
    assert len(acc._models) == 1


def test_prepared_dataloader_is_picklable():
    """Prepared loaders must survive pickling (reference: test_accelerator
    dataloader pickling :673) — needed for spawned dataloader workers."""
    import pickle

    from torch.utils.data import DataLoader, TensorDataset

    acc = Accelerator(cpu=True)
    dl = acc.prepare(DataLoader(TensorDataset(torch.arange(10).float()), batch_size=2))
    dl2 = pickle.loads(pickle.dumps(dl))
    a = [b[0].tolist() for b in dl]
    b = [b[0].tolist() for b in dl2]
    assert a == b


def test_parallelism_rank_properties():
    from accelerate_amd import ParallelismConfig

    acc = Accelerator(cpu=True)
    assert acc.data_parallel_rank == acc.process_index
    assert acc.tensor_parallel_rank == 0 and acc.context_parallel_rank == 0
    acc.parallelism_config = ParallelismConfig(tp_size=1)
    assert acc.torch_device_mesh is None
    assert acc.multi_device is False
    assert acc.optimizer_step_was_skipped is False
    assert acc.fp8_backend is None
    assert acc.should_save_model is True


def test_skip_first_batches_method():
    from torch.utils.data import DataLoader, TensorDataset

    acc = Accelerator(cpu=True)
    dl = acc.prepare(DataLoader(TensorDataset(torch.arange(10).float()), batch_size=2))
    rest = acc.skip_first_batches(dl, 2)
    assert [b[0].tolist() for b in rest] == [[4.0, 5.0], [6.0, 7.0], [8.0, 9.0]]


def test_maybe_context_parallel_noop():
    acc = Accelerator(cpu=True)
    bufs = [torch.randn(2, 8)]
    with acc.maybe_context_parallel(buffers=bufs):
        assert bufs[0].shape == (2, 8)  # world 1: untouched


def test_automatic_checkpoint_naming_and_total_limit(tmp_path):
    """save_state rotation (reference: ProjectConfiguration total_limit):
    automatic checkpoint_N naming, oldest folders deleted past the limit."""
    import os

    from accelerate_amd.utils import ProjectConfiguration

    acc = Accelerator(
        cpu=True,
        project_config=ProjectConfiguration(
            project_dir=str(tmp_path), automatic_checkpoint_naming=True, total_limit=2
        ),
    )
    model = torch.nn.Linear(4, 2)
    model = acc.prepare(model)
    for _ in range(3):
        acc.save_state()
    ckpts = sorted(os.listdir(tmp_path / "checkpoints"))
    assert ckpts == ["checkpoint_1", "checkpoint_2"], ckpts  # checkpoint_0 rotated out
    # load back the newest
    acc.load_state(str(tmp_path / "checkpoints" / "checkpoint_2"))


def test_fp16_skipped_step_detection_cpu():
    """step_was_skipped must flip on inf grads (scaler skip) and the
    AcceleratedScheduler must not advance the LR on skipped steps."""
    acc = Accelerator(cpu=True, mixed_precision="fp16")
    model = torch.nn.Linear(4, 2)
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    sched = torch.optim.lr_scheduler.StepLR(opt, step_size=1, gamma=0.5)
    model, opt, sched = acc.prepare(model, opt, sched)

    # healthy step: advances
    loss = model(torch.randn(2, 4)).sum()
    acc.backward(loss)
    opt.step()
    assert opt.step_was_skipped is False
    sched.step()
    lr_after_good = opt.param_groups[0]["lr"]

    # poisoned grads: scaler must skip and report it
    loss = model(torch.randn(2, 4)).sum()
    acc.backward(loss)
    for p in model.parameters():
        p.grad.fill_(float("inf"))
    before = [p.detach().clone() for p in model.parameters()]
    opt.step()
    assert opt.step_was_skipped is True
    for p, b in zip(model.parameters(), before):
        assert torch.equal(p.detach(), b), "skipped step must not move params"
    sched.step()
    assert opt.param_groups[0]["lr"] == lr_after_good, "LR must not advance on a skipped step"


def test_prepare_preserves_order_and_passthrough():
    """prepare() returns objects in argument order; non-wrappable objects
    pass through untouched (reference prepare semantics)."""
    from torch.utils.data import DataLoader, TensorDataset

    acc = Accelerator(cpu=True)
    model = torch.nn.Linear(4, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    dl = DataLoader(TensorDataset(torch.arange(4).float()), batch_size=2)
    sentinel = {"not": "preparable"}
    m, s, o, d = acc.prepare(model, sentinel, opt, dl)
    assert s is sentinel
    assert o.optimizer is opt
    assert hasattr(m, "forward")
    assert list(d)[0][0].numel() == 2


def test_free_memory_clears_references():
    acc = Accelerator(cpu=True)
    model = acc.prepare(torch.nn.Linear(4, 2))
    assert len(acc._models) == 1
    (model,) = acc.free_memory(model)
    assert acc._models == []
    assert model is None  # slots come back None so callers drop references


def test_dp_shard_without_fsdp_plugin_raises():
    """dp_shard is the FSDP dimension (reference state.py:995-1007):
    without a plugin, prepare_model must refuse rather than silently
    replicate where the user asked for sharding."""
    import pytest as _pytest
    import torch.nn as nn

    from accelerate_amd import Accelerator, ParallelismConfig

    acc = Accelerator(cpu=True)
    # bypass the world-size check (single process here): the plugin
    # requirement must trip FIRST in prepare_model regardless
    pc = ParallelismConfig.__new__(ParallelismConfig)
    pc.dp_replicate_size, pc.dp_shard_size, pc.tp_size, pc.cp_size = 1, 2, 1, 1
    pc._groups = {}
    acc.parallelism_config = pc
    with _pytest.raises(ValueError, match="FSDP plugin"):
        acc.prepare_model(nn.Linear(4, 4))
