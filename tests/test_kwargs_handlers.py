"""KwargsHandler plumb-through into Accelerator internals and small init
utilities (reference: tests/test_kwargs_handlers.py)."""

import torch
import torch.nn as nn

from accelerate_amd import (
    Accelerator,
    AutocastKwargs,
    DistributedDataParallelKwargs,
    GradScalerKwargs,
    InitProcessGroupKwargs,
    init_on_device,
    synchronize_rng_states,
)
from accelerate_amd.state import PartialState
from accelerate_amd.utils.dataclasses import FP8RecipeKwargs, ProfileKwargs


def _fresh():
    PartialState._reset_state()


def test_handlers_reach_accelerator_slots():
    _fresh()
    ddp = DistributedDataParallelKwargs(bucket_cap_mb=32, comm_dtype="bf16")
    scaler = GradScalerKwargs(init_scale=2.0**10, growth_interval=500)
    autocast = AutocastKwargs(enabled=False)
    fp8 = FP8RecipeKwargs(amax_history_len=8)
    prof = ProfileKwargs(activities=["cpu"])
    acc = Accelerator(cpu=True, kwargs_handlers=[ddp, scaler, autocast, fp8, prof])
    assert acc.ddp_handler is ddp and acc.ddp_handler.bucket_cap_mb == 32
    assert acc.scaler_handler is scaler
    assert acc.autocast_handler is autocast
    assert acc.fp8_recipe_handler is fp8
    assert acc.profile_handler is prof
    _fresh()


def test_to_kwargs_diffs_against_defaults():
    k = GradScalerKwargs(init_scale=2.0**10)
    d = k.to_kwargs()
    assert d == {"init_scale": 2.0**10}  # only the non-default field
    assert InitProcessGroupKwargs().to_kwargs() == {}


def test_init_on_device_meta():
    with init_on_device(torch.device("meta")):
        m = nn.Linear(8, 4)
    assert m.weight.device.type == "meta"
    # and CPU ctx leaves params materialized
    with init_on_device(torch.device("cpu")):
        m2 = nn.Linear(4, 2)
    assert m2.weight.device.type == "cpu"


def test_synchronize_rng_states_generator():
    _fresh()
    Accelerator(cpu=True)  # world 1: must be a no-op that doesn't crash
    g = torch.Generator().manual_seed(3)
    synchronize_rng_states(["generator"], generator=g)
    before = g.get_state().clone()
    synchronize_rng_states(["generator"], generator=g)
    assert torch.equal(g.get_state(), before)
    _fresh()


def test_ddp_kwargs_comm_hook_alias_feeds_engine():
    """comm_hook (reference alias) folds into comm_dtype and must NOT leak
    into the engine kwargs — regression: the SCALE launch crashed with
    'unexpected keyword argument comm_hook' at world > 1."""
    import torch.nn as nn

    from accelerate_amd.parallel.ddp import DistributedDataParallelEngine
    from accelerate_amd.utils import DDPCommunicationHookType, DistributedDataParallelKwargs

    h = DistributedDataParallelKwargs(comm_hook=DDPCommunicationHookType.BF16)
    d = h.to_dict()
    assert "comm_hook" not in d
    assert d["comm_dtype"] == "bf16"
    engine = DistributedDataParallelEngine(nn.Linear(4, 4), **d)  # must not raise
    assert engine.comm_dtype is not None
