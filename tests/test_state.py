import pytest
import torch

from accelerate_amd.state import AcceleratorState, GradientState, PartialState
from accelerate_amd.utils.dataclasses import DistributedType


def test_partial_state_singleton():
    s1 = PartialState()
    s2 = PartialState()
    assert s1.__dict__ is s2.__dict__
    assert s1.num_processes == 1
    assert s1.process_index == 0
    assert s1.is_main_process
    assert s1.distributed_type == DistributedType.NO


def test_reset_state():
    s = PartialState()
    assert s.initialized
    PartialState._reset_state()
    assert PartialState._shared_state == {}


def test_accelerator_state_mixed_precision():
    state = AcceleratorState(mixed_precision="bf16")
    assert state.mixed_precision == "bf16"
    # second instantiation with the same mp is fine
    state2 = AcceleratorState(mixed_precision="bf16")
    assert state2.mixed_precision == "bf16"
    # conflicting mp raises
    with pytest.raises(ValueError):
        AcceleratorState(mixed_precision="fp16")


def test_split_between_processes_single():
    s = PartialState()
    with s.split_between_processes([1, 2, 3]) as x:
        assert x == [1, 2, 3]


def test_gradient_state():
    gs = GradientState()
    assert gs.sync_gradients
    assert gs.num_steps == 1
    assert not gs.in_dataloader
    assert gs.remainder == -1
    gs._set_sync_gradients(False)
    assert not GradientState().sync_gradients


def test_rank_gated_decorators():
    s = PartialState()
    calls = []

    @s.on_main_process
    def f():
        calls.append(1)

    f()
    assert calls == [1]
