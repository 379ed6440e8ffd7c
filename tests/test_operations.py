import numpy as np
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st
import torch

from accelerate_amd.state import PartialState
from accelerate_amd.utils.operations import (
    broadcast,
    concatenate,
    convert_outputs_to_fp32,
    convert_to_fp32,
    find_batch_size,
    find_device,
    gather,
    gather_object,
    honor_type,
    listify,
    pad_across_processes,
    recursively_apply,
    reduce,
    send_to_device,
)


@pytest.fixture(autouse=True)
def state():
    return PartialState()


def test_send_to_device_nested():
    data = {"a": torch.randn(2), "b": [torch.randn(3), (torch.randn(1),)], "c": 5}
    out = send_to_device(data, "cpu")
    assert out["a"].device.type == "cpu"
    assert out["b"][1][0].device.type == "cpu"
    assert out["c"] == 5


def test_send_to_device_skip_keys():
    data = {"a": torch.randn(2), "keep": torch.randn(2)}
    out = send_to_device(data, "cpu", skip_keys=["keep"])
    assert out["keep"] is data["keep"]


def test_honor_type_namedtuple():
    from collections import namedtuple

    Point = namedtuple("Point", ["x", "y"])
    p = Point(1, 2)
    out = honor_type(p, (v * 2 for v in p))
    assert isinstance(out, Point)
    assert out.x == 2 and out.y == 4


def test_recursively_apply():
    data = ([torch.ones(2)], {"k": torch.ones(3)})
    out = recursively_apply(lambda t: t * 2, data)
    assert torch.equal(out[0][0], torch.full((2,), 2.0))
    assert torch.equal(out[1]["k"], torch.full((3,), 2.0))


def test_find_batch_size():
    assert find_batch_size({"x": torch.randn(4, 3)}) == 4
    assert find_batch_size([torch.randn(7, 2), torch.randn(3)]) == 7
    assert find_batch_size("no tensors") is None


def test_find_device():
    assert find_device({"a": [torch.randn(1)]}) == torch.device("cpu")


def test_convert_to_fp32():
    t = {"a": torch.randn(2, dtype=torch.bfloat16), "b": torch.ones(2, dtype=torch.int64)}
    out = convert_to_fp32(t)
    assert out["a"].dtype == torch.float32
    assert out["b"].dtype == torch.int64  # ints untouched


def test_convert_outputs_to_fp32_not_picklable():
    import pickle

    def fwd(x):
        return x

    wrapped = convert_outputs_to_fp32(fwd)
    assert wrapped(torch.ones(1, dtype=torch.bfloat16)).dtype == torch.float32
    with pytest.raises(Exception):
        pickle.dumps(wrapped.__wrapped__)


def test_listify():
    out = listify({"a": torch.tensor([1.5, 2.5]), "b": torch.tensor(3)})
    assert out == {"a": [1.5, 2.5], "b": 3}


def test_single_process_collectives_passthrough():
    t = torch.randn(3)
    assert torch.equal(gather(t), t)
    assert gather_object(["x"]) == ["x"]  # world 1: unchanged (reference parity)
    assert torch.equal(broadcast(t), t)
    r = reduce(t, "mean")
    assert torch.allclose(r, t)
    assert torch.equal(pad_across_processes(t), t)


def test_concatenate():
    data = [{"x": torch.ones(2, 3)}, {"x": torch.zeros(1, 3)}]
    out = concatenate(data)
    assert out["x"].shape == (3, 3)


class TestStructuredOpsProperties:
    """Property sweep: structure-preserving ops must round-trip arbitrary
    nested (dict/list/tuple/tensor) payloads (reference test_utils.py ops)."""

    @staticmethod
    def _payload(depth_seed, shape):
        t = torch.arange(float(shape[0] * shape[1])).reshape(shape)
        if depth_seed % 3 == 0:
            return {"a": t, "b": [t + 1, (t + 2,)]}
        if depth_seed % 3 == 1:
            return [t, {"x": t * 2}]
        return (t, t + 5)

    @given(depth_seed=st.integers(0, 8), rows=st.integers(1, 6), cols=st.integers(1, 6))
    @settings(max_examples=50, deadline=None, derandomize=True)
    def test_find_device_and_send(self, depth_seed, rows, cols):
        from accelerate_amd.utils.operations import find_device, send_to_device

        data = self._payload(depth_seed, (rows, cols))
        assert find_device(data).type == "cpu"
        out = send_to_device(data, torch.device("cpu"))
        flat_in, flat_out = [], []

        def collect(x, acc):
            if isinstance(x, torch.Tensor):
                acc.append(x)
            elif isinstance(x, dict):
                [collect(v, acc) for v in x.values()]
            elif isinstance(x, (list, tuple)):
                [collect(v, acc) for v in x]

        collect(data, flat_in)
        collect(out, flat_out)
        assert len(flat_in) == len(flat_out)
        for a, b in zip(flat_in, flat_out):
            assert torch.equal(a, b)

    @given(rows=st.integers(1, 7), cols=st.integers(1, 5), pad_index=st.integers(0, 3))
    @settings(max_examples=50, deadline=None, derandomize=True)
    def test_pad_across_processes_world1(self, rows, cols, pad_index):
        from accelerate_amd.utils.operations import pad_across_processes

        t = torch.randn(rows, cols)
        out = pad_across_processes(t, dim=0, pad_index=pad_index)
        # world 1: max length == own length, content unchanged
        assert torch.equal(out, t)

    @given(n=st.integers(1, 5))
    @settings(max_examples=30, deadline=None, derandomize=True)
    def test_concatenate_matches_torch_cat(self, n):
        from accelerate_amd.utils.operations import concatenate

        parts = [{"x": torch.randn(2, 3), "y": (torch.randn(4),)} for _ in range(n)]
        out = concatenate(parts)
        assert out["x"].shape == (2 * n, 3)
        assert out["y"][0].shape == (4 * n,)


def test_collective_ops_uneven_3proc():
    """pad/gather/reduce/broadcast/gather_object at world 3 with
    rank-uneven shapes (pad equalizes to the max before concatenation;
    gather_object flattens each rank's list — reference semantics)."""
    from tests.testing_utils import launch_distributed

    out = launch_distributed("tests/distributed_scripts/operations_script.py", nproc=3, timeout=300)
    assert "OPERATIONS_PASS" in out
