import numpy as np
import pytest
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
    assert gather_object(["x"]) == [["x"]]
    assert torch.equal(broadcast(t), t)
    r = reduce(t, "mean")
    assert torch.allclose(r, t)
    assert torch.equal(pad_across_processes(t), t)


def test_concatenate():
    data = [{"x": torch.ones(2, 3)}, {"x": torch.zeros(1, 3)}]
    out = concatenate(data)
    assert out["x"].shape == (3, 3)
