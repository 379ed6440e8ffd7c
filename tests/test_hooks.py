import torch
import torch.nn as nn

from accelerate_amd.hooks import (
    AlignDevicesHook,
    ModelHook,
    SequentialHook,
    add_hook_to_module,
    remove_hook_from_module,
)


class PreForwardHook(ModelHook):
    def pre_forward(self, module, *args, **kwargs):
        return (args[0] + 1,) + args[1:], kwargs


class PostForwardHook(ModelHook):
    def post_forward(self, module, output):
        return output + 1


def test_add_and_remove_hook():
    model = nn.Linear(2, 2)
    x = torch.randn(2, 2)
    base = model(x)
    add_hook_to_module(model, PostForwardHook())
    assert torch.allclose(model(x), base + 1)
    remove_hook_from_module(model)
    assert torch.allclose(model(x), base)
    assert not hasattr(model, "_hf_hook")


def test_pre_forward_hook():
    model = nn.Linear(2, 2)
    x = torch.randn(2, 2)
    expected = model(x + 1)
    add_hook_to_module(model, PreForwardHook())
    assert torch.allclose(model(x), expected)


def test_sequential_and_append():
    model = nn.Linear(2, 2)
    x = torch.randn(2, 2)
    base = model(x)
    add_hook_to_module(model, PostForwardHook())
    add_hook_to_module(model, PostForwardHook(), append=True)
    assert isinstance(model._hf_hook, SequentialHook)
    assert torch.allclose(model(x), base + 2)


def test_align_devices_hook_offload_cpu():
    model = nn.Linear(4, 4)
    w = model.weight.detach().clone()
    hook = AlignDevicesHook(execution_device="cpu", offload=True)
    add_hook_to_module(model, hook)
    # weights offloaded to meta between forwards
    assert model.weight.device == torch.device("meta")
    x = torch.randn(2, 4)
    out = model(x)
    assert model.weight.device == torch.device("meta")
    expected = torch.nn.functional.linear(x, hook.weights_map["weight"], hook.weights_map["bias"])
    assert torch.allclose(out, expected, atol=1e-6)
    # detaching restores weights
    remove_hook_from_module(model)
    assert torch.equal(model.weight, w)


def test_io_same_device():
    model = nn.Linear(4, 4)
    hook = AlignDevicesHook(execution_device="cpu", io_same_device=True)
    add_hook_to_module(model, hook)
    out = model(torch.randn(2, 4))
    assert out.device == torch.device("cpu")


def test_onload_prefetcher_learns_ring_order():
    """The async-onload prefetcher learns execution order on pass 1 and
    prefetches hook i+1 (wrapping) from pass 2 on."""
    from accelerate_amd.hooks import _OnloadPrefetcher

    class StubHook:
        def __init__(self):
            self.kicked = 0
            self._prefetched = None

        def _start_prefetch(self, stream):
            self.kicked += 1

    p = _OnloadPrefetcher()
    p._get_stream = lambda: None  # no CUDA in CPU CI
    hooks = [StubHook() for _ in range(3)]
    for h in hooks:  # pass 1: learning, no prefetches yet
        p.note(h)
    assert p.learning and all(h.kicked == 0 for h in hooks)
    p.note(hooks[0])  # ring closes
    assert not p.learning
    assert hooks[1].kicked == 1  # successor of 0 prefetched
    p.note(hooks[2])
    assert hooks[0].kicked == 1  # wrap-around: last prefetches first
    # an unconsumed prefetch is not re-issued
    hooks[1]._prefetched = {}
    p.note(hooks[0])
    assert hooks[1].kicked == 1


def test_pinned_cache_budget(monkeypatch):
    """_pinned_value pins once per name and respects the byte budget."""
    import accelerate_amd.hooks as H

    hook = H.AlignDevicesHook(offload=True)
    t = torch.randn(16)
    monkeypatch.setenv("ACCELERATE_AMD_PINNED_CACHE_MB", "1")
    a = hook._pinned_value("w", t)
    b = hook._pinned_value("w", t)
    assert a is b  # cached
    monkeypatch.setenv("ACCELERATE_AMD_PINNED_CACHE_MB", "0")
    hook2 = H.AlignDevicesHook(offload=True)
    c = hook2._pinned_value("w", t)
    assert c is t  # budget exhausted: pageable fallback
