"""2-process gloo oracle for ring (P2P rotation) context parallelism:
forward AND backward of `ring_attention` over sequence shards must match
full-sequence flash attention exactly (same math, different collective
pattern). Covers causal, non-causal, and GQA inputs; gradient homing (each
rank ends with dk/dv for ITS chunk) is checked against the reference's
shard slices. The end-to-end prepare() path for cp_impl='ring' is covered
by cp_prepare_script.py."""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig
from accelerate_amd.ops.attention import flash_attention
from accelerate_amd.parallel.cp import ring_attention


def check_ring_parity(causal, hkv, tag):
    n = dist.get_world_size()
    r = dist.get_rank()
    B, H, S, D = 2, 4, 48, 16
    g = torch.Generator().manual_seed(11 + causal * 7 + hkv)
    q = torch.randn(B, H, S, D, generator=g, requires_grad=True)
    k = torch.randn(B, hkv, S, D, generator=g, requires_grad=True)
    v = torch.randn(B, hkv, S, D, generator=g, requires_grad=True)
    w = torch.randn(B, H, S, D, generator=g)  # loss weights, shared by seed

    # reference: full-sequence attention on every rank (identical by seed)
    ref = flash_attention(q, k, v, causal=causal)
    (ref * w).sum().backward()
    ref_gq, ref_gk, ref_gv = q.grad.clone(), k.grad.clone(), v.grad.clone()

    # ring: each rank owns a contiguous sequence shard
    s = S // n
    q2 = q.detach().clone().requires_grad_()
    k2 = k.detach().clone().requires_grad_()
    v2 = v.detach().clone().requires_grad_()
    out = ring_attention(
        q2[:, :, r * s : (r + 1) * s],
        k2[:, :, r * s : (r + 1) * s],
        v2[:, :, r * s : (r + 1) * s],
        causal=causal,
    )
    assert out.shape == (B, H, s, D)
    assert torch.allclose(out, ref[:, :, r * s : (r + 1) * s], atol=2e-5), (
        f"{tag}: fwd mismatch {(out - ref[:, :, r * s:(r + 1) * s]).abs().max()}"
    )
    (out * w[:, :, r * s : (r + 1) * s]).sum().backward()
    # grads flow only into this rank's shard slices of q2/k2/v2
    for name, got, want in (
        ("dq", q2.grad, ref_gq),
        ("dk", k2.grad, ref_gk),
        ("dv", v2.grad, ref_gv),
    ):
        sl = got[:, :, r * s : (r + 1) * s]
        want_sl = want[:, :, r * s : (r + 1) * s]
        assert torch.allclose(sl, want_sl, atol=5e-5), (
            f"{tag}: {name} mismatch {(sl - want_sl).abs().max()}"
        )
    dist.barrier()


def main():
    import os

    world = int(os.environ.get("WORLD_SIZE", "2"))
    acc = Accelerator(cpu=True, parallelism_config=ParallelismConfig(cp_size=world, cp_impl="ring"))
    assert acc.parallelism_config.cp_impl == "ring"

    check_ring_parity(causal=True, hkv=4, tag="causal")
    check_ring_parity(causal=False, hkv=4, tag="full")
    check_ring_parity(causal=True, hkv=2, tag="gqa")

    # non-divisible sequence -> loud error, not silent truncation
    from accelerate_amd.parallel.cp import shard_sequence

    try:
        shard_sequence(torch.randn(2, world * 5 + 1), dim=1)
        raise AssertionError("expected ValueError for non-divisible sequence")
    except ValueError:
        pass

    if acc.is_main_process:
        print("RING_CP_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
