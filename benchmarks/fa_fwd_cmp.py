"""Forward-only attention kernel timing: legacy vs swapped (env
ACCELERATE_AMD_FA_FWD picks the impl; run once per value)."""
import os, sys, time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from accelerate_amd.ops import _load_extension

B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
torch.manual_seed(0)
q = torch.randn(B, Hq, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
ext = _load_extension(required=True)


def bench(causal, iters=20):
    for _ in range(3):
        ext.flash_attn_fwd(q, k, v, causal, D**-0.5, 0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ext.flash_attn_fwd(q, k, v, causal, D**-0.5, 0)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    fl = 2 * B * Hq * S * S * D * 2 * (0.5 if causal else 1.0)
    return dt * 1e3, fl / dt / 1e12


for c in (True, False):
    ms, tf = bench(c)
    print(f"impl={os.environ.get('ACCELERATE_AMD_FA_FWD', 'legacy'):8s} causal={c}: {ms:.3f} ms  {tf:.0f} TF/s")
