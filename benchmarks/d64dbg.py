import os, sys, torch
sys.path.insert(0, "/root/repo")
from accelerate_amd.ops import _load_extension
ext = _load_extension(required=True)
torch.manual_seed(0)
B, H, S, D = 1, 2, 256, 64
q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
out, lse = ext.flash_attn_fwd(q, k, v, False, D**-0.5, 0)
ref = torch.nn.functional.scaled_dot_product_attention(q.float(), k.float(), v.float())
d = (out.float() - ref).abs()
print("impl", os.environ.get("ACCELERATE_AMD_FA_FWD", "default"), "max diff", d.max().item())
if d.max() > 0.05:
    idx = (d > 0.05)
    rows = idx.any(-1).nonzero()[:5]
    print("bad rows (b,h,q):", rows.tolist())
    b, h, r = rows[0].tolist()
    cols = idx[b, h, r].nonzero().flatten()
    print("bad cols:", cols[:20].tolist(), "of", cols.numel())
