import os, torch
import accelerate_amd.ops.attention as fa

def compare(B,H,S,D,causal,bshd_dout):
    torch.manual_seed(0)
    q1 = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k1 = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    v1 = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    q2 = q1.detach().clone().requires_grad_(True)
    k2 = k1.detach().clone().requires_grad_(True)
    v2 = v1.detach().clone().requires_grad_(True)
    out1 = fa.flash_attention(q1,k1,v1,causal=causal)
    if bshd_dout:
        dout = torch.randn_like(out1)  # BSHD strides
    else:
        dout = torch.randn(B,H,S,D, device="cuda", dtype=torch.bfloat16)
    out1.backward(dout)
    os.environ["ACCELERATE_AMD_FA_BWD"] = "0"
    out2 = fa.flash_attention(q2,k2,v2,causal=causal)
    out2.backward(dout)
    del os.environ["ACCELERATE_AMD_FA_BWD"]
    fwd = (out1.float()-out2.float()).abs().max().item()
    rels = []
    for a,b in ((q1,q2),(k1,k2),(v1,v2)):
        rels.append(((a.grad.float()-b.grad.float()).abs().max()/(b.grad.float().abs().max()+1e-6)).item())
    print(f"B{B} H{H} S{S} D{D} causal={causal} bshd_dout={bshd_dout}: fwd={fwd:.4f} dq={rels[0]:.4f} dk={rels[1]:.4f} dv={rels[2]:.4f}", flush=True)

for bshd in (False, True):
    compare(2,8,512,128,True,bshd)
    compare(1,4,333,64,True,bshd)
    compare(2,4,256,128,True,bshd)
print("dout strides check:", flush=True)
q = torch.randn(1,2,64,64, device="cuda", dtype=torch.bfloat16)
out = fa.flash_attention(q.requires_grad_(True), q.detach(), q.detach())
print("out strides", out.stride(), "randn_like strides", torch.randn_like(out).stride())
