"""A/B: eager per-token decode loop vs hipGraph-captured decode (70B)."""
import json, os, sys, time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM

dev = torch.device("cuda:0")
cfg = LlamaConfig.llama3_70b()
with torch.device(dev):
    torch.set_default_dtype(torch.bfloat16)
    model = LlamaForCausalLM(cfg).eval()
    torch.set_default_dtype(torch.float32)
torch.cuda.synchronize()
ids = torch.randint(0, cfg.vocab_size, (1, 32), device=dev)
with torch.no_grad():
    for mode in (False, True, False, True):  # interleave to cancel drift
        model.generate(ids, max_new_tokens=2, graph_decode=mode)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        model.generate(ids, max_new_tokens=16, graph_decode=mode)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 16
        print(json.dumps({"mode": "graph" if mode else "eager", "s_per_token": round(dt, 4)}))
