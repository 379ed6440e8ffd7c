"""Llama-3-70B bf16 inference on ONE MI355X — the 288 GB HBM3E flagship demo.

70B bf16 weights are ~141 GB: they do not fit on any 80/96/141-GB part
without quantization or multi-GPU sharding; on MI355X the full model plus a
KV cache sits in HBM with >100 GB to spare. (Reference comparison point:
its big-model-inference tables, BASELINE.md, need CPU/disk offload for a
30B fp16 model on 24 GB cards at 2.4-34 s/token.)

Random-init weights (no network for checkpoints), decode measured per token.

  gpurun -- 'python benchmarks/llama70b_inference_demo.py'
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM

    int8 = "--int8" in sys.argv
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    config = LlamaConfig.llama3_70b()
    t0 = time.perf_counter()
    with torch.device(dev):
        torch.set_default_dtype(torch.bfloat16)
        model = LlamaForCausalLM(config).eval()
        torch.set_default_dtype(torch.float32)
    if not int8:
        from accelerate_amd.ops.linear import convert_linears_for_inference

        convert_linears_for_inference(model)  # fused decode GEMV
    if int8:
        # weight-only int8: halves HBM residency AND decode time (decode is
        # weight-bandwidth-bound; the fused w8a16 GEMV reads int8 directly)
        from accelerate_amd.utils import QuantizationConfig, load_and_quantize_model

        model = load_and_quantize_model(model, QuantizationConfig(load_in_8bit=True))
        torch.cuda.empty_cache()
    torch.cuda.synchronize()
    t_build = time.perf_counter() - t0
    params = sum(p.numel() for p in model.parameters()) + sum(
        b.numel() for n, b in model.named_buffers() if "qweight" in n or "scales" in n
    )
    w_gb = torch.cuda.memory_allocated() / 2**30

    ids = torch.randint(0, config.vocab_size, (1, 32), device=dev)
    with torch.no_grad():
        out = model.generate(ids, max_new_tokens=4)  # warmup (prefill + alloc)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = model.generate(ids, max_new_tokens=16)
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(
        json.dumps(
            {
                "bench": f"llama3-70b {'int8' if int8 else 'bf16'} inference, ONE MI355X, random init",
                "params_b": round(params / 1e9, 1),
                "weights_gb": round(w_gb, 1),
                "build_s": round(t_build, 1),
                "s_per_token_decode": round(dt / 16, 3),
                "peak_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 1),
                "new_tokens": int(out.shape[1] - ids.shape[1]),
            }
        )
    )


if __name__ == "__main__":
    main()
