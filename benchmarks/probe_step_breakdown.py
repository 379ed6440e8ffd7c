"""Where does the 16.8 ms/step go? Breaks the BERT-base step into
fwd / bwd / optimizer and times a hipGraph-captured fwd+bwd replay.
Writes findings to stdout (run under gpurun)."""

import time

import torch

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models import BertConfig, BertForSequenceClassification
from accelerate_amd.ops.optim import FusedAdamW

B, S = 16, 128


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    set_seed(0)
    acc = Accelerator(mixed_precision="bf16")
    model = BertForSequenceClassification(BertConfig.bert_base())
    opt = FusedAdamW(model.parameters(), lr=2e-5)
    model, opt = acc.prepare(model, opt)

    ids = torch.randint(0, 30522, (B, S), device="cuda")
    mask = torch.ones(B, S, dtype=torch.long, device="cuda")
    types = torch.zeros(B, S, dtype=torch.long, device="cuda")
    labels = torch.randint(0, 2, (B,), device="cuda")

    def fwd():
        return model(ids, attention_mask=mask, token_type_ids=types, labels=labels)["loss"]

    def fwd_bwd():
        opt.zero_grad(set_to_none=False)
        loss = fwd()
        acc.backward(loss)
        return loss

    def full():
        loss = fwd_bwd()
        opt.step()
        return loss

    # long eager run first: does NaN appear without any graph involvement?
    for i in range(100):
        loss = full()
        if i % 20 == 19:
            torch.cuda.synchronize()
            print(f"eager step {i+1}: loss={loss.item():.5f} finite={torch.isfinite(loss).item()}")

    with torch.no_grad():
        t_fwd_nograd = timeit(lambda: model(ids, attention_mask=mask, token_type_ids=types, labels=labels))
    t_fwd = timeit(fwd)
    t_fwdbwd = timeit(fwd_bwd)
    t_full = timeit(full)
    print(f"fwd(no_grad): {t_fwd_nograd:.3f} ms")
    print(f"fwd:          {t_fwd:.3f} ms")
    print(f"fwd+bwd:      {t_fwdbwd:.3f} ms")
    print(f"full step:    {t_full:.3f} ms  (optimizer = {t_full - t_fwdbwd:.3f} ms)")

    # ---- hipGraph capture of zero+fwd+bwd ----
    import faulthandler
    import gc

    faulthandler.enable()
    # Drop every reference to prior autograd graphs before capture: a stale
    # AccumulateGrad node pinned to the default stream segfaults ROCm's
    # capture_end (observed on MI355X, torch 2.10+rocm7.0).
    loss = None
    del loss
    gc.collect()
    torch.cuda.synchronize()
    fwd_bwd()  # materialize grads
    gc.collect()
    print("phase: pre-warmup done", flush=True)
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(3):
            fwd_bwd()
    torch.cuda.current_stream().wait_stream(side)
    print("phase: side-stream warmup done", flush=True)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        loss_out = fwd_bwd()
    print("phase: capture done", flush=True)

    def replay():
        g.replay()

    t_graph = timeit(replay, iters=50, warmup=5)
    torch.cuda.synchronize()
    print(f"hipGraph fwd+bwd replay: {t_graph:.3f} ms (eager was {t_fwdbwd:.3f}); loss after pure replay: {loss_out.item():.4f}")
    gw = acc.unwrap_model(model).bert.layers[0].ffn_in.weight.grad
    print(f"grad finite after replay: {torch.isfinite(gw).all().item()}  |grad| max {gw.abs().max().item():.3e}")

    def graph_full():
        g.replay()
        opt.step()

    t_graph_full = timeit(graph_full, iters=50, warmup=5)
    torch.cuda.synchronize()
    print(f"hipGraph + eager opt:    {t_graph_full:.3f} ms -> {B/ t_graph_full * 1000:.0f} samples/s")
    print(f"loss after graph+opt replays: {loss_out.item():.4f}")
    p = acc.unwrap_model(model).bert.layers[0].ffn_in.weight
    print(f"param finite: {torch.isfinite(p).all().item()}")

    # eager steps after the graph phase for comparison (is nan from capture or from training?)
    for i in range(5):
        loss = full()
    torch.cuda.synchronize()
    print(f"eager loss after graph phases: {loss.item():.4f}")


if __name__ == "__main__":
    main()
