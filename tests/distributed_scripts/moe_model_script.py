"""2-process gloo: LlamaMoEForCausalLM trains under EP (experts split
across ranks) + our DDP engine on the dense parts, loss decreases and
dense params stay replica-identical while experts stay distinct."""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models.llama_moe import LlamaMoEConfig, LlamaMoEForCausalLM
from accelerate_amd.parallel.ddp import DistributedDataParallelEngine


def main():
    acc = Accelerator(cpu=True)
    n, r = acc.num_processes, acc.process_index
    set_seed(0)
    model = LlamaMoEForCausalLM(LlamaMoEConfig.tiny_moe())
    # make each rank's experts distinct (as real EP training would)
    with torch.no_grad():
        for p in model.layers[0].mlp.experts.parameters():
            p.add_(0.01 * (r + 1))
    engine = DistributedDataParallelEngine(model)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    ids = torch.randint(0, 1024, (4, 32), generator=torch.Generator().manual_seed(9))
    losses = []
    for _ in range(6):
        opt.zero_grad()
        out = engine(ids[r::n], labels=ids[r::n])
        out["loss"].backward()
        engine.finalize()
        opt.step()
        losses.append(out["loss"].item())
    assert losses[-1] < losses[0], losses
    # dense params identical across ranks; expert params still distinct
    g = model.embed_tokens.weight
    gs = [torch.empty_like(g) for _ in range(n)]
    dist.all_gather(gs, g)
    assert torch.allclose(gs[0], gs[1], atol=1e-6)
    e = next(model.layers[0].mlp.experts.parameters()).flatten()[:8]
    es = [torch.empty_like(e) for _ in range(n)]
    dist.all_gather(es, e)
    assert not torch.allclose(es[0], es[1]), "experts must stay rank-local"
    if acc.is_main_process:
        print("MOE_MODEL_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
