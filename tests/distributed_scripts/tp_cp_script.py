"""4-process gloo oracle: TP (tp=2) x CP (cp=2) through plain prepare() +
maybe_context_parallel. Attention runs with tp-sharded heads while the cp
group all-gathers the sequence-sharded KV; gradients average over the cp
group (the grad domain at dp=1) while tp shards stay per-rank. One step
must match the single-process full-sequence reference.
"""

import torch
import torch.distributed as dist

from accelerate_amd import Accelerator, ParallelismConfig, set_seed
from accelerate_amd.models.llama import LlamaConfig, LlamaForCausalLM


def main():
    pc = ParallelismConfig(tp_size=2, cp_size=2, cp_impl="allgather")
    acc = Accelerator(cpu=True, parallelism_config=pc)
    assert acc.num_processes == 4

    set_seed(0)
    model = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))
    set_seed(0)
    ref = LlamaForCausalLM(LlamaConfig.tiny(num_hidden_layers=2))

    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)
    model, opt = acc.prepare(model, opt)

    S = 16
    g = torch.Generator().manual_seed(11)
    ids = torch.randint(0, 1024, (2, S), generator=g)  # same batch on all ranks

    # forward parity first (this rank's sequence shard of the logits)
    with torch.no_grad():
        ref_out = ref(ids)["logits"]

    shard = [ids.clone()]
    with acc.maybe_context_parallel(buffers=shard, buffer_seq_dims=[1]):
        local = shard[0]
        assert local.shape[1] == S // 2
        with torch.no_grad():
            out = model(local)["logits"]
        cp_rank = pc.coords(acc.process_index)["cp"]
        want = ref_out[:, cp_rank * (S // 2) : (cp_rank + 1) * (S // 2)]
        assert torch.allclose(out, want, atol=1e-4), f"fwd diverges {(out - want).abs().max()}"

        opt.zero_grad()
        loss = model(local)["logits"].float().pow(2).mean()
        acc.backward(loss)
    opt.step()

    ref_opt.zero_grad()
    ref(ids)["logits"].float().pow(2).mean().backward()
    ref_opt.step()

    with torch.no_grad():
        out2 = model(ids)["logits"]  # full sequence, outside the cp scope
        want2 = ref(ids)["logits"]
    assert torch.allclose(out2, want2, atol=1e-4), f"step diverges {(out2 - want2).abs().max()}"
    dist.barrier()
    if acc.is_main_process:
        print("TP_CP_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
