"""2-process gloo: Llama split into pipeline stages trains identically to
the unsplit model (loss + updated weights), GPipe and 1F1B."""

import torch

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.models.llama import (
    LlamaConfig,
    LlamaForCausalLM,
    build_llama_pipeline_stages,
    causal_lm_loss,
)
from accelerate_amd.parallel.pp import PipelineParallelEngine


def run(acc, schedule):
    r = acc.process_index
    set_seed(0)
    config = LlamaConfig.tiny(num_hidden_layers=4)
    model = LlamaForCausalLM(config)
    ref = LlamaForCausalLM(config)
    ref.load_state_dict(model.state_dict())

    stages = build_llama_pipeline_stages(model, acc.num_processes)
    engine = PipelineParallelEngine(stage=stages[r], num_microbatches=2, schedule=schedule)
    opt = torch.optim.SGD(engine.parameters(), lr=0.05)
    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05)

    g = torch.Generator().manual_seed(7)
    for step in range(2):
        ids = torch.randint(0, 1024, (4, 32), generator=g)
        opt.zero_grad()
        loss = engine.train_step(
            inputs=ids if engine.is_first else None,
            targets=ids if engine.is_last else None,
            loss_fn=causal_lm_loss if engine.is_last else None,
        )
        opt.step()
        ref_opt.zero_grad()
        ref_out = ref(ids, labels=ids)
        ref_out["loss"].backward()
        ref_opt.step()
        if engine.is_last:
            assert torch.allclose(loss, ref_out["loss"], atol=1e-5), (loss, ref_out["loss"])

    ref_stages = build_llama_pipeline_stages(ref, acc.num_processes)
    for p_eng, p_ref in zip(stages[r].parameters(), ref_stages[r].parameters()):
        assert torch.allclose(p_eng, p_ref, atol=1e-5), (p_eng - p_ref).abs().max()


def main():
    acc = Accelerator(cpu=True)
    assert acc.num_processes >= 2
    run(acc, "gpipe")
    run(acc, "1f1b")
    if acc.is_main_process:
        print("PP_LLAMA_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
