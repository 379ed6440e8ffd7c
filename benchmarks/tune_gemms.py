"""One-off hipBLASLt/rocBLAS algorithm tuning for the bench GEMM shapes
(torch TunableOp). Writes profiles/tunableop_mi355x.csv which bench.py
loads read-only. Run under gpurun; commit the CSV."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.cuda.tunable as tunable

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "profiles", "tunableop_mi355x.csv")


def main():
    tunable.enable(True)
    tunable.tuning_enable(True)
    tunable.set_max_tuning_duration(10)
    tunable.set_max_tuning_iterations(30)

    from accelerate_amd import Accelerator, set_seed
    from accelerate_amd.models import BertConfig, BertForSequenceClassification
    from accelerate_amd.ops.optim import FusedAdamW

    set_seed(0)
    acc = Accelerator()
    model = BertForSequenceClassification(BertConfig.bert_base()).to(torch.bfloat16)
    opt = FusedAdamW(model.parameters(), lr=2e-5)
    model, opt = acc.prepare(model, opt)
    ids = torch.randint(0, 30522, (16, 128), device="cuda")
    mask = torch.ones(16, 128, dtype=torch.long, device="cuda")
    types = torch.zeros(16, 128, dtype=torch.long, device="cuda")
    labels = torch.randint(0, 2, (16,), device="cuda")
    for i in range(3):
        opt.zero_grad()
        out = model(ids, attention_mask=mask, token_type_ids=types, labels=labels)
        acc.backward(out["loss"])
        opt.step()
        print(f"tuning pass {i} done", flush=True)
    torch.cuda.synchronize()
    tunable.write_file(OUT)
    print(f"wrote {OUT} with {len(tunable.get_results())} results")


if __name__ == "__main__":
    main()
