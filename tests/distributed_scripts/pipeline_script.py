"""2-process gloo pipeline-inference oracle: a 4-block Sequential split into
2 stages; the streamed microbatch output on the last rank must equal the
single-process forward."""

import torch
import torch.nn as nn

from accelerate_amd import Accelerator, set_seed
from accelerate_amd.inference import prepare_pipeline


def main():
    acc = Accelerator(cpu=True)
    set_seed(0)
    model = nn.Sequential(
        nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 32), nn.ReLU(), nn.Linear(32, 4)
    )
    x = torch.randn(12, 8)
    with torch.no_grad():
        expected = model(x)

    pipe = prepare_pipeline(model, num_chunks=3, gather_output=True)
    out = pipe(x if acc.is_main_process else None)
    assert out is not None
    assert torch.allclose(out.cpu(), expected, atol=1e-5), (out.cpu() - expected).abs().max()
    if acc.is_main_process:
        print("PIPELINE_PASS")
    acc.end_training()


if __name__ == "__main__":
    main()
