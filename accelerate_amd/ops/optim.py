"""Fused CDNA4 AdamW (replaces torch fused/foreach Adam and DeepSpeed
FusedAdam — reference: SURVEY.md §2.9 N7).

One kernel launch per step for the whole parameter set (multi-tensor
chunked), float4-vectorized, HBM-bound. Numerics mirror torch.optim.AdamW
exactly; `tests/test_kernels.py::test_fused_adamw` compares against the
eager fp32 reference.

On CPU (unit tests) the step falls back to torch.optim.AdamW math; on a GPU
the HIP extension is required — no silent eager fallback.
"""

from typing import Optional

import torch

from . import _load_extension


class FusedAdamW(torch.optim.Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 1e-2,
        *,
        grad_scale_tensor: Optional[torch.Tensor] = None,
        found_inf_tensor: Optional[torch.Tensor] = None,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._grad_scale_tensor = grad_scale_tensor
        self._found_inf_tensor = found_inf_tensor

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            params, grads, exp_avgs, exp_avg_sqs = [], [], [], []
            cpu_params = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                    state["exp_avg_sq"] = torch.zeros_like(p, memory_format=torch.preserve_format)
                state["step"] += 1
                if p.is_cuda and p.dtype == torch.float32:
                    params.append(p)
                    grads.append(p.grad if p.grad.is_contiguous() else p.grad.contiguous())
                    exp_avgs.append(state["exp_avg"])
                    exp_avg_sqs.append(state["exp_avg_sq"])
                else:
                    cpu_params.append((p, state))

            if params:
                ext = _load_extension(required=True)
                step = self.state[params[0]]["step"]
                beta1, beta2 = group["betas"]
                ext.fused_adamw(
                    params,
                    grads,
                    exp_avgs,
                    exp_avg_sqs,
                    step,
                    group["lr"],
                    beta1,
                    beta2,
                    group["eps"],
                    group["weight_decay"],
                    self._grad_scale_tensor,
                    self._found_inf_tensor,
                )
            for p, state in cpu_params:
                self._single_tensor_step(p, state, group)
        return loss

    @staticmethod
    def _single_tensor_step(p, state, group):
        # reference torch.optim.AdamW math (CPU fallback for unit tests)
        beta1, beta2 = group["betas"]
        grad = p.grad
        exp_avg, exp_avg_sq = state["exp_avg"], state["exp_avg_sq"]
        step = state["step"]
        p.mul_(1 - group["lr"] * group["weight_decay"])
        exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
        exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
        bias_correction1 = 1 - beta1**step
        bias_correction2 = 1 - beta2**step
        step_size = group["lr"] / bias_correction1
        denom = (exp_avg_sq.sqrt() / (bias_correction2**0.5)).add_(group["eps"])
        p.addcdiv_(exp_avg, denom, value=-step_size)
