"""Fused CDNA4 AdamW (replaces torch fused/foreach Adam and DeepSpeed
FusedAdam — reference: SURVEY.md §2.9 N7).

One kernel launch per step for the whole parameter set (multi-tensor
chunked), float4-vectorized, HBM-bound. Numerics mirror torch.optim.AdamW
exactly; `tests/test_kernels.py::test_fused_adamw` compares against the
eager fp32 reference.

hipGraph-capturable by construction: the multi-tensor plan (device pointer
table + chunk prefix) is built once and cached; `step` and `lr` live in
device memory, the step increment happens in-graph, and the host writes lr
into the device scalar only when the scheduler changes it. Capturing
``opt.step()`` inside a torch.cuda.graph therefore replays correctly with
advancing bias corrections.

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
        # per-group cached execution plans: {group_idx: plan dict}
        self._plans = {}

    def _build_plan(self, plan_key, params, grads, exp_avgs, exp_avg_sqs, device, masters=None):
        import itertools

        gi = plan_key[0] if isinstance(plan_key, tuple) else plan_key
        n = len(params)
        lists = [params, grads, exp_avgs, exp_avg_sqs] + ([masters] if masters else [])
        ptrs = [t.data_ptr() for t in itertools.chain(*lists)]
        numels = [p.numel() for p in params]
        prefix = [0]
        for numel in numels:
            prefix.append(prefix[-1] + (numel + 16383) // 16384)
        # pinned staging + async H2D: capture-safe (a pageable copy would
        # implicitly sync and a hipHostMalloc would invalidate hipGraph
        # capture) — the pinned buffers are allocated ONCE per plan shape at
        # the first (out-of-capture) build and reused in-place on rebuilds,
        # so a rebuild triggered INSIDE capture only writes + async-copies.
        pin = device.type == "cuda" if hasattr(device, "type") else False
        cache = getattr(self, "_pinned_cache", None)
        if cache is None:
            cache = self._pinned_cache = {}
        cached = cache.get(plan_key)
        if cached is None or cached[0].numel() != len(ptrs) + len(numels):
            cpu_an = torch.empty(len(ptrs) + len(numels), dtype=torch.int64, pin_memory=pin)
            cpu_cp = torch.empty(len(prefix), dtype=torch.int32, pin_memory=pin)
            cache[plan_key] = (cpu_an, cpu_cp)
        else:
            cpu_an, cpu_cp = cached
        cpu_an.copy_(torch.tensor(ptrs + numels, dtype=torch.int64))
        cpu_cp.copy_(torch.tensor(prefix, dtype=torch.int32))
        addrs_numels = cpu_an.to(device, non_blocking=True)
        chunk_prefix = cpu_cp.to(device, non_blocking=True)
        group = self.param_groups[gi]
        old = self._plans.get(plan_key)
        if old is not None:
            step_t, lr_t, last_lr = old["step"], old["lr"], old["last_lr"]
        else:
            restored = getattr(self, "_restored_steps", {}).pop(gi, None)
            step_t = restored.to(device) if restored is not None else torch.zeros(1, dtype=torch.float32, device=device)
            lr_t = torch.full((1,), float(group["lr"]), dtype=torch.float32, device=device)
            last_lr = float(group["lr"])
        plan = {
            "key": tuple(ptrs),
            "addrs_numels": addrs_numels,
            "chunk_prefix": chunk_prefix,
            "n_tensors": n,
            "total_chunks": prefix[-1],
            # device-side hyperparams (stable storage across plan rebuilds so
            # captured graphs keep pointing at live step/lr scalars)
            "step": step_t,
            "lr": lr_t,
            "last_lr": last_lr,
            "_pinned": (cpu_an, cpu_cp),
        }
        self._plans[plan_key] = plan
        return plan

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        ext = None
        for gi, group in enumerate(self.param_groups):
            # fp32 and bf16(+master) GPU params run as two fused plans
            buckets = {"fp32": ([], [], [], [], None), "bf16": ([], [], [], [], [])}
            cpu_params = []
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32, memory_format=torch.preserve_format)
                    state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32, memory_format=torch.preserve_format)
                    if p.is_cuda and p.dtype == torch.bfloat16:
                        state["master"] = p.detach().to(torch.float32)
                if p.is_cuda and p.dtype == torch.float32:
                    b = buckets["fp32"]
                elif p.is_cuda and p.dtype == torch.bfloat16 and p.grad.dtype == torch.bfloat16:
                    b = buckets["bf16"]
                else:
                    state["step"] = state.get("step", 0) + 1
                    cpu_params.append((p, state))
                    continue
                b[0].append(p)
                b[1].append(p.grad if p.grad.is_contiguous() else p.grad.contiguous())
                b[2].append(state["exp_avg"])
                b[3].append(state["exp_avg_sq"])
                if b[4] is not None:
                    b[4].append(state["master"])

            for kind, (params, grads, exp_avgs, exp_avg_sqs, masters) in buckets.items():
                if not params:
                    continue
                if ext is None:
                    ext = _load_extension(required=True)
                import itertools

                plan_key = (gi, kind)
                plan = self._plans.get(plan_key)
                lists = [params, grads, exp_avgs, exp_avg_sqs] + ([masters] if masters else [])
                key = tuple(t.data_ptr() for t in itertools.chain(*lists))
                if plan is None or plan["key"] != key:
                    plan = self._build_plan(plan_key, params, grads, exp_avgs, exp_avg_sqs, params[0].device, masters)
                lr = float(group["lr"])
                if lr != plan["last_lr"]:
                    plan["lr"].fill_(lr)
                    plan["last_lr"] = lr
                beta1, beta2 = group["betas"]
                ext.fused_adamw_planned(
                    plan["addrs_numels"],
                    plan["chunk_prefix"],
                    plan["n_tensors"],
                    plan["total_chunks"],
                    plan["step"],
                    plan["lr"],
                    beta1,
                    beta2,
                    group["eps"],
                    group["weight_decay"],
                    self._grad_scale_tensor,
                    self._found_inf_tensor,
                    kind == "bf16",
                )
            for p, state in cpu_params:
                self._single_tensor_step(p, state, group)
        return loss

    def refresh_hyperparams(self):
        """Push host-side group['lr'] changes into the device scalars.

        Call after ``scheduler.step()`` when the optimizer kernel was captured
        in a hipGraph (the captured kernel reads lr from device memory)."""
        for plan_key, plan in self._plans.items():
            gi = plan_key[0] if isinstance(plan_key, tuple) else plan_key
            group = self.param_groups[gi]
            if float(group["lr"]) != plan["last_lr"]:
                plan["lr"].fill_(float(group["lr"]))
                plan["last_lr"] = float(group["lr"])

    def state_dict(self):
        self._sync_steps_from_device()
        return super().state_dict()

    def _sync_steps_from_device(self):
        # under graph replay the host never sees step increments; read back
        for plan_key, plan in self._plans.items():
            gi = plan_key[0] if isinstance(plan_key, tuple) else plan_key
            step = int(plan["step"].item())
            for p in self.param_groups[gi]["params"]:
                if p in self.state and p.is_cuda:
                    self.state[p]["step"] = step

    def load_state_dict(self, state_dict):
        super().load_state_dict(state_dict)
        self._plans = {}
        self._restored_steps = {}
        # restore device step counters from the loaded per-param steps
        for gi, group in enumerate(self.param_groups):
            steps = [self.state[p].get("step", 0) for p in group["params"] if p in self.state]
            if steps:
                self._restored_steps[gi] = torch.tensor([float(max(steps))], dtype=torch.float32)

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
