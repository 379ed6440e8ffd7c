"""Loss scaler for fp16 mixed precision.

Replaces ``torch.amp.GradScaler`` + the ``_amp_foreach_non_finite_check_and
_unscale_`` CUDA kernel (reference: SURVEY.md §2.9 N5) with our CDNA4
unscale+check kernel and plain host-side growth/backoff logic. The
found-inf flag stays on-device through step-skipping; one small D2H read
happens in ``update()`` per step (same as torch's scaler).
"""

from collections import defaultdict
from typing import Optional

import torch

from . import _load_extension


class GradScaler:
    def __init__(
        self,
        init_scale: float = 2.0**16,
        growth_factor: float = 2.0,
        backoff_factor: float = 0.5,
        growth_interval: int = 2000,
        enabled: bool = True,
    ):
        self._enabled = enabled
        self._init_scale = init_scale
        self._scale: Optional[torch.Tensor] = None
        self._growth_factor = growth_factor
        self._backoff_factor = backoff_factor
        self._growth_interval = growth_interval
        self._growth_tracker = 0
        self._per_optimizer_states = defaultdict(dict)

    def is_enabled(self):
        return self._enabled

    def _lazy_init(self, device):
        if self._scale is None:
            self._scale = torch.full((1,), self._init_scale, dtype=torch.float32, device=device)

    def get_scale(self) -> float:
        if not self._enabled:
            return 1.0
        return self._init_scale if self._scale is None else float(self._scale.item())

    def scale(self, outputs):
        if not self._enabled:
            return outputs
        if isinstance(outputs, torch.Tensor):
            self._lazy_init(outputs.device)
            return outputs * self._scale
        return type(outputs)(self.scale(o) for o in outputs)

    def unscale_(self, optimizer):
        if not self._enabled:
            return
        state = self._per_optimizer_states[id(optimizer)]
        if state.get("unscaled", False):
            raise RuntimeError("unscale_() has already been called on this optimizer since the last update().")
        self._lazy_init(self._scale.device if self._scale is not None else "cuda")
        inv_scale = self._scale.reciprocal()
        found_inf = torch.zeros((1,), dtype=torch.float32, device=self._scale.device)
        grads_gpu, grads_other = [], []
        for group in optimizer.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                if p.grad.is_cuda and p.grad.dtype == torch.float32 and p.grad.is_contiguous():
                    grads_gpu.append(p.grad)
                else:
                    grads_other.append(p.grad)
        if grads_gpu:
            ext = _load_extension(required=True)
            ext.unscale_and_check(grads_gpu, inv_scale, found_inf)
        for g in grads_other:
            g.mul_(inv_scale.to(g.device))
            if not torch.isfinite(g).all():
                found_inf.fill_(1.0)
        state["found_inf"] = found_inf
        state["unscaled"] = True

    def step(self, optimizer, *args, **kwargs):
        if not self._enabled:
            return optimizer.step(*args, **kwargs)
        state = self._per_optimizer_states[id(optimizer)]
        if not state.get("unscaled", False):
            self.unscale_(optimizer)
        # one host read: skip the step on overflow (torch's scaler does the same)
        if float(state["found_inf"].item()) == 0.0:
            return optimizer.step(*args, **kwargs)
        return None

    def update(self, new_scale=None):
        if not self._enabled:
            return
        if self._scale is None:
            return
        if new_scale is not None:
            if isinstance(new_scale, torch.Tensor):
                self._scale.copy_(new_scale)
            else:
                self._scale.fill_(float(new_scale))
        else:
            found = any(
                float(s["found_inf"].item()) != 0.0 for s in self._per_optimizer_states.values() if "found_inf" in s
            )
            if found:
                self._scale.mul_(self._backoff_factor)
                self._growth_tracker = 0
            else:
                self._growth_tracker += 1
                if self._growth_tracker >= self._growth_interval:
                    self._scale.mul_(self._growth_factor)
                    self._growth_tracker = 0
        self._per_optimizer_states = defaultdict(dict)

    def state_dict(self):
        return {
            "scale": self.get_scale(),
            "growth_factor": self._growth_factor,
            "backoff_factor": self._backoff_factor,
            "growth_interval": self._growth_interval,
            "_growth_tracker": self._growth_tracker,
        }

    def load_state_dict(self, state_dict):
        self._init_scale = state_dict["scale"]
        if self._scale is not None:
            self._scale.fill_(state_dict["scale"])
        self._growth_factor = state_dict["growth_factor"]
        self._backoff_factor = state_dict["backoff_factor"]
        self._growth_interval = state_dict["growth_interval"]
        self._growth_tracker = state_dict["_growth_tracker"]
