"""Optimizer wrapper for the distributed / mixed-precision world.

Behavior parity with the reference's AcceleratedOptimizer
(reference optimizer.py:38-213), structured differently for this stack:

- ``step()`` and ``zero_grad()`` are GATED on
  ``GradientState.sync_gradients`` — inside an accumulation window both
  are silent no-ops, so user training loops need no accumulation branches.
- Under fp16 the step routes through our GradScaler; a skipped step
  (inf/nan grads) is detected by watching the scaler's found-inf verdict
  for THIS optimizer after ``scaler.step`` rather than by monkeypatching
  the step method.
- Optimizer state tensors are migrated to the accelerator device up front
  so a CPU-constructed optimizer never mixes devices.

On MI355X the wrapped optimizer is typically our fused HIP AdamW
(``accelerate_amd.ops.optim.FusedAdamW``); any torch optimizer works.
Subclassing ``torch.optim.Optimizer`` keeps user-side isinstance checks
working — every attribute of the inner optimizer is delegated.
"""

import inspect

import torch

from .state import AcceleratorState, GradientState
from .utils.operations import honor_type

_TRANSIENT_ATTRS = ()  # everything we hold is picklable


def _state_to_device(obj, device):
    """Recursively move optimizer-state tensors (momenta etc.) to device."""
    if isinstance(obj, torch.Tensor):
        return obj.to(device)
    if isinstance(obj, dict):
        return type(obj)({key: _state_to_device(val, device) for key, val in obj.items()})
    if isinstance(obj, (list, tuple)):
        return honor_type(obj, (_state_to_device(item, device) for item in obj))
    return obj


class AcceleratedOptimizer(torch.optim.Optimizer):
    def __init__(self, optimizer, device_placement=True, scaler=None):
        self.optimizer = optimizer
        self.scaler = scaler
        self.accelerator_state = AcceleratorState()
        self.gradient_state = GradientState()
        self.device_placement = device_placement
        self._last_step_skipped = False
        # per-rank-state marker (tp/fsdp shard moments) set on the RAW
        # optimizer before wrapping must survive onto the wrapper, which is
        # what checkpointing inspects
        self._sharded = getattr(optimizer, "_sharded", False)
        if device_placement:
            migrated = _state_to_device(optimizer.state_dict(), self.accelerator_state.device)
            optimizer.load_state_dict(migrated)

    # -- delegation (state/param_groups/defaults live on the inner optimizer)

    @property
    def state(self):
        return self.optimizer.state

    @state.setter
    def state(self, value):
        self.optimizer.state = value

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    @param_groups.setter
    def param_groups(self, value):
        self.optimizer.param_groups = value

    @property
    def defaults(self):
        return self.optimizer.defaults

    @defaults.setter
    def defaults(self, value):
        self.optimizer.defaults = value

    def add_param_group(self, param_group):
        self.optimizer.add_param_group(param_group)

    def state_dict(self):
        return self.optimizer.state_dict()

    def load_state_dict(self, state_dict):
        self.optimizer.load_state_dict(state_dict)

    def train(self):
        inner = getattr(self.optimizer, "train", None)
        if callable(inner):
            inner()

    def eval(self):
        inner = getattr(self.optimizer, "eval", None)
        if callable(inner):
            inner()

    # -- the accumulation-gated core --------------------------------------

    def zero_grad(self, set_to_none=None):
        if not self.gradient_state.sync_gradients:
            return  # grads keep accumulating inside the window
        sig = inspect.signature(self.optimizer.zero_grad)
        if "set_to_none" in sig.parameters:
            self.optimizer.zero_grad(set_to_none=True if set_to_none is None else set_to_none)
        elif set_to_none is not None:
            raise ValueError("this optimizer's zero_grad() does not accept `set_to_none`")
        else:
            self.optimizer.zero_grad()

    def step(self, closure=None):
        if not self.gradient_state.sync_gradients:
            return  # accumulation window: no parameter update yet
        if self.scaler is None:
            self.optimizer.step(closure)
            self._last_step_skipped = False
            return
        # fp16: the scaler unscales, checks for inf/nan, and either runs or
        # skips the real step. Ask the scaler for its verdict instead of
        # instrumenting optimizer.step: per-optimizer found_inf flags are
        # recorded by scaler.step and readable until update() clears them.
        self.scaler.step(self.optimizer, closure)
        self._last_step_skipped = self._scaler_found_inf()
        self.scaler.update()

    def _scaler_found_inf(self) -> bool:
        record = getattr(self.scaler, "_per_optimizer_states", {}).get(id(self.optimizer))
        if not record:
            return False
        if "found_inf" in record:  # our HIP GradScaler: one device flag
            return bool(record["found_inf"].item())
        flags = record.get("found_inf_per_device", {})  # torch.amp.GradScaler
        return any(bool(flag.item()) for flag in flags.values())

    @property
    def step_was_skipped(self) -> bool:
        """True when the last ``step()`` was dropped by the GradScaler
        (non-finite grads). The AcceleratedScheduler consults this so the
        LR schedule doesn't advance on skipped steps."""
        return self._last_step_skipped

    def _switch_parameters(self, parameters_map):
        """Re-point param groups at swapped parameters (the FSDP master-shard
        optimizer-param swap; reference accelerator.py:1714-1726)."""
        for group in self.optimizer.param_groups:
            group["params"] = [parameters_map.get(p, p) for p in group["params"]]
