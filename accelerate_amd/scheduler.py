"""LR-scheduler wrapper keeping the schedule in GLOBAL-batch units.

Behavior parity with the reference's AcceleratedScheduler
(reference scheduler.py:25-99), restructured for this stack:

- A ``step()`` call is honored only when the linked optimizers actually
  updated this iteration: accumulation windows tick the internal step
  counter (so schedulers keyed on `_step_count` stay aligned) without
  moving the LR, and fp16 skipped steps freeze the schedule entirely.
- Without ``split_batches`` each process sees 1/N of the global batch, so
  one training iteration advances the schedule N times — the LR curve a
  single-process run with the same GLOBAL batch would produce.
"""

from .state import AcceleratorState, GradientState


class AcceleratedScheduler:
    def __init__(
        self,
        scheduler,
        optimizers,
        step_with_optimizer: bool = True,
        split_batches: bool = False,
        num_batch_shards: int = None,
    ):
        self.scheduler = scheduler
        self.optimizers = list(optimizers) if isinstance(optimizers, (list, tuple)) else [optimizers]
        self.split_batches = split_batches
        self.step_with_optimizer = step_with_optimizer
        self.gradient_state = GradientState()
        # how many ways the global batch is sharded: the dp degree under
        # TP/CP worlds (tp/cp ranks replicate the batch), else num_processes
        self.num_batch_shards = num_batch_shards

    # -- stepping ----------------------------------------------------------

    def _any_optimizer_skipped(self) -> bool:
        return any(getattr(opt, "step_was_skipped", False) for opt in self.optimizers)

    def _advance(self, *args, **kwargs):
        # OneCycle-style schedulers with a hard total_steps raise past the
        # end; clamp when drop_last didn't trim the tail batch.
        limit = getattr(self.scheduler, "total_steps", None)
        if limit is not None and self.scheduler._step_count > limit:
            return
        self.scheduler.step(*args, **kwargs)

    def step(self, *args, **kwargs):
        if not self.step_with_optimizer:
            self.scheduler.step(*args, **kwargs)  # decoupled: always advance
            return
        if not self.gradient_state.sync_gradients:
            # accumulation window: optimizers no-op'd; keep counters aligned
            if self.gradient_state.adjust_scheduler:
                self.scheduler._step_count += 1
            return
        if self._any_optimizer_skipped():
            return  # fp16 inf/nan step: LR stays put
        if self.split_batches:
            ticks = 1
        elif self.num_batch_shards is not None:
            ticks = self.num_batch_shards
        else:
            ticks = AcceleratorState().num_processes
        for _ in range(ticks):
            self._advance(*args, **kwargs)

    # -- passthrough -------------------------------------------------------

    def get_last_lr(self):
        return self.scheduler.get_last_lr()

    def get_lr(self):
        return self.scheduler.get_lr()

    def print_lr(self, *args, **kwargs):
        return self.scheduler.print_lr(*args, **kwargs)

    def state_dict(self):
        return self.scheduler.state_dict()

    def load_state_dict(self, state_dict):
        self.scheduler.load_state_dict(state_dict)
