"""Multi-process-aware logging (reference: logging.py)."""

import functools
import logging
import os


class MultiProcessAdapter(logging.LoggerAdapter):
    """LoggerAdapter that logs on the main process only by default; pass
    ``main_process_only=False`` to log everywhere, ``in_order=True`` to log
    round-robin by rank (reference: logging.py:23-92)."""

    @staticmethod
    def _should_log(main_process_only):
        from .state import PartialState

        state = PartialState()
        return not main_process_only or (main_process_only and state.is_main_process)

    def log(self, level, msg, *args, **kwargs):
        if int(os.environ.get("ACCELERATE_DISABLE_RICH", "0")) == 1:
            pass
        main_process_only = kwargs.pop("main_process_only", True)
        in_order = kwargs.pop("in_order", False)
        kwargs.setdefault("stacklevel", 2)

        if self.isEnabledFor(level):
            if self._should_log(main_process_only):
                msg, kwargs = self.process(msg, kwargs)
                self.logger.log(level, msg, *args, **kwargs)
            elif in_order:
                from .state import PartialState

                state = PartialState()
                for i in range(state.num_processes):
                    if i == state.process_index:
                        msg, kwargs = self.process(msg, kwargs)
                        self.logger.log(level, msg, *args, **kwargs)
                    state.wait_for_everyone()

    @functools.lru_cache(None)
    def warning_once(self, *args, **kwargs):
        self.warning(*args, **kwargs)


def get_logger(name: str, log_level: str = None) -> MultiProcessAdapter:
    """Multi-process logger; level from ``ACCELERATE_LOG_LEVEL`` when not given
    (reference: logging.py:93)."""
    if log_level is None:
        log_level = os.environ.get("ACCELERATE_LOG_LEVEL", None)
    logger = logging.getLogger(name)
    if log_level is not None:
        logger.setLevel(log_level.upper())
        logger.root.setLevel(log_level.upper())
    return MultiProcessAdapter(logger, {})
