"""Rich traceback install (reference: utils/rich.py) — pretty multi-process
tracebacks when the optional `rich` package is present."""

from .imports import is_rich_available

if is_rich_available():
    from rich.traceback import install

    install(show_locals=False)
else:
    raise ModuleNotFoundError(
        "To use the rich extension, install rich with `pip install rich`"
    )
