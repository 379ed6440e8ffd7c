"""Tracker integration registry (reference tracking.py built-ins)."""

import pytest


def test_all_reference_tracker_integrations_registered():
    """Parity with the reference's 9 built-ins (reference tracking.py:179-1244)
    plus our JSONL tracker; missing packages raise ImportError at construction
    (lazy imports), never at module import."""
    from accelerate_amd.tracking import LOGGER_TYPE_TO_CLASS

    expected = {
        "tensorboard", "wandb", "mlflow", "jsonl",
        "comet_ml", "aim", "clearml", "dvclive", "swanlab", "trackio",
    }
    assert expected == set(LOGGER_TYPE_TO_CLASS)
    for name, cls in LOGGER_TYPE_TO_CLASS.items():
        assert cls.name == name
        assert isinstance(cls.requires_logging_directory, bool)


def test_unavailable_tracker_raises_at_construction():
    import pytest as _pytest

    from accelerate_amd.tracking import AimTracker

    with _pytest.raises(ImportError):
        AimTracker("run", logging_dir=".")
