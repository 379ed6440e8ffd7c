"""Launch the 2-process gloo oracle script (the reference launches
test_sync.py the same way)."""

from pathlib import Path

from testing_utils import launch_distributed

SCRIPT = Path(__file__).parent / "distributed_scripts" / "grad_sync_script.py"


def test_distributed_grad_sync_oracle():
    out = launch_distributed(SCRIPT, nproc=2)
    assert "PARITY_PASS" in out
    assert "NOSYNC_PASS" in out
    assert "COLLECTIVES_PASS" in out
    assert "ACCUM_PASS" in out
