"""The DLRM public-API example trains: loss falls well below ln(2) on its
mixed dense+sparse synthetic target (the user-journey twin of bench.py)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_dlrm_example_learns():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "dlrm_criteo", "train.py"),
         "--steps", "80", "--batch-size", "256", "--dim", "16",
         "--num-sparse", "8", "--rows", "2e4"],
        capture_output=True, text=True, timeout=560,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    vals = {}
    for line in out.stdout.splitlines():
        if line.startswith("LOSS_"):
            k, v = line.split()
            vals[k] = float(v)
    assert set(vals) == {"LOSS_FIRST", "LOSS_LAST"}, out.stdout[-500:]
    assert vals["LOSS_LAST"] < 0.55, vals  # well below ln(2) = 0.693
    assert vals["LOSS_LAST"] < vals["LOSS_FIRST"]
