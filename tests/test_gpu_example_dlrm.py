"""GPU user-journey: the DLRM public-API example with --fused (MFMA dense
layers through ctx/DataLoader, f32 user model + bf16 fused compute)."""
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.gpu
@pytest.mark.timeout(600)
def test_dlrm_example_fused_learns_on_gpu():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", "dlrm_criteo", "train.py"),
         "--steps", "80", "--batch-size", "1024", "--dim", "32",
         "--rows", "2e4", "--fused"],
        capture_output=True, text=True, timeout=560,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    vals = dict(
        line.split() for line in out.stdout.splitlines()
        if line.startswith("LOSS_")
    )
    assert float(vals["LOSS_LAST"]) < 0.5, vals
    assert float(vals["LOSS_LAST"]) < float(vals["LOSS_FIRST"])
