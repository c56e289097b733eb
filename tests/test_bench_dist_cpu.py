"""Run bench.py exactly as the driver will (torch.distributed.run, world=2)
on CPU/gloo: catches multi-rank wiring bugs (DDP + sparse a2a + pipeline
threads) before the round-end 8-GPU scaling run."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(900)
def test_bench_torchrun_world2_cpu():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr=127.0.0.1", "--master-port=29533",
            os.path.join(REPO, "bench.py"),
            "--gpus", "2", "--steps", "3", "--warmup", "1",
            "--batch-size", "64", "--rows", "1e4", "--num-sparse", "4",
            "--dim", "16", "--device", "cpu", "--graph", "0",
        ],
        capture_output=True, text=True, timeout=800, cwd=REPO, env=env,
    )
    assert out.returncode == 0, (out.stderr[-3000:], out.stdout[-500:])
    line = [l for l in out.stdout.strip().splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == 2
    assert res["config"]["global_batch"] == 128
    assert res["value"] > 0
