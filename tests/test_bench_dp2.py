"""The driver's multi-rank launch path: torchrun --nproc-per-node 2 of
bench.py on CPU (gloo). Guards the exact command shape the round-end
scaling bench uses."""

import json
import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_torchrun_dp2_bench():
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    repo = env["PYTHONPATH"]
    proc = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
            "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
            "--master-port", "29733", os.path.join(repo, "bench.py"),
            "--gpus", "2", "--steps", "2", "--warmup", "1", "--batch", "8",
            "--workers", "1", "--waves", "2", "--upstreams", "1",
        ],
        env=env,
        capture_output=True,
        timeout=280,
        cwd=repo,
    )
    assert proc.returncode == 0, proc.stderr.decode()[-2000:]
    line = [l for l in proc.stdout.decode().splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0
    assert out["scaling"] == "weak"
