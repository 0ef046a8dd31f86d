"""Multi-process distributed bench path: world_size=2 over gloo on CPU —
validates the torchrun contract (rank gathering, rank-0 JSON line) that the
driver uses for the N-GPU scaling bench."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_bench_two_ranks_gloo():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    proc = subprocess.run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node=2",
            "--master-addr=127.0.0.1",
            "--master-port=29517",
            "bench.py",
            "--gpus",
            "2",
            "--steps",
            "4",
            "--warmup",
            "1",
        ],
        cwd=REPO,
        env=env,
        capture_output=True,
        text=True,
        timeout=280,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout}\nstderr:\n{proc.stderr}"
    json_lines = [
        line for line in proc.stdout.splitlines() if line.startswith('{"metric"')
    ]
    assert len(json_lines) == 1, proc.stdout
    result = json.loads(json_lines[0])
    assert result["n_gpus"] == 2
    assert result["steps"] == 4
    assert result["value"] > 0
    assert result["higher_is_better"] is False
    assert result["scaling"] == "weak"
    assert result["reconciles_per_sec"] > 0
    # ONE shared-store operator on rank 0 (contended shape)
    assert result["config"]["parallelism"].startswith("ONE shared-store operator")
    assert result["config"]["global_batch"] == 2  # CR size = N
    # secondary configs ride along (configs #4/#5 + async compose)
    assert "churn_ms" in result
    assert result["contention"]["n_crs"] >= 2
    assert result["async_fabric"]["attach_p50_ms"] > 1000


def test_bench_single_process_no_dist():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--skip-extras"],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=120,
    )
    assert proc.returncode == 0, proc.stderr
    result = json.loads(proc.stdout.strip().splitlines()[-1])
    assert result["n_gpus"] == 1
    assert result["value"] > 0
