"""Multi-process benchmark contract test (gloo, world_size=2, CPU).

Verifies the distributed path of bench.py the same way the round-end
driver invokes it, and that exactly one JSON line is emitted by rank 0.
"""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port() -> int:
    # ask the kernel for an ephemeral port so the test can't collide with
    # another process holding a fixed rendezvous port
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_bench(nproc: int, steps: int = 2, warmup: int = 0) -> dict:
    if nproc > 1:
        cmd = [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            f"--nproc-per-node={nproc}",
            "--master-addr",
            "127.0.0.1",
            "--master-port",
            str(_free_port()),
            os.path.join(REPO, "bench.py"),
        ]
    else:
        cmd = [sys.executable, os.path.join(REPO, "bench.py")]
    cmd += [
        "--gpus",
        str(nproc),
        "--steps",
        str(steps),
        "--warmup",
        str(warmup),
    ]

    result = subprocess.run(
        cmd, capture_output=True, text=True, timeout=600, cwd=REPO
    )
    assert result.returncode == 0, result.stderr[-2000:]

    json_lines = [
        line
        for line in result.stdout.splitlines()
        if line.startswith("{") and '"metric"' in line
    ]
    assert len(json_lines) == 1, result.stdout
    return json.loads(json_lines[0])


def test_bench_single_process():
    record = run_bench(1)
    assert record["metric"] == "codegen_runs_per_s"
    assert record["n_gpus"] == 1
    assert record["value"] > 0
    assert record["higher_is_better"] is True
    assert record["scaling"] == "weak"
    assert record["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(600)
def test_bench_world_size_two_gloo():
    record = run_bench(2)
    assert record["n_gpus"] == 2
    assert record["value"] > 0
    assert record["config"]["parallelism"] == "dp2"
