"""The driver contract: bench.py must run single- and multi-process
(gloo on CPU here; the same code path is RCCL on GPU boxes)."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def _check_json_line(line):
    result = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in result, key
    assert result["data"] == "synthetic"
    assert result["value"] > 0
    return result


def test_bench_single_process():
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "1",
         "--warmup", "0", "--num-voxels", "256", "--voxels-per-step",
         "64", "--chunk", "64"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    result = _check_json_line(out.stdout.strip().splitlines()[-1])
    assert result["n_gpus"] == 1


@pytest.mark.slow
def test_bench_two_ranks_gloo():
    env = dict(os.environ)
    import socket
    with socket.socket() as s:      # a currently-free rendezvous port
        s.bind(("127.0.0.1", 0))
        port = str(s.getsockname()[1])
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", port, str(REPO / "bench.py"), "--gpus",
         "2", "--steps", "1", "--warmup", "0", "--num-voxels", "256",
         "--voxels-per-step", "64", "--chunk", "64"],
        capture_output=True, text=True, timeout=600, cwd=str(REPO),
        env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{")]
    assert len(json_lines) == 1, "exactly rank 0 prints the result"
    result = _check_json_line(json_lines[-1])
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "voxel-sharded dp2"
