"""bench.py contract: runs under torch.distributed.run with world=2 on CPU
(gloo) and prints one valid JSON line (the driver depends on this)."""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _check_json_line(stdout: str, n_gpus: int):
    lines = [l for l in stdout.splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output:\n{stdout[-2000:]}"
    result = json.loads(lines[-1])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in result, f"missing {key}"
    assert result["n_gpus"] == n_gpus
    assert result["value"] > 0
    return result


def test_bench_single_process_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "3", "--warmup", "1",
         "--model", "test-llama", "--prompt-len", "8", "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json_line(out.stdout, 1)


def test_bench_serve_stack_single_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "3", "--warmup", "1",
         "--model", "test-llama", "--prompt-len", "8", "--device", "cpu", "--stack", "serve"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    result = _check_json_line(out.stdout, 1)
    assert result["config"]["stack"] == "serve"


def test_bench_serve_stack_two_rank_cpu():
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", str(port),
         "bench.py", "--gpus", "2", "--steps", "3", "--warmup", "1",
         "--model", "test-llama", "--prompt-len", "8", "--device", "cpu", "--stack", "serve"],
        cwd=REPO, capture_output=True, text=True, timeout=420,
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-2000:])
    result = _check_json_line(out.stdout, 2)
    assert result["config"]["parallelism"] == "swarm-pp2"


def test_bench_two_rank_pipeline_cpu():
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", str(port),
         "bench.py", "--gpus", "2", "--steps", "3", "--warmup", "1",
         "--model", "test-llama", "--prompt-len", "8", "--device", "cpu", "--stack", "pipeline"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    result = _check_json_line(out.stdout, 2)
    assert result["config"]["parallelism"] == "pp2"
