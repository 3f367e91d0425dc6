"""The driver depends on bench.py's exact contract: keep it green on CPU."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json_line(output: str) -> dict:
    for line in reversed(output.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {output[-2000:]}")


def test_bench_single_process():
    proc = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1", "--model", "tiny",
         "--batch", "4", "--seq-len", "32", "--target-batch-size", "16"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    result = _last_json_line(proc.stdout)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in result, f"missing bench contract key {key}"
    assert result["n_gpus"] == 1 and result["steps"] == 2 and result["warmup"] == 1
    assert result["value"] > 0 and result["data"] == "synthetic"
    assert result["scaling"] == "weak" and result["higher_is_better"] is True


def test_bench_torchrun_two_ranks():
    """The driver's N>1 launch shape: torch.distributed.run with gloo on CPU."""
    from hivemind_amd.utils.networking import get_free_port

    port = str(get_free_port())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", port, "bench.py",
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--model", "tiny",
         "--batch", "4", "--seq-len", "32", "--target-batch-size", "16"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    result = _last_json_line(proc.stdout)
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "dp2"


def test_bench_torchrun_two_ranks_dpu_overlapped_rounds():
    """DPU + per-epoch state averaging on 2 gloo ranks: grad and state rounds
    overlap across epochs, the exact pattern whose opposite-order collective
    launches could deadlock before the CollectiveSequencer (VERDICT round 1
    weak #6). 8 steps x tiny epochs force several interleaved rounds; the run
    must finish and report the dist data plane."""
    from hivemind_amd.utils.networking import get_free_port

    port = str(get_free_port())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", port, "bench.py",
         "--gpus", "2", "--steps", "8", "--warmup", "2", "--model", "tiny",
         "--batch", "4", "--seq-len", "32", "--target-batch-size", "16", "--dpu"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    result = _last_json_line(proc.stdout)
    assert result["epochs_in_timed_window"] >= 2
    assert result["grad_data_plane"] == "rccl", result


def test_bench_torchrun_two_ranks_int8_compression():
    """BASELINE config 2 shape: blockwise-int8 quantized gradient averaging
    through the full Optimizer on 2 gloo ranks."""
    from hivemind_amd.utils.networking import get_free_port

    port = str(get_free_port())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", port, "bench.py",
         "--gpus", "2", "--steps", "6", "--warmup", "2", "--model", "tiny",
         "--batch", "4", "--seq-len", "32", "--target-batch-size", "16",
         "--grad-compression", "int8"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    result = _last_json_line(proc.stdout)
    assert result["epochs_in_timed_window"] >= 1
    assert result["grad_data_plane"] == "rccl", result
    assert result["config"]["grad_compression"] == "int8"


def test_bench_torchrun_two_ranks_powersgd():
    """BASELINE config 3 shape: rank-r PowerSGD gradient averaging through
    the full Optimizer on 2 gloo ranks (two chained rounds per global step)."""
    from hivemind_amd.utils.networking import get_free_port

    port = str(get_free_port())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", port, "bench.py",
         "--gpus", "2", "--steps", "6", "--warmup", "2", "--model", "tiny",
         "--batch", "4", "--seq-len", "32", "--target-batch-size", "16",
         "--powersgd-rank", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=900,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    result = _last_json_line(proc.stdout)
    assert result["epochs_in_timed_window"] >= 1
    assert result["config"]["powersgd_rank"] == 2
    assert result["grad_data_plane"] == "rccl", result
