"""bench.py driver-contract tests: one JSON line from rank 0 with the agreed
schema, single-process and under torch.distributed.run (gloo, 2 ranks)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _last_json(output: str) -> dict:
    for line in reversed(output.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{output[-2000:]}")


@pytest.mark.timeout(300)
def test_bench_serve_schema():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "llama-tiny", "--batch", "2",
         "--prompt-len", "16", "--steps", "4", "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json(out.stdout)
    assert REQUIRED <= set(r)
    assert r["metric"] == "served_tokens_per_s"
    assert r["scaling"] == "weak" and r["higher_is_better"] is True
    assert r["data"] == "synthetic"
    assert r["config"]["model"] == "llama-tiny"
    assert r["value"] > 0 and r["ms_per_step"] > 0


@pytest.mark.timeout(300)
def test_bench_train_schema():
    out = subprocess.run(
        [sys.executable, "bench.py", "--mode", "train", "--model",
         "llama-tiny", "--steps", "2", "--warmup", "1", "--seq-len", "32"],
        cwd=ROOT, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json(out.stdout)
    assert REQUIRED <= set(r)
    assert r["metric"] == "fine_tune_tokens_per_s"


@pytest.mark.timeout(600)
def test_bench_torchrun_2rank_gloo():
    """The exact launch shape the driver uses for N>1 (loopback rendezvous)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29912", "bench.py", "--gpus", "2", "--model",
         "llama-tiny", "--batch", "2", "--prompt-len", "16", "--steps", "4",
         "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=580)
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json(out.stdout)
    assert r["n_gpus"] == 2
    assert r["config"]["parallelism"] == "dp2"
    assert r["config"]["global_batch"] == 4  # 2 per replica × 2 replicas


@pytest.mark.timeout(600)
def test_bench_torchrun_tp2_gloo():
    """TP serving bench: 2 ranks serve ONE shared batch (tokens counted
    once, parallelism tp2) — the llama2-70b TP=8 accounting at CPU scale."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29913", "bench.py", "--gpus", "2", "--tp", "2",
         "--model", "llama-tiny", "--batch", "2", "--prompt-len", "16",
         "--steps", "4", "--warmup", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=580)
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json(out.stdout)
    assert r["config"]["parallelism"] == "tp2"
    assert r["config"]["global_batch"] == 2  # ONE shared batch, not 2x


@pytest.mark.timeout(600)
def test_bench_train_torchrun_2rank_gloo():
    """Train bench under the driver's N>1 launch shape — exercises the
    ZeRO-1 path (reduce-scatter/shard-AdamW/all-gather) end to end through
    bench.py on gloo."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29914", "bench.py", "--gpus", "2", "--mode",
         "train", "--model", "llama-tiny", "--steps", "2", "--warmup", "1",
         "--seq-len", "32"],
        cwd=ROOT, capture_output=True, text=True, timeout=580)
    assert out.returncode == 0, out.stderr[-2000:]
    r = _last_json(out.stdout)
    assert r["metric"] == "fine_tune_tokens_per_s"
    assert r["n_gpus"] == 2
