"""Driver contract: bench.py must emit ONE JSON line with the agreed
schema (BASELINE.json metric/config) from rank 0, CPU fallback included."""

import json
import os
import subprocess
import sys


def test_bench_json_contract():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--batch-size", "8"],
        cwd=repo, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["metric"] == "train_samples_per_s"
    assert d["unit"] == "samples/s"
    assert isinstance(d["value"], float) and d["value"] > 0
    assert d["n_gpus"] == 1 and d["steps"] == 1 and d["warmup"] == 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None
    assert d["data"] == "synthetic"
    assert d["config"]["model"] == "tiger-amazon-beauty"
    assert d["config"]["global_batch"] == 8
    assert d["config"]["parallelism"] == "dp1"
    assert d["config"]["hip_graph"] is False  # no GPU here
    assert d["config"]["seq_len"] == 61
    assert isinstance(d["ms_per_step"], float)


def test_bench_world2_gloo_contract():
    """The driver's torchrun launch shape: 2 ranks over gloo on CPU; ONE
    JSON line from rank 0 with n_gpus=2 and the global batch."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--batch-size", "4"],
        cwd=repo, capture_output=True, text=True, timeout=900, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 8
