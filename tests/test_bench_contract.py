"""bench.py contract: runs end-to-end (CPU, tiny config) and prints one
valid JSON line with the required fields."""
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from conftest import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "vgg16", "--batch-size", "2",
         "--steps", "2", "--warmup", "1", "--density", "0.05",
         "--dense-baseline-steps", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                  "dtype", "data", "config"):
        assert field in d, field
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["data"] == "synthetic"
    assert d["value"] > 0
    assert d["config"]["compressor"] == "oktopk"
    assert d["config"]["speedup_vs_dense"] is not None


def test_bench_world2_torchrun():
    """The driver's multi-GPU invocation shape: torch.distributed.run with
    --nproc-per-node 2 (gloo on CPU here; RCCL on the GPU box).  Rank 0
    prints the one JSON line; value aggregates over the world."""
    env = dict(os.environ, OKTOPK_BACKEND="gloo")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "2", "--model",
         "vgg16", "--batch-size", "2", "--steps", "2", "--warmup", "1",
         "--density", "0.05", "--dense-baseline-steps", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4


def test_bench_dense_compressor_json():
    """--compressor dense: no dense arm, speedup null, allreduce phase only."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "vgg16", "--batch-size", "2",
         "--steps", "2", "--warmup", "1", "--compressor", "dense"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["config"]["compressor"] == "dense"
    assert d["config"]["speedup_vs_dense"] is None
    assert d["config"]["dense_ms_per_step"] is None
