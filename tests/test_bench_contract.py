"""bench.py contract: runs end-to-end (CPU, tiny config) and prints one
valid JSON line with the required fields."""
import json
import os
import subprocess
import sys

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
