"""End-to-end CLI under torchrun (2 gloo ranks, CPU): train 1 epoch with
checkpointing, then resume with --pretrain."""
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from conftest import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_cli(tmp, extra):
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()), "-m", "oktopk_amd.train",
           "--dnn", "resnet20", "--batch-size", "2", "--density", "0.05",
           "--dense-warmup", "0",  # exercise the SPARSE path, not warmup
           "--iters-per-epoch", "3", "--max-epochs", "1", "--dtype", "fp32",
           "--logdir", os.path.join(tmp, "logs"),
           "--checkpoint-dir", os.path.join(tmp, "ck")] + extra
    return subprocess.run(cmd, cwd=REPO, capture_output=True, text=True, timeout=420)


def test_cli_train_and_resume(tmp_path):
    tmp = str(tmp_path)
    out = _run_cli(tmp, [])
    assert out.returncode == 0, out.stderr[-2000:]
    ck = os.path.join(tmp, "ck", "checkpoint.epoch.0.pth")
    assert os.path.exists(ck)
    metrics = os.path.join(tmp, "logs", "metrics.jsonl")
    lines = [json.loads(l) for l in open(metrics)]
    assert any(d["tag"] == "train/loss" for d in lines)

    out2 = _run_cli(tmp, ["--pretrain", ck, "--max-epochs", "1"])
    assert out2.returncode == 0, out2.stderr[-2000:]


def test_cli_chunked_balanced_flags(tmp_path):
    """--pipeline-chunks and --balanced-allgather through the full CLI
    (2 gloo ranks): both engine modes train end-to-end."""
    tmp = str(tmp_path)
    out = _run_cli(tmp, ["--pipeline-chunks", "2"])
    assert out.returncode == 0, out.stderr[-2000:]
    out2 = _run_cli(tmp, ["--balanced-allgather"])
    assert out2.returncode == 0, out2.stderr[-2000:]


def test_baseline_config1_vgg_cpu_world2(tmp_path):
    """BASELINE.json config #1 verbatim: VGG-16 Ok-Topk density=1%, CPU,
    world_size=2 (plumbing check, no GPU)."""
    tmp = str(tmp_path)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()), "-m", "oktopk_amd.train",
           "--dnn", "vgg16", "--batch-size", "4", "--density", "0.01",
           "--compressor", "oktopk", "--dense-warmup", "0",
           "--iters-per-epoch", "2",
           "--max-epochs", "1", "--dtype", "fp32",
           "--logdir", os.path.join(tmp, "logs")]
    out = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                         timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    metrics = os.path.join(tmp, "logs", "metrics.jsonl")
    lines = [json.loads(l) for l in open(metrics)]
    assert any(d["tag"] == "train/loss" for d in lines)


def test_cli_elastic_happy_path(tmp_path):
    """--elastic arms the agent/runner without disturbing a failure-free
    run (the failure path itself is covered by the ElasticRunner death
    tests in test_elastic_checkpoint.py)."""
    tmp = str(tmp_path)
    out = _run_cli(tmp, ["--elastic"])
    assert out.returncode == 0, out.stderr[-2000:]
    assert os.path.exists(os.path.join(tmp, "ck", "checkpoint.epoch.0.pth"))
