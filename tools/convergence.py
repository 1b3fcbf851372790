#!/usr/bin/env python3
"""Convergence evidence: BERT-base MLM+NSP on a fixed synthetic batch set,
oktopk density=0.1% vs dense, same seed — loss trajectories should track
(error feedback defers, never loses, gradient mass).  Also logs the EPS
oracle (||sparse - dense||/||dense||, reference PROFILING_NORM)."""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer


def run(compressor, steps, profiling_norm=False, dtype="bf16", density=0.001,
        chunks=1):
    torch.manual_seed(0)
    cfg = EngineConfig.preset("bert", compressor=compressor, density=density,
                              dense_warmup_iters=0, profiling_norm=profiling_norm,
                              pipeline_chunks=chunks)
    tr = Trainer("bert_base", batch_size=8, seq_len=128, cfg=cfg, dtype=dtype)
    losses = []
    for i in range(steps):
        tr.batches.randomize_()
        losses.append(tr.step())
    eps = []
    red = getattr(tr.opt, "reducer", None)
    if red is not None:
        eps = [e for _, e in red.eps_log]
    return losses, eps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=150)
    ap.add_argument("--compressor", default="oktopk")
    ap.add_argument("--density", type=float, default=0.001)
    ap.add_argument("--pipeline-chunks", type=int, default=1)
    args = ap.parse_args()
    t0 = time.time()
    dense_losses, _ = run("dense", args.steps)
    dense32_losses, _ = run("dense", args.steps, dtype="fp32")
    ok_losses, _ = run(args.compressor, args.steps, density=args.density,
                       chunks=args.pipeline_chunks)
    _, eps = run("oktopk", 10, profiling_norm=True)
    print(f"steps={args.steps} wall={time.time()-t0:.0f}s")
    for i in range(0, args.steps, max(1, args.steps // 15)):
        print(f"step {i:4d}  dense {dense_losses[i]:8.4f}  dense-fp32 {dense32_losses[i]:8.4f}  oktopk {ok_losses[i]:8.4f}")
    print(f"final: dense {dense_losses[-1]:.4f}  dense-fp32 {dense32_losses[-1]:.4f}  oktopk {ok_losses[-1]:.4f}")
    print("EPS (first 10 steps):", " ".join(f"{e:.3f}" for e in eps))


if __name__ == "__main__":
    main()
