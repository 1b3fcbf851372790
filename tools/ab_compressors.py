#!/usr/bin/env python3
"""Compressor A/B harness: run every compressor mode
on the same model/shape and report ms/step + engine phase breakdown.

Single-process mode exercises the full selection/merge pipeline (comm is
identity); under torchrun it A/Bs the real collectives.
"""
import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oktopk_amd.comm import init_from_env
from oktopk_amd.config import EngineConfig
from oktopk_amd.trainer import Trainer

COMPRESSORS = ["dense", "oktopk", "topkA", "topkA2", "topkAopt", "topkSA",
               "gtopk", "gaussiank", "gaussiankSA"]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_large")
    ap.add_argument("--batch-size", type=int, default=8)
    ap.add_argument("--density", type=float, default=0.001)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--compressors", default=",".join(COMPRESSORS))
    args = ap.parse_args()

    comm = init_from_env()
    sync = torch.cuda.synchronize if torch.cuda.is_available() else (lambda: None)
    rows = []
    for comp in args.compressors.split(","):
        if comp == "gtopk" and comm.size & (comm.size - 1):
            continue
        cfg = EngineConfig.preset("bert", compressor=comp, density=args.density,
                                  dense_warmup_iters=0)
        tr = Trainer(args.model, batch_size=args.batch_size, seq_len=128,
                     comm=comm, cfg=cfg,
                     dtype="bf16" if torch.cuda.is_available() else "fp32")
        for _ in range(args.warmup):
            tr.step()
        tr.capture_graph() and tr.step()
        red = getattr(tr.opt, "reducer", None)
        if red is not None:
            red.timers = {}
        comm.barrier(); sync()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            tr.step()
        sync(); comm.barrier()
        ms = 1000 * (time.perf_counter() - t0) / args.steps
        phases = {}
        if red is not None:
            for _, ph in red.timers.items():
                for k, v in ph.items():
                    phases[k] = phases.get(k, 0.0) + 1000 * v / args.steps
        rows.append((comp, ms, phases))
        del tr
        if torch.cuda.is_available():
            torch.cuda.empty_cache()
    if comm.rank == 0:
        print(f"# {args.model} bs{args.batch_size} density={args.density} "
              f"P={comm.size} steps={args.steps}")
        print(f"{'compressor':12s} {'ms/step':>9s}  phases(ms)")
        for comp, ms, ph in rows:
            pstr = " ".join(f"{k}={v:.2f}" for k, v in sorted(ph.items()))
            print(f"{comp:12s} {ms:9.2f}  {pstr}")


if __name__ == "__main__":
    main()
