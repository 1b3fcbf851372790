#!/usr/bin/env python3
"""8-GPU flag A/B decision procedure, automated (docs/round2_plan.md
"Flag decision procedure").  Run ON an MI355X node:

    torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
        tools/scale_readiness.py [--steps 40] [--out profiles/scale_r03.json]

Sweeps, in order, on the headline config (bert_base s128 bs8/GPU oktopk):
  1. --pipeline-chunks {1,2,4,8}
  2. --balanced-allgather on/off
  3. --wire-dtype bf16/fp32
  4. density {0.001, 0.01} + the dense baseline (the crossover point)
and reports ms/step per arm plus the winner per knob (3% materiality
threshold, matching the plan).  Works at any world size (world 1 gives
the no-comm baseline of every arm).  Rank 0 writes the JSON.
"""
import argparse
import itertools
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def measure(comm, steps, warmup, **cfg_over):
    from oktopk_amd.config import EngineConfig
    from oktopk_amd.trainer import Trainer

    compressor = cfg_over.pop("compressor", "oktopk")
    density = cfg_over.pop("density", 0.001)
    cfg = EngineConfig.preset("bert", compressor=compressor, density=density,
                              dense_warmup_iters=0, **cfg_over)
    tr = Trainer("bert_base", batch_size=8, seq_len=128, comm=comm, cfg=cfg,
                 dtype="bf16" if torch.cuda.is_available() else "fp32")
    for _ in range(warmup):
        tr.step()
    tr.capture_graph()
    dev_sync = torch.cuda.synchronize if torch.cuda.is_available() else (lambda: None)
    comm.barrier()
    dev_sync()
    t0 = time.perf_counter()
    for _ in range(steps):
        tr.step()
    dev_sync()
    comm.barrier()
    el = torch.tensor([time.perf_counter() - t0], device=comm.device)
    comm.allreduce_(el, op="max")
    del tr
    if torch.cuda.is_available():
        torch.cuda.empty_cache()
    return 1000.0 * float(el.cpu().item()) / steps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=40)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--out", default=os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "profiles", "scale_readiness.json"))
    args = ap.parse_args()

    from oktopk_amd.comm import init_from_env

    comm = init_from_env()
    rank, world = comm.rank, comm.size
    res = {}

    def log(k, v):
        res[k] = round(v, 3)
        if rank == 0:
            print(f"{k:40s} {v:8.3f} ms/step", flush=True)

    # 1. pipeline chunks
    for c in (1, 2, 4, 8):
        log(f"chunks={c}", measure(comm, args.steps, args.warmup,
                                   pipeline_chunks=c))
    # 2. balanced allgather (chunks=1 only)
    log("balanced_allgather=on", measure(comm, args.steps, args.warmup,
                                         balanced_allgather=True))
    # 3. wire dtype
    log("wire=fp32", measure(comm, args.steps, args.warmup, wire_dtype="fp32"))
    log("wire=bf16", measure(comm, args.steps, args.warmup, wire_dtype="bf16"))
    # 4. density crossover vs dense
    for d in (0.001, 0.01):
        log(f"density={d}", measure(comm, args.steps, args.warmup, density=d))
    log("dense", measure(comm, args.steps, args.warmup, compressor="dense"))

    if rank == 0:
        base = res["chunks=1"]
        verdict = {
            "world": world,
            "chunks_winner": min((res[f"chunks={c}"], c) for c in (1, 2, 4, 8))[1],
            "chunks_win_pct": round(100 * (base - min(
                res[f"chunks={c}"] for c in (2, 4, 8))) / base, 2),
            "balanced_wins": res["balanced_allgather=on"] < base * 0.97,
            "wire_bf16_wins": res["wire=bf16"] < res["wire=fp32"] * 0.97,
            "speedup_vs_dense@0.1%": round(res["dense"] / res["density=0.001"], 3),
            "speedup_vs_dense@1%": round(res["dense"] / res["density=0.01"], 3),
            "note": "flip a default only on a reproducible >3% win "
                    "(docs/round2_plan.md); at world 1 every sparse arm is "
                    "pure overhead and the verdicts are not meaningful",
        }
        out = {"config": {"model": "bert_base", "seq_len": 128,
                          "batch_per_gpu": 8, "world": world,
                          "steps": args.steps},
               "ms_per_step": res, "verdict": verdict}
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(out, f, indent=1)
        print(f"wrote {args.out}")
        print(json.dumps(verdict, indent=1))


if __name__ == "__main__":
    main()
