#!/usr/bin/env python3
"""Experiment sweep driver (parity with the reference's ssh-fanout sweep,
/root/reference/BERT/scripts/driver_sweep.py — torchrun single-node instead
of ssh worker files): runs a grid of {compressor} x {density} benches and
collects the JSON lines into one results file."""
import argparse
import itertools
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_base")
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--compressors", default="oktopk,dense,topkAopt,gaussiank")
    ap.add_argument("--densities", default="0.01,0.001")
    ap.add_argument("--steps", type=int, default=15)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--out", default="sweep_results.jsonl")
    args = ap.parse_args()

    results = []
    for comp, dens in itertools.product(
        args.compressors.split(","), args.densities.split(",")
    ):
        cmd = [sys.executable]
        if args.gpus > 1:
            cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                   f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1"]
        cmd += [os.path.join(REPO, "bench.py"), "--gpus", str(args.gpus),
                "--model", args.model, "--compressor", comp, "--density", dens,
                "--steps", str(args.steps), "--warmup", str(args.warmup),
                "--dense-baseline-steps", "0"]
        print("::", " ".join(cmd), flush=True)
        out = subprocess.run(cmd, capture_output=True, text=True, cwd=REPO)
        line = out.stdout.strip().splitlines()[-1] if out.stdout.strip() else "{}"
        try:
            d = json.loads(line)
        except json.JSONDecodeError:
            d = {"error": out.stderr[-500:], "compressor": comp, "density": dens}
        results.append(d)
        print(json.dumps(d))
    with open(args.out, "w") as f:
        for d in results:
            f.write(json.dumps(d) + "\n")
    print(f"wrote {args.out} ({len(results)} rows)")


if __name__ == "__main__":
    main()
