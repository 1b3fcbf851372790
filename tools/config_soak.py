"""Random-config engine soak: N trials across every compressor, size,
density, wire dtype, chunking and warmup combination; asserts finite
outputs.  (CPU-runnable; complements the hypothesis fuzz which explores
fewer dimensions more deeply.)

    python tools/config_soak.py [--trials 400] [--seed 0]
"""
import argparse
import os
import random
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from oktopk_amd import AllReducer, Comm, EngineConfig
from oktopk_amd.config import OkTopkConfig

COMPRESSORS = ["oktopk", "topkA", "topkA2", "topkAopt", "topkSA", "gtopk",
               "gaussiank", "gaussiankconcat", "gaussiankSA", "dense"]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=400)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    random.seed(args.seed)
    fails = 0
    for trial in range(args.trials):
        n = random.randint(1, 50_000)
        density = random.choice([0.001, 0.01, 0.05, 0.3, 1.0])
        comp = random.choice(COMPRESSORS)
        chunks = random.choice([1, 1, 1, 2, 3, 5]) if comp == "oktopk" else 1
        wire = random.choice(["bf16", "fp32"])
        cfg = EngineConfig(
            compressor=comp, density=density, wire_dtype=wire,
            oktopk=OkTopkConfig(dense_warmup_iters=random.choice([0, 1]),
                                pipeline_chunks=chunks))
        eng = AllReducer(Comm(None), cfg)
        g = torch.Generator().manual_seed(trial)
        scale = random.choice([1.0, 1e-5, 1e4])
        try:
            for _ in range(3):
                out = eng.run("w", torch.randn(n, generator=g) * scale)
                assert torch.isfinite(out).all(), (comp, n, density)
        except Exception as e:  # noqa: BLE001 - soak reports, doesn't raise
            fails += 1
            print("FAIL", comp, n, density, chunks, wire, repr(e)[:200])
    print(f"soak done: {args.trials} trials, {fails} failures")
    return 1 if fails else 0


if __name__ == "__main__":
    raise SystemExit(main())
