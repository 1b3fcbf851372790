"""Random-config engine soak: N trials across every compressor, size,
density, wire dtype, chunking and warmup combination; asserts finite
outputs.  (CPU-runnable; complements the hypothesis fuzz which explores
fewer dimensions more deeply.)

World-1 soak:
    python tools/config_soak.py [--trials 400] [--seed 0]
Distributed soak (gloo, worlds up to 8, with adversarially skewed and
empty per-rank selections — the RCCL-assumption hardening sweep of
docs/round2_plan.md):
    python tools/config_soak.py --worlds 2,3,5,8 --trials 40
"""
import argparse
import os
import random
import socket
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

COMPRESSORS = ["oktopk", "topkA", "topkA2", "topkAopt", "topkSA", "gtopk",
               "gaussiank", "gaussiankconcat", "gaussiankSA", "dense"]

# per-rank gradient shapes designed to stress the variable-size paths:
#   flat    — iid noise
#   empty   — all-zero gradient (selection degenerates to nothing)
#   tiny    — 1e-6 scale (thresholds near denormal)
#   huge    — 1e4 scale (one rank dominates every selection)
#   spike   — all mass in a narrow index band (repartition quantiles collapse)
SKEWS = ["flat", "empty", "tiny", "huge", "spike"]


def make_grad(n, skew, seed):
    g = torch.Generator().manual_seed(seed)
    t = torch.randn(n, generator=g)
    if skew == "empty":
        return torch.zeros(n)
    if skew == "tiny":
        return t * 1e-6
    if skew == "huge":
        return t * 1e4
    if skew == "spike":
        out = torch.zeros(n)
        lo = int(torch.randint(0, max(1, n - max(1, n // 16)), (1,), generator=g))
        hi = min(n, lo + max(1, n // 16))
        out[lo:hi] = t[lo:hi] * 100.0
        return out
    return t


def one_trial(comm, trial, rank, log=print):
    from oktopk_amd import AllReducer, EngineConfig
    from oktopk_amd.config import OkTopkConfig

    rnd = random.Random(10_000 + trial)  # same schedule on every rank
    n = rnd.randint(max(comm.size, 8), 50_000)
    density = rnd.choice([0.001, 0.01, 0.05, 0.3, 1.0])
    comp = rnd.choice(COMPRESSORS)
    chunks = rnd.choice([1, 1, 1, 2, 3, 5]) if comp == "oktopk" else 1
    wire = rnd.choice(["bf16", "fp32"])
    cfg = EngineConfig(
        compressor=comp, density=density, wire_dtype=wire,
        oktopk=OkTopkConfig(dense_warmup_iters=rnd.choice([0, 1]),
                            region_repartition_interval=rnd.choice([1, 2, 64]),
                            pipeline_chunks=chunks))
    eng = AllReducer(comm, cfg)
    # skew assignment is rank-dependent but drawn from the shared schedule
    skews = [rnd.choice(SKEWS) for _ in range(comm.size)]
    try:
        for it in range(3):
            t = make_grad(n, skews[rank], 31 * trial + 7 * rank + it)
            out = eng.run("w", t)
            assert torch.isfinite(out).all(), (comp, n, density, skews)
        return None
    except Exception as e:  # noqa: BLE001 - soak reports, doesn't raise
        return f"FAIL {comp} n={n} d={density} chunks={chunks} wire={wire} " \
               f"skews={skews}: {repr(e)[:200]}"


def _dist_child(rank, world, port, trials, q):
    import torch.distributed as dist

    from oktopk_amd import Comm

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    comm = Comm(dist.group.WORLD)
    fails = []
    for trial in range(trials):
        err = one_trial(comm, trial, rank)
        if err:
            fails.append(err)
    if rank == 0:
        q.put(fails)
    dist.destroy_process_group()


def soak_world(world, trials):
    if world == 1:
        from oktopk_amd import Comm

        comm = Comm(None)
        fails = []
        for trial in range(trials):
            err = one_trial(comm, trial, 0)
            if err:
                fails.append(err)
        return fails
    import torch.multiprocessing as mp

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dist_child, args=(r, world, port, trials, q))
             for r in range(world)]
    for p in procs:
        p.start()
    fails = q.get(timeout=3600)
    for p in procs:
        p.join(60)
        if p.is_alive():
            p.terminate()
            fails.append(f"world={world}: rank hung")
    return fails


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=400)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--worlds", default="1",
                    help="comma list of gloo world sizes (1 = in-process)")
    args = ap.parse_args()
    random.seed(args.seed)
    total_fails = 0
    for world in [int(w) for w in args.worlds.split(",")]:
        fails = soak_world(world, args.trials)
        for f in fails:
            print(f"world={world} {f}")
        print(f"soak world={world}: {args.trials} trials, {len(fails)} failures")
        total_fails += len(fails)
    return 1 if total_fails else 0


if __name__ == "__main__":
    raise SystemExit(main())
