#!/usr/bin/env python3
"""Third real-data loss family: word-LM perplexity, sparse vs dense.

The reference's PTB task (LSTM recipe, perplexity eval —
/root/reference/VGG/dl_trainer.py:709-784) needs the PTB files, which this
offline image lacks; REAL English text it does have: this repository's own
documentation.  The corpus is every tracked *.md file (~100 KB of natural
technical prose), split 90/10 into train/valid, tokenized by the PTB
reader, and a PTBLSTM word model trains through the full stack
(DistributedOptimizer -> sparse engine -> SGD + clip) at gloo world 2;
held-out perplexity per epoch is the metric.

    python tools/convergence_real_ptb.py [--epochs 30] [--density 0.02]
Writes profiles/convergence_real_ptb.json.
"""
import argparse
import glob
import json
import math
import os
import socket
import sys
import tempfile

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def build_corpus(tmpdir: str) -> str:
    """Concatenate the repo's markdown into train/valid files (line-level
    90/10 split, deterministic)."""
    lines = []
    for pattern in ("*.md", "docs/*.md", "profiles/README.md",
                    "oktopk_amd/ops/csrc/README.md", "tools/README.md"):
        for path in sorted(glob.glob(os.path.join(REPO, pattern))):
            with open(path, encoding="utf-8") as f:
                lines.extend(l.strip() for l in f if l.strip())
    train, valid = [], []
    for i, l in enumerate(lines):
        (valid if i % 10 == 9 else train).append(l)
    with open(os.path.join(tmpdir, "ptb.train.txt"), "w") as f:
        f.write("\n".join(train))
    with open(os.path.join(tmpdir, "ptb.valid.txt"), "w") as f:
        f.write("\n".join(valid))
    with open(os.path.join(tmpdir, "ptb.test.txt"), "w") as f:
        f.write("\n".join(valid))
    return tmpdir


def train_rank(corpus_dir, compressor, density, epochs, seq_len=32, bs=16,
               lr=2.0, clip=0.25):
    import torch.distributed as dist

    from oktopk_amd.comm import Comm
    from oktopk_amd.config import EngineConfig, OkTopkConfig
    from oktopk_amd.data.ptb import PTBReader, ptb_batchify
    from oktopk_amd.models import create_net
    from oktopk_amd.optimizer import DistributedOptimizer

    comm = Comm(dist.group.WORLD) if dist.is_initialized() else Comm(None)
    rank, world = comm.rank, comm.size
    reader = PTBReader(corpus_dir)
    V = reader.vocab_size
    torch.manual_seed(0)
    model = create_net("lstm", vocab_size=V, emb=256, hidden=256, layers=2,
                       dropout=0.2)
    steps_per_epoch = sum(1 for _ in ptb_batchify(
        reader.train_ids, bs, seq_len, rank=rank, world=world))
    cfg = EngineConfig.preset(
        "lstm", compressor=compressor, density=density,
        dense_warmup_iters=steps_per_epoch)
    opt = DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=lr),
        model.named_parameters(), comm=comm, cfg=cfg, norm_clip=clip)

    crit = torch.nn.CrossEntropyLoss()
    curve = []
    for epoch in range(epochs):
        model.train()
        for x, y in ptb_batchify(reader.train_ids, bs, seq_len,
                                 rank=rank, world=world):
            opt.zero_grad()
            out, _ = model(x)
            loss = crit(out.view(-1, V), y.reshape(-1))
            loss.backward()
            opt.step()
        model.eval()
        with torch.no_grad():
            tot, n = 0.0, 0
            for x, y in ptb_batchify(reader.valid_ids, bs, seq_len):
                out, _ = model(x)
                tot += float(crit(out.view(-1, V), y.reshape(-1))) * y.numel()
                n += y.numel()
        ppl = math.exp(min(tot / max(n, 1), 20.0))
        curve.append(round(ppl, 2))
    opt.stop()
    return curve, V


def _child(rank, world, port, corpus_dir, compressor, density, epochs, q):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        curve, V = train_rank(corpus_dir, compressor, density, epochs)
        if rank == 0:
            q.put((curve, V))
    finally:
        dist.destroy_process_group()


def run(corpus_dir, compressor, density, epochs, world):
    if world == 1:
        return train_rank(corpus_dir, compressor, density, epochs)
    import queue as _q
    import time as _t

    import torch.multiprocessing as mp

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_child, args=(r, world, port, corpus_dir,
                                              compressor, density, epochs, q))
             for r in range(world)]
    for p in procs:
        p.start()
    out = None
    deadline = _t.monotonic() + 3600
    while out is None:
        try:
            out = q.get(timeout=5)
        except _q.Empty:
            if any(p.exitcode not in (None, 0) for p in procs) or \
                    _t.monotonic() > deadline:
                for p in procs:
                    if p.is_alive():
                        p.terminate()
                raise RuntimeError(f"child failed: {[p.exitcode for p in procs]}")
    for p in procs:
        p.join(60)
        if p.is_alive():
            p.terminate()
    return out


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--epochs", type=int, default=30)
    ap.add_argument("--density", type=float, default=0.02)
    ap.add_argument("--world", type=int, default=2)
    ap.add_argument("--tolerance", type=float, default=1.2,
                    help="oktopk final ppl must be <= dense * tolerance")
    ap.add_argument("--out", default=os.path.join(REPO, "profiles",
                                                  "convergence_real_ptb.json"))
    args = ap.parse_args()

    with tempfile.TemporaryDirectory() as tmp:
        corpus = build_corpus(tmp)
        curves = {}
        V = None
        for comp in ("dense", "oktopk"):
            curves[comp], V = run(corpus, comp, args.density, args.epochs,
                                  args.world)
            print(f"{comp:8s} ppl " +
                  " ".join(f"{x:8.1f}" for x in curves[comp][-8:]), flush=True)
    dense_f, ok_f = curves["dense"][-1], curves["oktopk"][-1]
    verdict = {
        "dense_final_ppl": dense_f,
        "oktopk_final_ppl": ok_f,
        "untrained_ppl": V,  # uniform baseline = vocab size
        "tracks_dense": ok_f <= dense_f * args.tolerance and ok_f < V / 3,
    }
    out = {"setup": {"corpus": "repo *.md files (real English prose), "
                               "90/10 line split",
                     "vocab": V, "model": "PTBLSTM emb256 h256 x2",
                     "world": args.world, "density": args.density,
                     "epochs": args.epochs,
                     "metric": "held-out perplexity"},
           "ppl_curves": curves, "verdict": verdict}
    with open(args.out, "w") as f:
        json.dump(out, f, indent=1)
    print("verdict:", verdict)
    return 0 if verdict["tracks_dense"] else 1


if __name__ == "__main__":
    raise SystemExit(main())
