#!/usr/bin/env python3
"""Attention A/B: flash kernel (attention_fa.hip) vs torch SDPA (AOTriton)
vs the legacy seq-128 kernel vs the eager matmul chain — fwd-only and
fwd+bwd, BERT-base head shapes at seq 128/256/512 (VERDICT r01 item 8:
"an A/B table like r01-f").

    python tools/attnbench.py [--bs 8] [--heads 12] [--iters 200]
Writes the table to stdout; copy it into profiles/README.md.
"""
import argparse
import math
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def timeit(fn, iters, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def make_qkv(b, s, nh, hd, grad):
    torch.manual_seed(3)
    t = (torch.randn(b, s, 3 * nh * hd) * 0.5).bfloat16().cuda()
    return t.requires_grad_(grad)


def run_shape(b, s, nh, hd, iters, dropout_p):
    from oktopk_amd.ops.fused_attn import _FlashAttention, _FusedAttention

    res = {}
    scale = 1.0 / math.sqrt(hd)

    def sdpa(qkv):
        q, k, v = (qkv.view(b, s, 3, nh, hd).permute(2, 0, 3, 1, 4)
                   .unbind(0))
        o = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, dropout_p=dropout_p)
        return o.transpose(1, 2).reshape(b, s, nh * hd)

    def eager(qkv):
        q, k, v = (qkv.view(b, s, 3, nh, hd).permute(2, 0, 3, 1, 4)
                   .unbind(0))
        p = torch.softmax(q @ k.transpose(-1, -2) * scale, dim=-1)
        if dropout_p:
            p = torch.nn.functional.dropout(p, dropout_p)
        return (p @ v).transpose(1, 2).reshape(b, s, nh * hd)

    variants = {
        "sdpa": sdpa,
        "eager": eager,
        "flash(ours)": lambda qkv: _FlashAttention.apply(
            qkv, None, nh, dropout_p, True),
    }
    if s == 128:
        variants["legacy(ours)"] = lambda qkv: _FusedAttention.apply(
            qkv, None, nh, dropout_p, True)

    for name, fn in variants.items():
        qkv = make_qkv(b, s, nh, hd, grad=False)
        with torch.no_grad():
            res[(name, "fwd")] = timeit(lambda: fn(qkv), iters)
        qkv = make_qkv(b, s, nh, hd, grad=True)
        gy = torch.randn(b, s, nh * hd).bfloat16().cuda()

        def step():
            out = fn(qkv)
            qkv.grad = None
            out.backward(gy)

        res[(name, "fwd+bwd")] = timeit(step, iters)
    return res


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--bs", type=int, default=8)
    ap.add_argument("--heads", type=int, default=12)
    ap.add_argument("--hd", type=int, default=64)
    ap.add_argument("--iters", type=int, default=100)
    ap.add_argument("--dropout", type=float, default=0.0)
    ap.add_argument("--seqs", default="128,256,512")
    args = ap.parse_args()
    assert torch.cuda.is_available()

    print(f"# attention A/B  bs={args.bs} heads={args.heads} hd={args.hd} "
          f"dropout={args.dropout} (us/call)")
    header = None
    for s in [int(x) for x in args.seqs.split(",")]:
        r = run_shape(args.bs, s, args.heads, args.hd, args.iters,
                      args.dropout)
        names = sorted({k[0] for k in r})
        if header != names:
            header = names
            print("| seq | phase | " + " | ".join(names) + " |")
            print("|" + "---|" * (len(names) + 2))
        for phase in ["fwd", "fwd+bwd"]:
            row = [f"{r[(n, phase)]:.1f}" if (n, phase) in r else "-"
                   for n in names]
            print(f"| {s} | {phase} | " + " | ".join(row) + " |", flush=True)


if __name__ == "__main__":
    main()
