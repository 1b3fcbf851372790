"""Communication micro-benchmarks (reference: BERT/tests/communication/
{point_to_point,all_to_all,gloo_communication_handler}.py — manual 2-process
throughput scripts).

Run:  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 \
          tools/commbench.py [--sizes-mb 1,16,128] [--iters 20]

Measures p2p send/recv, allreduce, all_to_all_single and
all_gather_into_tensor goodput over the active backend (RCCL on GPU over
xGMI, gloo on CPU).  Rank 0 prints one JSON line per (op, size).
"""
import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def _sync(dev):
    if dev.type == "cuda":
        torch.cuda.synchronize()


def bench(fn, dev, iters, warmup=3):
    for _ in range(warmup):
        fn()
    _sync(dev)
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    _sync(dev)
    dist.barrier()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes-mb", default="1,16,128")
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend)
    rank, world = dist.get_rank(), dist.get_world_size()
    dev = torch.device("cuda", rank % max(torch.cuda.device_count(), 1)) \
        if backend == "nccl" else torch.device("cpu")
    if dev.type == "cuda":
        torch.cuda.set_device(dev)

    for mb in [float(x) for x in args.sizes_mb.split(",")]:
        n = int(mb * (1 << 20) / 4)
        buf = torch.randn(n, device=dev)
        out = torch.empty(world * n, device=dev)

        results = {}
        results["allreduce"] = bench(lambda: dist.all_reduce(buf), dev, args.iters)
        results["all_to_all"] = bench(
            lambda: dist.all_to_all_single(buf.clone(), buf), dev, args.iters
        )
        results["allgather"] = bench(
            lambda: dist.all_gather_into_tensor(out, buf), dev, args.iters
        )

        def p2p():
            # ring: rank -> rank+1
            dst, src = (rank + 1) % world, (rank - 1) % world
            recv = torch.empty_like(buf)
            reqs = [dist.isend(buf, dst), dist.irecv(recv, src)]
            for r in reqs:
                r.wait()

        results["p2p_ring"] = bench(p2p, dev, args.iters)

        if rank == 0:
            for op, sec in results.items():
                # algorithmic bytes moved per rank
                bytes_moved = {
                    "allreduce": 2 * buf.numel() * 4 * (world - 1) / world,
                    "all_to_all": buf.numel() * 4 * (world - 1) / world,
                    "allgather": buf.numel() * 4 * (world - 1),
                    "p2p_ring": buf.numel() * 4,
                }[op]
                print(json.dumps({
                    "op": op, "mb": mb, "world": world, "backend": backend,
                    "ms": round(sec * 1e3, 3),
                    "gbps": round(bytes_moved / sec / 1e9, 2),
                }))

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
