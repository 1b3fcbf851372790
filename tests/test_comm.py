"""Comm-layer tests over gloo world 2 (the CPU plumbing config of
BASELINE.json #1). Mirrors the reference's 2-process comm micro-tests
(/root/reference/BERT/tests/communication/)."""
import torch

from conftest import run_dist


def _allgatherv(rank):
    from oktopk_amd.comm import Comm
    import torch.distributed as dist

    c = Comm(dist.group.WORLD)
    t = torch.arange(3 + rank * 2, dtype=torch.float32) + rank * 100
    buf, sizes = c.allgatherv(t)
    assert sizes == [3, 5]
    assert torch.equal(buf[:3], torch.tensor([0.0, 1.0, 2.0]))
    assert torch.equal(buf[3:], torch.tensor([100.0, 101.0, 102.0, 103.0, 104.0]))


def test_allgatherv():
    run_dist(_allgatherv, 2)


def _alltoallv(rank):
    from oktopk_amd.comm import Comm
    import torch.distributed as dist

    c = Comm(dist.group.WORLD)
    # rank r sends [r*10+d] of size (d+1) to dest d
    if rank == 0:
        send = torch.tensor([0, 1, 1], dtype=torch.int32)
        send_splits = [1, 2]
    else:
        send = torch.tensor([10, 11, 11], dtype=torch.int32)
        send_splits = [1, 2]
    recv_splits = c.alltoall_sizes(send_splits, c.device)
    out = c.alltoallv(send, send_splits, recv_splits)
    if rank == 0:
        assert recv_splits == [1, 1]
        assert out.tolist() == [0, 10]
    else:
        assert recv_splits == [2, 2]
        assert out.tolist() == [1, 1, 11, 11]


def test_alltoallv():
    run_dist(_alltoallv, 2)


def _sizes(rank):
    from oktopk_amd.comm import Comm
    import torch.distributed as dist

    c = Comm(dist.group.WORLD)
    sizes = c.allgather_sizes(5 + rank, c.device)
    assert sizes.tolist() == [5, 6]


def test_allgather_sizes():
    run_dist(_sizes, 2)


def test_world1_noop_comm():
    from oktopk_amd.comm import Comm

    c = Comm(None)
    assert c.size == 1 and c.rank == 0
    t = torch.randn(4)
    ref = t.clone()
    c.allreduce_(t)
    assert torch.equal(t, ref)
    buf, sizes = c.allgatherv(t)
    assert torch.equal(buf, ref) and sizes == [4]
    out = c.alltoallv(t, [4], [4])
    assert torch.equal(out, ref)


def _ddp_worker(rank):
    import torch
    from oktopk_amd.comm import Comm
    from oktopk_amd.ddp_trainer import ssgd_with_ddp
    import torch.distributed as dist

    model, opt = ssgd_with_ddp("caffe_cifar", comm=Comm(dist.group.WORLD))
    x = torch.randn(2, 3, 32, 32, generator=torch.Generator().manual_seed(rank))
    y = torch.randint(0, 10, (2,), generator=torch.Generator().manual_seed(rank))
    loss = torch.nn.functional.cross_entropy(model(x), y)
    opt.zero_grad()
    loss.backward()
    opt.step()
    # DDP keeps replicas in sync: parameters identical across ranks
    p = next(model.parameters()).detach().clone()
    p0 = p.clone()
    dist.broadcast(p0, src=0)
    assert torch.allclose(p, p0)


def test_ddp_alternative_trainer():
    run_dist(_ddp_worker, 2)


def _async_matches_sync(rank):
    """Async Comm variants == sync results; interleaved enqueue of several
    in-flight collectives drains correctly (docs/overlap_design.md step 1)."""
    import torch
    import torch.distributed as dist
    from oktopk_amd.comm import Comm

    comm = Comm(dist.group.WORLD)
    P = comm.size
    dev = comm.device

    send_sizes = [(rank + d) % 5 + 1 for d in range(P)]
    sync_sizes = comm.alltoall_sizes(send_sizes, dev)
    a1 = comm.alltoall_sizes_async(send_sizes, dev)

    n = 3 + rank
    sync_g = comm.allgather_sizes(n, dev)
    a2 = comm.allgather_sizes_async(n, dev)

    payload = torch.arange(sum(send_sizes), dtype=torch.float32) + 100 * rank
    a3 = comm.alltoallv_async(payload, send_sizes, sync_sizes)

    v = torch.arange(n, dtype=torch.float32) + 10 * rank
    a4 = comm.allgatherv_async(v, [int(x) for x in sync_g])

    # waits deferred past all enqueues, resolved out of order
    got_v, _ = comm.allgatherv(v, sizes=[int(x) for x in sync_g])
    assert torch.equal(a4.wait(), got_v)
    sync_p = comm.alltoallv(payload, send_sizes, sync_sizes)
    assert torch.equal(a3.wait(), sync_p)
    assert [int(x) for x in a2.wait()] == [int(x) for x in sync_g]
    assert a1.wait() == sync_sizes
    assert a1.wait() == sync_sizes  # idempotent


def test_async_collectives_world3():
    run_dist(_async_matches_sync, 3)


def test_async_collectives_world1():
    import torch
    from oktopk_amd.comm import Comm

    c = Comm(None)
    assert c.alltoall_sizes_async([4], torch.device("cpu")).wait() == [4]
    t = torch.arange(4.0)
    assert torch.equal(c.alltoallv_async(t, [4], [4]).wait(), t)
    assert torch.equal(
        c.allgatherv_async(t, [4]).wait(), t)


def _comm_fuzz(rank):
    """Randomized sizes through alltoallv/allgatherv, sync and async, many
    rounds — the exact variable-size patterns the engine emits."""
    import random
    import torch
    import torch.distributed as dist
    from oktopk_amd.comm import Comm

    comm = Comm(dist.group.WORLD)
    P = comm.size
    rng = random.Random(42)  # SAME stream on every rank: sizes agree
    for round_i in range(25):
        sizes = [[rng.randint(0, 9) for _ in range(P)] for _ in range(P)]
        send_sizes = sizes[rank]
        recv_sizes = [sizes[src][rank] for src in range(P)]
        payload = torch.arange(sum(send_sizes), dtype=torch.float32) + 1000 * rank
        if round_i % 2 == 0:
            got = comm.alltoallv(payload, send_sizes, recv_sizes)
        else:
            got = comm.alltoallv_async(payload, send_sizes, recv_sizes).wait()
        # verify contents from each source
        off = 0
        for src in range(P):
            n = recv_sizes[src]
            seg = got[off:off + n]
            base = sum(sizes[src][:rank])
            expect = torch.arange(base, base + n, dtype=torch.float32) + 1000 * src
            assert torch.equal(seg, expect), (round_i, src)
            off += n

        n_mine = rng.randint(0, 7) + rank  # rank-dependent but derived from shared rng
        v = torch.full((n_mine,), float(rank))
        all_sizes = [int(x) for x in comm.allgather_sizes(n_mine, comm.device)]
        if round_i % 2 == 0:
            got2, _ = comm.allgatherv(v, sizes=all_sizes)
        else:
            got2 = comm.allgatherv_async(v, all_sizes).wait()
        expect2 = torch.cat([torch.full((all_sizes[r],), float(r)) for r in range(P)])
        assert torch.equal(got2, expect2), round_i


def test_comm_fuzz_world3():
    run_dist(_comm_fuzz, 3)
