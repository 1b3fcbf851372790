"""Engine edge cases that only bite at scale: all-zero gradients (empty
selections everywhere), tiny tensors (n < world), density=1.0, empty
alltoallv legs."""
import pytest
import torch

from conftest import run_dist

from oktopk_amd import AllReducer, Comm, EngineConfig
from oktopk_amd.config import OkTopkConfig


def make(comp="oktopk", density=0.05, **okkw):
    okkw.setdefault("dense_warmup_iters", 0)
    return AllReducer(Comm(None), EngineConfig(compressor=comp, density=density,
                                               oktopk=OkTopkConfig(**okkw)))


def test_all_zero_gradient_world1():
    eng = make()
    t = torch.zeros(4096)
    out = eng.run("w", t)
    assert out.abs().sum() == 0
    # and again (thresholds now zero/degenerate)
    out = eng.run("w", torch.zeros(4096))
    assert torch.isfinite(out).all()


def test_tiny_tensor_world1():
    eng = make(density=1.0)
    for it in range(3):
        t = torch.randn(3, generator=torch.Generator().manual_seed(it))
        out = eng.run("w", t)
        assert torch.isfinite(out).all()


def test_density_one_world1():
    eng = make(density=1.0)
    t = torch.randn(1000, generator=torch.Generator().manual_seed(0))
    ref = t.clone()
    out = eng.run("w", t)
    # everything above-threshold except ties at the minimum |value|
    assert (out != 0).sum() >= 990
    nz = out.nonzero().view(-1)
    assert torch.allclose(out[nz], ref[nz])


def _zero_grads_dist(rank):
    import torch.distributed as dist

    eng = AllReducer(Comm(dist.group.WORLD),
                     EngineConfig(compressor="oktopk", density=0.05,
                                  oktopk=OkTopkConfig(dense_warmup_iters=0)))
    for it in range(3):
        out = eng.run("w", torch.zeros(2048))
        assert out.abs().sum() == 0


def test_all_zero_gradient_world2():
    run_dist(_zero_grads_dist, 2)


def _one_rank_zero(rank):
    """One rank has zero grads (empty selection), the other doesn't —
    exercises empty alltoallv legs and empty allgather segments."""
    import torch.distributed as dist

    eng = AllReducer(Comm(dist.group.WORLD),
                     EngineConfig(compressor="oktopk", density=0.05,
                                  oktopk=OkTopkConfig(dense_warmup_iters=0)))
    for it in range(4):
        if rank == 0:
            t = torch.zeros(2048)
        else:
            t = torch.randn(2048, generator=torch.Generator().manual_seed(it))
        out = eng.run("w", t)
        assert torch.isfinite(out).all()


def test_one_rank_all_zero_world2():
    run_dist(_one_rank_zero, 2)


def _tiny_dist(rank):
    import torch.distributed as dist

    eng = AllReducer(Comm(dist.group.WORLD),
                     EngineConfig(compressor="oktopk", density=1.0,
                                  oktopk=OkTopkConfig(
                                      dense_warmup_iters=0,
                                      region_repartition_interval=1)))
    for it in range(3):
        t = torch.randn(3, generator=torch.Generator().manual_seed(100 * rank + it))
        out = eng.run("w", t)
        assert torch.isfinite(out).all()


def test_tiny_tensor_world2_repartition_every_step():
    run_dist(_tiny_dist, 2)


def _dense_fallback_odd_n(rank):
    """topkSA dense-region fallback with n not divisible by P (the last
    uniform region is LARGER than ceil(n/P)) must reproduce the dense mean."""
    import torch.distributed as dist

    eng = AllReducer(Comm(dist.group.WORLD),
                     EngineConfig(compressor="topkSA", density=1.0,
                                  oktopk=OkTopkConfig(dense_warmup_iters=0)))
    n = 1003  # 1003 = 2*501 + 1 -> regions [501, 502]
    t = torch.randn(n, generator=torch.Generator().manual_seed(rank))
    ref = sum(
        torch.randn(n, generator=torch.Generator().manual_seed(r)) for r in range(2)
    ) / 2
    out = eng.run("w", t.clone())
    # density 1.0 selects everything except each rank's strict-> minimum
    # element (which stays in its residual, faithful EF semantics) -> the
    # fallback must reproduce the dense mean everywhere but <= P coords
    bad = ((out - ref).abs() > 1e-5).sum().item()
    assert bad <= 2, bad
    # and the region-stitching must be structurally right: large errors
    # (region offset bugs) would corrupt whole blocks
    assert (out - ref).abs().max() < 1.0


def test_topkSA_dense_fallback_odd_n_world2():
    run_dist(_dense_fallback_odd_n, 2)


def test_chunked_rejects_balanced_allgather():
    """pipeline_chunks>1 + balanced_allgather is an unsupported combination
    and must fail loudly, mirroring the profiling_norm guard."""
    import pytest

    from oktopk_amd.allreducer import AllReducer
    from oktopk_amd.comm import Comm
    from oktopk_amd.config import EngineConfig, OkTopkConfig

    eng = AllReducer(
        Comm(None),
        EngineConfig(compressor="oktopk",
                     oktopk=OkTopkConfig(dense_warmup_iters=0,
                                         pipeline_chunks=2,
                                         balanced_allgather=True)))
    with pytest.raises(ValueError, match="balanced_allgather"):
        eng.run("w", torch.randn(64))


def _adversarial_repartition(rank):
    """World-8 repartition with adversarially skewed index distributions
    (VERDICT r01 weak 3): all ranks concentrate mass in ONE narrow band, a
    degenerate rank selects nothing (uniform quantile contribution), one
    rank dominates magnitudes.  Boundaries must stay monotone and the EF
    mass invariant must hold."""
    import torch.distributed as dist

    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig

    world = dist.get_world_size()
    n = 16384
    cfg = EngineConfig(
        compressor="oktopk", density=0.02,
        oktopk=OkTopkConfig(dense_warmup_iters=0,
                            region_repartition_interval=1))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    in_sum = torch.zeros(n)
    out_sum = torch.zeros(n)
    for it in range(6):
        g = torch.Generator().manual_seed(31 * rank + it)
        t = torch.zeros(n)
        if rank == 0:
            pass  # degenerate: empty selection every iteration
        elif rank == 1:
            t = torch.randn(n, generator=g) * 1e4  # magnitude-dominant
        else:
            # everyone else: all mass inside the same narrow band
            band = slice(n // 2, n // 2 + n // 64)
            t[band] = torch.randn(n // 64, generator=g) * 10
        in_sum += t
        out = eng.run("w", t.clone())
        out_sum += out
        assert torch.isfinite(out).all()
        b = eng.states["w"].boundaries
        assert b is not None and bool((b[1:] >= b[:-1]).all()), b
        assert int(b[0]) == 0 and int(b[-1]) == n
    dist.all_reduce(in_sum)
    res = eng.states["w"].residual.clone()
    dist.all_reduce(res)
    err = (world * out_sum + res - in_sum).abs().max().item()
    assert err < 2e-2, f"EF mass leak under skew: {err}"


def test_adversarial_repartition_world8():
    run_dist(_adversarial_repartition, 8)
