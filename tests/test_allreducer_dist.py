"""Distributed engine tests: gloo world 2/4 on CPU.

The EPS oracle mirrors the reference's built-in correctness check
(settings.PROFILING_NORM, /root/reference/VGG/allreducer.py:584-606,1072-1080):
the sparse result is compared against the global-top-k of the dense-allreduced
gradient.  The mass-conservation test is the error-feedback invariant: no
gradient mass is ever lost, only deferred through residuals.
"""
import torch

from conftest import run_dist

N = 8192
DENSITY = 0.02
ITERS = 8


def _grad(rank, it, n=N):
    g = torch.Generator().manual_seed(1000 * rank + it)
    return torch.randn(n, generator=g)


def _dense_mean(it, world, n=N):
    return sum(_grad(r, it, n) for r in range(world)) / world


def _dense_ok(rank):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    import torch.distributed as dist

    eng = AllReducer(Comm(dist.group.WORLD), EngineConfig(compressor="dense"))
    world = dist.get_world_size()
    for it in range(3):
        t = _grad(rank, it)
        out = eng.run("w", t)
        ref = _dense_mean(it, world)
        assert torch.allclose(out, ref, atol=1e-5), (it, (out - ref).abs().max())


def test_dense_world2():
    run_dist(_dense_ok, 2)


def _mass_conservation(rank, comp):
    """P * sum(results) + sum_r(residual_r) == sum_r sum_it(grad_r_it)."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    world = dist.get_world_size()
    cfg = EngineConfig(
        compressor=comp,
        density=DENSITY,
        oktopk=OkTopkConfig(dense_warmup_iters=2),
    )
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    out_sum = torch.zeros(N)
    in_sum = torch.zeros(N)
    for it in range(ITERS):
        t = _grad(rank, it)
        in_sum += t
        out = eng.run("w", t.clone())
        out_sum += out
    # allreduce the inputs and residuals to get global mass
    dist.all_reduce(in_sum)
    res = eng.states["w"].residual.clone()
    dist.all_reduce(res)
    lhs = world * out_sum + res
    err = (lhs - in_sum).abs().max().item()
    assert err < 1e-3, f"{comp}: EF mass leak {err}"


def test_mass_conservation_oktopk():
    run_dist(_mass_conservation, 2, args=("oktopk",))


def test_mass_conservation_topkA():
    run_dist(_mass_conservation, 2, args=("topkA",))


def test_mass_conservation_gtopk():
    run_dist(_mass_conservation, 2, args=("gtopk",))


def test_mass_conservation_topkSA():
    run_dist(_mass_conservation, 2, args=("topkSA",))


def test_mass_conservation_gaussiank():
    run_dist(_mass_conservation, 2, args=("gaussiank",))


def _eps_oracle(rank, comp, eps_bound):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    cfg = EngineConfig(
        compressor=comp,
        density=DENSITY,
        profiling_norm=True,
        oktopk=OkTopkConfig(dense_warmup_iters=0),
    )
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    for it in range(ITERS):
        eng.run("w", _grad(rank, it))
    eps = [e for _, e in eng.eps_log]
    assert len(eps) == ITERS
    # sparse result should be a reasonable approximation of the dense mean:
    # ||sparse - dense|| / ||dense|| < eps_bound (density=2% of random noise
    # keeps ~sqrt relative mass; this bounds gross algorithmic errors, e.g.
    # wrong indices, double counting, lost regions)
    assert all(e < eps_bound for e in eps), eps
    # the selected entries must MATCH the dense values there: error
    # restricted to selected support must be ~0 (checked via conservation
    # tests); here ensure eps is not trivially 1.0 (empty result)
    assert min(eps) < 0.995, eps


def test_eps_oracle_oktopk():
    run_dist(_eps_oracle, 2, args=("oktopk", 1.05))


def _dense_profiling(rank):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    import torch.distributed as dist

    cfg = EngineConfig(compressor="dense", profiling_norm=True)
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    for it in range(2):
        eng.run("w", _grad(rank, it))


def test_eps_oracle_dense_is_zero():
    run_dist(_dense_profiling, 2)


def _oktopk_selected_match_dense(rank):
    """On the selected support, ok-topk values must equal the dense mean
    exactly (the algorithm sums exact contributions of ranks whose local
    threshold admitted the index; with warmup residuals cleared and iteration
    0, every rank contributes its full value at globally-selected indices
    only if locally selected — so compare against the dense mean restricted
    to indices where ALL ranks selected locally)."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    world = dist.get_world_size()
    cfg = EngineConfig(
        compressor="oktopk", density=DENSITY, oktopk=OkTopkConfig(dense_warmup_iters=0)
    )
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    t = _grad(rank, 0)
    out = eng.run("w", t.clone())
    dense = _dense_mean(0, world)
    # indices where every rank's |g| cleared its own local tau: at iteration 0
    # tau_r = exact kth value of rank r's grad
    k = int(N * DENSITY)
    taus = [torch.topk(_grad(r, 0).abs(), k).values[-1].item() for r in range(world)]
    all_sel = torch.ones(N, dtype=torch.bool)
    for r in range(world):
        all_sel &= _grad(r, 0).abs() > taus[r]
    support = out.nonzero().view(-1)
    both = all_sel[support]
    sel = support[both]
    if sel.numel():
        assert torch.allclose(out[sel], dense[sel], atol=1e-5)


def test_oktopk_values_exact_on_full_support():
    run_dist(_oktopk_selected_match_dense, 2)


def test_world4_oktopk():
    run_dist(_mass_conservation, 4, args=("oktopk",))


def _bf16_wire(rank):
    """bf16 wire values: result must match the fp32-wire engine within bf16
    rounding on the selected entries."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    def build(wire):
        return AllReducer(
            Comm(dist.group.WORLD),
            EngineConfig(compressor="oktopk", density=DENSITY, wire_dtype=wire,
                         oktopk=OkTopkConfig(dense_warmup_iters=0)),
        )

    e32, e16 = build("fp32"), build("bf16")
    for it in range(4):
        t = _grad(rank, it)
        o32 = e32.run("w", t.clone())
        o16 = e16.run("w", t.clone())
        nz = o32.nonzero().view(-1)
        # selected support should largely agree; values within bf16 rounding
        assert torch.allclose(o16[nz], o32[nz], rtol=0.02, atol=1e-3), (
            (o16[nz] - o32[nz]).abs().max()
        )


def test_bf16_wire_world2():
    run_dist(_bf16_wire, 2)


def _bf16_wire_all_compressors(rank):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    for comp in ("topkA", "gtopk", "topkSA", "gaussiank"):
        eng = AllReducer(
            Comm(dist.group.WORLD),
            EngineConfig(compressor=comp, density=0.05, wire_dtype="bf16",
                         oktopk=OkTopkConfig(dense_warmup_iters=0)),
        )
        for it in range(3):
            out = eng.run("w", _grad(rank, it, 4096))
            assert torch.isfinite(out).all()


def test_bf16_wire_topkA_gtopk_world2():
    run_dist(_bf16_wire_all_compressors, 2)


def _balanced_equivalence(rank):
    """Balanced round-2 redistribution (BERT/bert/allreducer.py:615-715)
    returns the SAME entries in the SAME order as the pad-to-max path, so
    the full oktopk stream is bit-identical under either setting."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    world = dist.get_world_size()
    outs = {}
    for balanced in (False, True):
        cfg = EngineConfig(
            compressor="oktopk", density=DENSITY,
            oktopk=OkTopkConfig(dense_warmup_iters=0,
                                region_repartition_interval=4,
                                balanced_allgather=balanced),
        )
        eng = AllReducer(Comm(dist.group.WORLD), cfg)
        acc = []
        for it in range(ITERS):
            t = _grad(rank, it)
            acc.append(eng.run("w", t).clone())
        outs[balanced] = acc
    for it, (a, b) in enumerate(zip(outs[False], outs[True])):
        assert torch.equal(a, b), (it, (a - b).abs().max())


def test_balanced_allgather_world2():
    run_dist(_balanced_equivalence, 2)


def test_balanced_allgather_world4():
    run_dist(_balanced_equivalence, 4)


def _balanced_fp32_wire(rank):
    """Balanced redistribution with fp32 wire (packed stride 2n) — the
    packing geometry differs from bf16; both must agree with pad-to-max."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    outs = {}
    for balanced in (False, True):
        cfg = EngineConfig(
            compressor="oktopk", density=DENSITY, wire_dtype="fp32",
            oktopk=OkTopkConfig(dense_warmup_iters=0,
                                region_repartition_interval=3,
                                balanced_allgather=balanced),
        )
        eng = AllReducer(Comm(dist.group.WORLD), cfg)
        acc = []
        for it in range(6):
            acc.append(eng.run("w", _grad(rank, it)).clone())
        outs[balanced] = acc
    for a, b in zip(outs[False], outs[True]):
        assert torch.equal(a, b)


def test_balanced_fp32_wire_world3():
    run_dist(_balanced_fp32_wire, 3)


def _elastic_shrink_midstream(rank):
    """set_comm after a shrink: boundaries reset, stream continues finite
    (reference err_callback -> update_nworker, VGG/dl_trainer.py:472)."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    world = dist.get_world_size()
    cfg = EngineConfig(compressor="oktopk", density=DENSITY,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    for it in range(4):
        eng.run("w", _grad(rank, it))
    # shrink to the first half of ranks (gloo new_group)
    half = max(1, world // 2)
    group = dist.new_group(ranks=list(range(half)))
    if rank < half:
        eng.set_comm(Comm(group))
        assert eng.states["w"].boundaries is None  # reset on shrink
        for it in range(4, 8):
            out = eng.run("w", _grad(rank, it))
            assert torch.isfinite(out).all()


def test_elastic_shrink_midstream_world4():
    run_dist(_elastic_shrink_midstream, 4)


def _gtopk_world8(rank):
    """3-round binomial tree at the 8-GPU xGMI island shape.  gTopK is NOT
    EF-complete beyond one merge round (entries truncated at intermediate
    merges are unrecoverable — the reference's add_residuals credits only
    against the FINAL set, VGG/compression.py:151-160), so the invariant
    here is: finite, rank-identical results, and the strict-EF subset
    (entries absent from the final set) is credited."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    cfg = EngineConfig(compressor="gtopk", density=DENSITY,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    for it in range(4):
        out = eng.run("w", _grad(rank, it))
        assert torch.isfinite(out).all()
        # all ranks hold the identical broadcast result
        ref = out.clone()
        dist.broadcast(ref, src=0)
        assert torch.equal(out, ref)
    assert torch.isfinite(eng.states["w"].residual).all()


def test_world8_gtopk():
    run_dist(_gtopk_world8, 8)


def _gaussiank_sa_ring(rank):
    """gaussiankSA's ring-order pairwise reduce-scatter (reference
    VGG/allreducer.py:1531-1578, faithful since round 2) must merge exactly
    the union of every rank's Gaussian-threshold selections: expected =
    sum_r scatter(sel_r) / P, where sel_r re-derives each rank's adaptive
    threshold the way the engine does (fresh engines, residual = 0)."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.allreducer import _gaussian_threshold
    from oktopk_amd.config import OkTopkConfig
    from oktopk_amd import ops
    import torch.distributed as dist

    world = dist.get_world_size()
    density = DENSITY
    cfg = EngineConfig(compressor="gaussiankSA", density=density,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    t = _grad(rank, 0)
    out = eng.run("w", t.clone())

    expected = torch.zeros(N)
    k = max(1, int(N * density))
    for r in range(world):
        g = _grad(r, 0)
        tau = _gaussian_threshold(g, density)
        for _ in range(3):
            cnt = ops.count_gt(g, tau)
            if cnt < 2 * k / 3:
                tau *= 0.5
            elif cnt > 4 * k / 3:
                tau *= 1.5
            else:
                break
        expected[g.abs() > tau] += g[g.abs() > tau]
    expected /= world
    assert torch.allclose(out, expected, atol=1e-5), (out - expected).abs().max()
    # residual keeps exactly the unselected mass
    my_tau = eng.states["w"].tau_local
    res = eng.states["w"].residual
    keep = t.abs() <= my_tau
    assert torch.allclose(res[keep], t[keep], atol=1e-6)
    assert torch.all(res[~keep] == 0)


def test_gaussiank_sa_ring_world4():
    run_dist(_gaussiank_sa_ring, 4)


def test_gaussiank_sa_ring_world3():
    # odd world: ring wrap-around (dst/src = (rank±i) mod P) paths
    run_dist(_gaussiank_sa_ring, 3)


def test_mass_conservation_gaussiankSA_world4():
    run_dist(_mass_conservation, 4, args=("gaussiankSA",))


def test_world3_gtopk():
    # non-power-of-two world: the alive-list merge must leave the unpaired
    # trailing survivor carrying its packet forward (the reference's
    # participate_ranks logic indexes out of range here)
    run_dist(_gtopk_world8, 3)


def test_world6_gtopk():
    run_dist(_gtopk_world8, 6)


def test_world8_oktopk():
    # oktopk IS EF-complete at any world size — full mass invariant
    run_dist(_mass_conservation, 8, args=("oktopk",))


def _chunked_equivalence(rank):
    """pipeline_chunks=C == C independent engines on the C slices
    (docs/overlap_design.md) — bit-equal outputs and residuals."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    C = 3
    n = N - 40  # non-divisible: last chunk takes the tail
    # dense_warmup_iters=0: the warmup dense allreduce is full-tensor in the
    # chunked engine but per-slice in the reference engines — RCCL/gloo ring
    # segmentation differs by message size, giving 1-ulp differences that
    # would break bit-equality (warmup covered by _chunked_warmup below)
    okc = dict(dense_warmup_iters=0, region_repartition_interval=4)
    cfg_c = EngineConfig(compressor="oktopk", density=DENSITY,
                         oktopk=OkTopkConfig(pipeline_chunks=C, **okc))
    eng_c = AllReducer(Comm(dist.group.WORLD), cfg_c)

    cfg_1 = EngineConfig(compressor="oktopk", density=DENSITY,
                         oktopk=OkTopkConfig(**okc))
    engs = [AllReducer(Comm(dist.group.WORLD), cfg_1) for _ in range(C)]

    base = max(8, (n // C) & ~7)
    bounds = [i * base for i in range(C)] + [n]

    for it in range(ITERS):
        t = _grad(rank, it, n)
        ref = t.clone()
        out_c = eng_c.run("w", t.clone())
        parts = []
        for i in range(C):
            lo, hi = bounds[i], bounds[i + 1]
            parts.append(engs[i].run("w", ref[lo:hi].clone()))
        out_ref = torch.cat(parts)
        assert torch.equal(out_c, out_ref), (it, (out_c - out_ref).abs().max())
    for i in range(C):
        lo, hi = bounds[i], bounds[i + 1]
        assert torch.equal(eng_c.states[f"w/c{i}"].residual,
                           engs[i].states["w"].residual)


def _chunked_warmup(rank):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    cfg = EngineConfig(compressor="oktopk", density=DENSITY,
                       oktopk=OkTopkConfig(pipeline_chunks=2,
                                           dense_warmup_iters=2))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    world = dist.get_world_size()
    for it in range(3):
        out = eng.run("w", _grad(rank, it))
        if it < 2:  # dense warmup: exact dense mean
            assert torch.allclose(out, _dense_mean(it, world), atol=1e-5)
        assert torch.isfinite(out).all()


def test_chunked_warmup_world2():
    run_dist(_chunked_warmup, 2)


def test_chunked_equivalence_world1():
    run_dist(_chunked_equivalence, 1)


def test_chunked_equivalence_world2():
    run_dist(_chunked_equivalence, 2)


def test_chunked_equivalence_world4():
    run_dist(_chunked_equivalence, 4)


def _hook_optimizer_chunked(rank):
    """Hook-driven DistributedOptimizer with pipeline_chunks=2: per-bucket
    chunked states don't collide, DP replicas stay in sync."""
    import torch
    from oktopk_amd.comm import Comm
    from oktopk_amd.config import EngineConfig, OkTopkConfig
    from oktopk_amd.optimizer import DistributedOptimizer
    import torch.distributed as dist

    torch.manual_seed(7)  # same init everywhere
    model = torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(), torch.nn.Linear(128, 10))
    inner = torch.optim.SGD(model.parameters(), lr=0.05)
    cfg = EngineConfig(compressor="oktopk", density=0.1,
                       oktopk=OkTopkConfig(dense_warmup_iters=0,
                                           pipeline_chunks=2))
    opt = DistributedOptimizer(inner, model.named_parameters(),
                               comm=Comm(dist.group.WORLD), cfg=cfg)
    g = torch.Generator().manual_seed(100 + rank)  # different data per rank
    for _ in range(3):
        opt.zero_grad()
        x = torch.randn(8, 64, generator=g)
        model(x).sum().backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    assert torch.isfinite(flat).all()
    ref = flat.clone()
    dist.broadcast(ref, src=0)
    assert torch.allclose(flat, ref, atol=1e-6), (flat - ref).abs().max()
    # chunked states exist per bucket
    assert any(k.endswith("/c0") for k in opt.reducer.states)


def test_hook_optimizer_chunked_world2():
    run_dist(_hook_optimizer_chunked, 2)


def _chunked_zero_grads(rank):
    """Zero gradients through the chunked async pipeline: empty selections
    on every rank/chunk must not wedge the collectives (0-size
    all_to_all/all_gather payloads)."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    cfg = EngineConfig(compressor="oktopk", density=0.01,
                       oktopk=OkTopkConfig(dense_warmup_iters=0,
                                           pipeline_chunks=3))
    eng = AllReducer(Comm(dist.group.WORLD), cfg)
    for it in range(3):
        out = eng.run("w", torch.zeros(4096))
        assert torch.equal(out, torch.zeros(4096))
    # then a real gradient still flows
    out = eng.run("w", _grad(rank, 9, 4096))
    assert torch.isfinite(out).all() and out.abs().sum() > 0


def test_chunked_zero_grads_world2():
    run_dist(_chunked_zero_grads, 2)


def _balanced_tiny_s(rank):
    """Balanced redistribution when global survivor count S <= P (some
    balanced blocks empty): still bit-equal to pad-to-max."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import torch.distributed as dist

    outs = {}
    for bal in (False, True):
        cfg = EngineConfig(compressor="oktopk", density=0.001,
                           oktopk=OkTopkConfig(dense_warmup_iters=0,
                                               balanced_allgather=bal))
        eng = AllReducer(Comm(dist.group.WORLD), cfg)
        g0 = torch.zeros(900)
        g0[rank * 3] = rank + 1.0  # one entry per rank
        outs[bal] = [eng.run("w", g0.clone()).clone() for _ in range(4)]
    for a, b in zip(outs[False], outs[True]):
        assert torch.equal(a, b)


def test_balanced_tiny_s_world4():
    run_dist(_balanced_tiny_s, 4)
