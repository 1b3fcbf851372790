"""Property-based fuzz of the engine invariants (hypothesis).

For any tensor size, density and compressor: results stay finite and the
error-feedback mass invariant holds at world 1 (result + residual ==
accumulated input, modulo topkA2's documented truncation)."""
import pytest
import torch
from hypothesis import given, settings, strategies as st

from oktopk_amd import AllReducer, Comm, EngineConfig
from oktopk_amd.config import OkTopkConfig

CONSERVING = ["oktopk", "topkA", "topkAopt", "topkSA", "gtopk", "gaussiank",
              "gaussiankconcat", "gaussiankSA", "dense"]


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=4096),
    density=st.floats(min_value=0.001, max_value=1.0),
    comp=st.sampled_from(CONSERVING),
    iters=st.integers(min_value=1, max_value=4),
    seed=st.integers(min_value=0, max_value=10_000),
    scale=st.sampled_from([1.0, 1e-6, 1e4]),
)
def test_engine_invariants_world1(n, density, comp, iters, seed, scale):
    cfg = EngineConfig(compressor=comp, density=density,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(None), cfg)
    g = torch.Generator().manual_seed(seed)
    total_in = torch.zeros(n)
    total_out = torch.zeros(n)
    for _ in range(iters):
        t = torch.randn(n, generator=g) * scale
        total_in += t
        out = eng.run("w", t.clone())
        assert torch.isfinite(out).all()
        total_out += out
    res = eng.states["w"].residual
    assert torch.isfinite(res).all()
    err = (total_out + res - total_in).abs().max().item()
    tol = max(1e-4 * scale, 1e-6)
    assert err <= tol, (comp, n, density, err)


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=2048),
    density=st.floats(min_value=0.001, max_value=0.5),
    seed=st.integers(min_value=0, max_value=1000),
)
def test_compact_matches_reference_fuzz(n, density, seed):
    from oktopk_amd.ops import reference as R

    t = torch.randn(n, generator=torch.Generator().manual_seed(seed))
    tau = R.kth_abs_value(t, max(1, int(n * density)))
    idx, val = R.compact_gt(t, tau)
    assert (t.abs()[idx.long()] > tau).all()
    assert int((t.abs() > tau).sum()) == idx.numel()


@settings(max_examples=20, deadline=None)
@given(
    n=st.integers(min_value=1, max_value=4096),
    density=st.floats(min_value=0.001, max_value=1.0),
    chunks=st.integers(min_value=2, max_value=5),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_chunked_engine_invariants_world1(n, density, chunks, seed):
    """Chunked engine keeps the EF mass invariant for any size/density/C
    (incl. n smaller than the chunk alignment)."""
    cfg = EngineConfig(compressor="oktopk", density=density,
                       oktopk=OkTopkConfig(dense_warmup_iters=0,
                                           pipeline_chunks=chunks))
    eng = AllReducer(Comm(None), cfg)
    g = torch.Generator().manual_seed(seed)
    total_in = torch.zeros(n)
    total_out = torch.zeros(n)
    for _ in range(3):
        t = torch.randn(n, generator=g)
        total_in += t
        out = eng.run("w", t.clone())
        assert torch.isfinite(out).all()
        total_out += out
    parts = [stt.residual for key, stt in
             sorted(((k, s) for k, s in eng.states.items() if "/c" in k),
                    key=lambda kv: int(kv[0].rsplit("c", 1)[1]))]
    res = torch.cat(parts)
    assert res.numel() == n  # chunks tile the tensor exactly
    err = (total_out + res - total_in).abs().max().item()
    assert err <= 1e-4, (n, density, chunks, err)


@settings(max_examples=30, deadline=None)
@given(
    sizes=st.lists(st.integers(min_value=8, max_value=3000), min_size=1,
                   max_size=5),
    density=st.floats(min_value=0.005, max_value=0.5),
    warmup=st.integers(min_value=0, max_value=2),
    iters=st.integers(min_value=1, max_value=5),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_run_many_equals_serial_world1(sizes, density, warmup, iters, seed):
    """The batched drain contract: run_many over ANY bucket list must be
    bit-equal to serial run() calls, including mixed warmup/sparse items
    and ragged sizes (the hook-muted hipGraph/grad-accum drain path)."""
    def mkcfg():
        return EngineConfig(
            compressor="oktopk", density=density,
            oktopk=OkTopkConfig(dense_warmup_iters=warmup,
                                local_threshold_recompute_interval=3,
                                global_threshold_recompute_interval=2,
                                region_repartition_interval=2))

    eng_a, eng_b = AllReducer(Comm(None), mkcfg()), AllReducer(Comm(None), mkcfg())
    g = torch.Generator().manual_seed(seed)
    for it in range(iters):
        gs = [torch.randn(n, generator=g) for n in sizes]
        outs = [eng_a.run(f"b{j}", x.clone()) for j, x in enumerate(gs)]
        items = [(f"b{j}", x.clone(), None) for j, x in enumerate(gs)]
        eng_b.run_many(items)
        for j in range(len(sizes)):
            assert torch.equal(outs[j], items[j][1]), (it, j)
    for j in range(len(sizes)):
        assert torch.equal(eng_a.states[f"b{j}"].residual,
                           eng_b.states[f"b{j}"].residual)
