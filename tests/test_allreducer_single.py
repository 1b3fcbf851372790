"""World-of-1 engine semantics: every compressor must run and preserve the
error-feedback invariant (result + residual carries the full gradient mass)."""
import pytest
import torch

from oktopk_amd import AllReducer, Comm, EngineConfig
from oktopk_amd.config import OkTopkConfig


def make_engine(compressor, density=0.05, warmup=0, **okkw):
    cfg = EngineConfig(
        compressor=compressor,
        density=density,
        oktopk=OkTopkConfig(dense_warmup_iters=warmup, **okkw),
    )
    return AllReducer(Comm(None), cfg)


ALL = ["dense", "oktopk", "topkA", "topkA2", "topkAopt", "gaussiank", "topkSA", "gtopk", "gaussiankSA"]


@pytest.mark.parametrize("comp", ALL)
def test_runs_world1(comp):
    eng = make_engine(comp)
    g = torch.Generator().manual_seed(42)
    for it in range(5):
        t = torch.randn(4096, generator=g)
        out = eng.run("w", t)
        assert out.shape == (4096,)
        assert torch.isfinite(out).all()


def test_dense_world1_identity():
    eng = make_engine("dense")
    t = torch.randn(100, generator=torch.Generator().manual_seed(0))
    ref = t.clone()
    out = eng.run("w", t)
    assert torch.allclose(out, ref)


@pytest.mark.parametrize("comp", ["oktopk", "topkA", "gaussiank", "topkSA", "gtopk"])
def test_error_feedback_conservation(comp):
    """After each iteration, result + residual == accumulated gradient mass
    (nothing is lost, only deferred)."""
    eng = make_engine(comp, density=0.02)
    g = torch.Generator().manual_seed(7)
    total_in = torch.zeros(2000)
    total_out = torch.zeros(2000)
    for it in range(6):
        t = torch.randn(2000, generator=g)
        total_in += t
        out = eng.run("w", t.clone())
        total_out += out
    residual = eng.states["w"].residual
    assert torch.allclose(total_out + residual, total_in, atol=1e-4), (
        f"{comp}: EF leak max={ (total_out + residual - total_in).abs().max() }"
    )


def test_oktopk_world1_selects_topk():
    """World 1: ok-topk result must contain exactly the top-k entries of the
    (EF-restored) gradient on exact-recompute iterations."""
    eng = make_engine("oktopk", density=0.01)
    t = torch.randn(10_000, generator=torch.Generator().manual_seed(3))
    ref = t.clone()
    out = eng.run("w", t)
    k = 100
    top = torch.topk(ref.abs(), k).indices
    nz = out.nonzero().view(-1)
    # strict-> selection excludes the k-th element itself (reference
    # compressbythreshold uses abs > tau, VGG/compression.py:122-132)
    assert k - 2 <= nz.numel() <= k
    assert set(nz.tolist()) <= set(top.tolist())
    assert torch.allclose(out[nz], ref[nz])


def test_warmup_uses_dense():
    eng = make_engine("oktopk", warmup=3)
    t = torch.randn(500, generator=torch.Generator().manual_seed(1))
    ref = t.clone()
    out = eng.run("w", t)
    assert torch.allclose(out, ref)  # dense passthrough in warmup
    assert eng.states["w"].counter == 1


def test_unknown_compressor_rejected():
    with pytest.raises(ValueError):
        AllReducer(Comm(None), EngineConfig(compressor="nope"))


def test_timing_table_populated():
    eng = make_engine("oktopk")
    t = torch.randn(1000, generator=torch.Generator().manual_seed(5))
    eng.run("w", t)
    tbl = eng.timing_table("w")
    assert "compress" in tbl and "allgather" in tbl


def test_dynamic_density_schedule():
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig

    cfg = EngineConfig(compressor="oktopk", density=0.5,
                       dynamic_densities=(0.25, 0.1, 0.05),
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(None), cfg)
    assert eng.get_current_density() == 0.25
    eng.train_epoch = 1
    assert eng.get_current_density() == 0.1
    eng.train_epoch = 99
    assert eng.get_current_density() == 0.05


def test_trainer_evaluate_cifar():
    from oktopk_amd.config import EngineConfig, OkTopkConfig
    from oktopk_amd.trainer import Trainer

    cfg = EngineConfig(compressor="dense")
    tr = Trainer("caffe_cifar", batch_size=4, cfg=cfg, dtype="fp32")
    m = tr.evaluate()
    assert "top1" in m and 0.0 <= m["top1"] <= 100.0


def test_sgd_path_grad_clip():
    import torch
    from oktopk_amd.comm import Comm
    from oktopk_amd.config import EngineConfig
    from oktopk_amd.optimizer import DistributedOptimizer

    torch.manual_seed(0)
    model = torch.nn.Linear(10, 10)
    inner = torch.optim.SGD(model.parameters(), lr=1.0)
    opt = DistributedOptimizer(inner, model.named_parameters(),
                               cfg=EngineConfig(compressor="dense"),
                               norm_clip=0.1)
    p0 = model.weight.detach().clone()
    opt.zero_grad()
    (model(torch.randn(4, 10) * 100).sum()).backward()
    opt.step()
    delta = (model.weight.detach() - p0).norm()
    # lr=1, clipped grad norm <= 0.1 -> total update norm <= ~0.1
    assert delta <= 0.11, delta


def test_profiling_grad_dump(tmp_path):
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    import json, os

    cfg = EngineConfig(compressor="oktopk", density=0.05,
                       profiling_grad_iters=(1,),
                       profiling_grad_dir=str(tmp_path),
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(None), cfg)
    for it in range(2):
        eng.run("w", torch.randn(2048, generator=torch.Generator().manual_seed(it)))
    files = os.listdir(tmp_path)
    assert files, "no dump written"
    d = json.load(open(os.path.join(tmp_path, files[0])))
    assert d["iter"] == 1 and len(d["abs_quantiles"]) == 21


def test_randomize_batches_in_place():
    from oktopk_amd.config import EngineConfig
    from oktopk_amd.trainer import Trainer

    tr = Trainer("caffe_cifar", batch_size=2, cfg=EngineConfig(compressor="dense"),
                 dtype="fp32")
    ptr = tr.batches.x.data_ptr()
    before = tr.batches.x.clone()
    tr.batches.randomize_()
    assert tr.batches.x.data_ptr() == ptr  # in-place (graph-safe)
    assert not torch.allclose(tr.batches.x, before)


def test_eps_oracle_with_grad_src():
    """The EPS oracle must read grad_src (not the stale fp32 buffer) when
    the fused-EF path is active; at iteration 0 world-1 the oktopk result IS
    the dense top-k, so EPS ~ 0."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig

    cfg = EngineConfig(compressor="oktopk", density=0.05, profiling_norm=True,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(None), cfg)
    g = torch.randn(4096, generator=torch.Generator().manual_seed(0))
    t = torch.zeros(4096)  # stale buffer (simulates flat_grad before upcast)
    eng.run("w", t, grad_src=g.to(torch.bfloat16))
    eps = eng.eps_log[0][1]
    assert eps < 0.35, eps  # bf16 wire rounding + strict-> tie loss only


def test_lr_schedule_step_decay():
    from oktopk_amd.config import EngineConfig
    from oktopk_amd.trainer import Trainer

    tr = Trainer("caffe_cifar", batch_size=2, cfg=EngineConfig(compressor="dense"),
                 dtype="fp32", lr=0.1)
    tr.set_epoch(0)
    assert tr.opt.param_groups[0]["lr"] == pytest.approx(0.1)
    tr.set_epoch(81)
    assert tr.opt.param_groups[0]["lr"] == pytest.approx(0.01)
    tr.set_epoch(122)
    assert tr.opt.param_groups[0]["lr"] == pytest.approx(0.001)
    tr.set_epoch(5)  # schedules are absolute, not cumulative
    assert tr.opt.param_groups[0]["lr"] == pytest.approx(0.1)


def test_chunked_state_roundtrip():
    """Chunked engine states (w/c{i} + w/meta) survive a state_dict round
    trip through the optimizer-style rebuild path."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig
    from oktopk_amd.allreducer import TensorState

    cfg = EngineConfig(compressor="oktopk", density=0.02,
                       oktopk=OkTopkConfig(dense_warmup_iters=0,
                                           pipeline_chunks=2))
    eng = AllReducer(Comm(None), cfg)
    g = torch.Generator().manual_seed(5)
    for it in range(3):
        eng.run("w", torch.randn(4096, generator=g))
    saved = {n: s.state_dict() for n, s in eng.states.items()}
    assert set(saved) == {"w/c0", "w/c1", "w/meta"}
    assert saved["w/meta"]["counter"] == 3

    eng2 = AllReducer(Comm(None), cfg)
    for n, sd in saved.items():
        eng2.states[n] = TensorState(residual=sd["residual"].clone())
        eng2.states[n].load_state_dict(sd)
    t = torch.randn(4096, generator=torch.Generator().manual_seed(99))
    o1 = eng.run("w", t.clone())
    o2 = eng2.run("w", t.clone())
    assert torch.equal(o1, o2)


def test_randk_research_log():
    """profiling_norm also logs the rand-k floor (reference randk norm
    arrays): informed selection (EPS) must beat random selection."""
    from oktopk_amd import AllReducer, Comm, EngineConfig
    from oktopk_amd.config import OkTopkConfig

    torch.manual_seed(0)
    cfg = EngineConfig(compressor="oktopk", density=0.02, profiling_norm=True,
                       oktopk=OkTopkConfig(dense_warmup_iters=0))
    eng = AllReducer(Comm(None), cfg)
    g = torch.Generator().manual_seed(3)
    for it in range(4):
        eng.run("w", torch.randn(8192, generator=g))
    assert len(eng.randk_log) == 4 and len(eng.eps_log) == 4
    assert len(eng.upbound_log) == 4
    for (_, eps), (_, rk), (_, ub) in zip(eng.eps_log, eng.randk_log,
                                          eng.upbound_log):
        assert eps < rk      # informed beats random
        assert 0.0 <= ub < 1.0  # truncation floor is a proper fraction


def test_momentum_correction_matches_sgd_momentum():
    """momentum_correction (pre-reduce momentum, reference
    distributed_optimizer.py:56,81-88) with an inner momentum-0 SGD equals
    plain SGD momentum (dampening 0) on the dense path."""
    from oktopk_amd.comm import Comm
    from oktopk_amd.config import EngineConfig
    from oktopk_amd.optimizer import DistributedOptimizer

    torch.manual_seed(1)
    m1 = torch.nn.Linear(16, 8)
    m2 = torch.nn.Linear(16, 8)
    m2.load_state_dict(m1.state_dict())

    opt1 = DistributedOptimizer(
        torch.optim.SGD(m1.parameters(), lr=0.1, momentum=0.0),
        m1.named_parameters(), comm=Comm(None),
        cfg=EngineConfig(compressor="dense"), momentum_correction=0.9)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9, dampening=0.0)

    g = torch.Generator().manual_seed(2)
    for _ in range(4):
        x = torch.randn(4, 16, generator=g)
        opt1.zero_grad()
        m1(x).sum().backward()
        opt1.step()
        opt2.zero_grad()
        m2(x).sum().backward()
        opt2.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_lr_schedules_per_recipe():
    """Per-recipe LR schedules (reference _adjust_learning_rate_*,
    VGG/dl_trainer.py:507-570)."""
    from oktopk_amd.comm import Comm
    from oktopk_amd.trainer import Trainer

    tr = Trainer(model_name="lstman4", batch_size=2, comm=Comm(None),
                 dtype="fp32", lr=1.0)
    tr.adjust_learning_rate(0)
    assert abs(tr.opt.param_groups[0]["lr"] - 1.0) < 1e-9
    tr.adjust_learning_rate(2)
    assert abs(tr.opt.param_groups[0]["lr"] - 1.0 / 1.01 ** 2) < 1e-9

    tr2 = Trainer(model_name="lstm", batch_size=2, comm=Comm(None),
                  dtype="fp32", lr=1.0)
    tr2.adjust_learning_rate(10)
    assert abs(tr2.opt.param_groups[0]["lr"] - 1.0) < 1e-9
    tr2.adjust_learning_rate(70)
    assert abs(tr2.opt.param_groups[0]["lr"] - 0.01) < 1e-9
    tr2.adjust_learning_rate(85)
    assert abs(tr2.opt.param_groups[0]["lr"] - 0.001) < 1e-9
