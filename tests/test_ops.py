"""Unit tests for the op layer (torch reference path, CPU).

The same assertions re-run against the HIP kernels in test_ops_gpu.py.
"""
import pytest
import torch

from oktopk_amd.ops import reference as R


def test_kth_abs_value_matches_topk():
    g = torch.Generator().manual_seed(0)
    t = torch.randn(10_000, generator=g)
    k = 100
    tau = R.kth_abs_value(t, k)
    assert (t.abs() >= tau).sum().item() >= k
    assert (t.abs() > tau).sum().item() < k + 10  # ties only


def test_compact_gt_sorted_and_correct():
    g = torch.Generator().manual_seed(1)
    t = torch.randn(5000, generator=g)
    tau = 1.5
    idx, val = R.compact_gt(t, tau)
    assert idx.dtype == torch.int32
    assert (idx[1:] > idx[:-1]).all()  # strictly ascending
    mask = t.abs() > tau
    assert idx.numel() == int(mask.sum())
    assert torch.equal(val, t[mask])


def test_compact_count_agree():
    t = torch.randn(1000, generator=torch.Generator().manual_seed(2))
    for tau in (0.0, 0.5, 2.0, 100.0):
        idx, _ = R.compact_gt(t, tau)
        assert idx.numel() == R.count_gt(t, tau)


def test_scatter_add_duplicates():
    dest = torch.zeros(10)
    idx = torch.tensor([1, 1, 3, 9], dtype=torch.int32)
    val = torch.tensor([1.0, 2.0, 3.0, 4.0])
    R.scatter_add_(dest, idx, val)
    assert dest[1] == 3.0 and dest[3] == 3.0 and dest[9] == 4.0
    assert dest.sum() == 10.0


def test_fill_sparse_scaled():
    out = torch.randn(8)
    idx = torch.tensor([0, 7], dtype=torch.int32)
    val = torch.tensor([2.0, -4.0])
    R.fill_sparse_scaled_(out, idx, val, 0.5)
    assert out[0] == 1.0 and out[7] == -2.0
    assert out[1:7].abs().sum() == 0


def test_isin_sorted():
    a = torch.tensor([0, 5, 9, 12], dtype=torch.int32)
    b = torch.tensor([5, 12, 40], dtype=torch.int32)
    m = R.isin_sorted(a, b)
    assert m.tolist() == [False, True, False, True]
    assert R.isin_sorted(a, torch.empty(0, dtype=torch.int32)).sum() == 0


def test_ef_restore_snapshot():
    t = torch.tensor([1.0, 2.0])
    r = torch.tensor([0.5, -1.0])
    R.ef_restore_snapshot_(t, r)
    assert torch.equal(t, torch.tensor([1.5, 1.0]))
    assert torch.equal(r, t)


def test_fused_sgd_matches_torch():
    torch.manual_seed(3)
    p = torch.randn(100)
    gr = torch.randn(100)
    buf = torch.zeros(100)

    p_ref = p.clone().requires_grad_(True)
    opt = torch.optim.SGD([p_ref], lr=0.1, momentum=0.9, weight_decay=1e-4, nesterov=True)
    p_ref.grad = gr.clone()
    opt.step()
    opt.zero_grad()
    p_ref.grad = gr.clone()
    opt.step()

    for _ in range(2):
        R.fused_sgd_(p, gr, buf, 0.1, 0.9, 1e-4, True)
    assert torch.allclose(p, p_ref.detach(), atol=1e-6)


def test_fused_adam_sane():
    p = torch.zeros(10)
    gr = torch.ones(10)
    m = torch.zeros(10)
    v = torch.zeros(10)
    R.fused_adam_(p, gr, m, v, lr=0.01, beta1=0.9, beta2=0.999, eps=1e-8, weight_decay=0.0)
    assert (p < 0).all()  # moved against the gradient
    assert torch.allclose(m, torch.full((10,), 0.1))


def test_compact_adaptive_ef_matches_unfused():
    g = torch.Generator().manual_seed(11)
    for grad_dtype in (None, torch.bfloat16):
        t = torch.randn(5000, generator=g)
        r = torch.randn(5000, generator=g) * 0.1
        grad = None
        if grad_dtype is not None:
            grad = torch.randn(5000, generator=g).to(grad_dtype)
        t2, r2 = t.clone(), r.clone()
        taus = [0.5, 0.6, 0.8]
        idx, val, chosen, cnt = R.compact_adaptive_ef(t, r, grad, taus, 400)
        if grad is not None:
            t2.copy_(grad.float() + r2)
        else:
            t2.add_(r2)
        r2.copy_(t2)
        idx2, val2, chosen2, cnt2 = R.compact_adaptive(t2, taus, 400)
        assert torch.equal(idx, idx2) and torch.equal(val, val2)
        assert chosen == chosen2 and cnt == cnt2
        # round-2 contract: the fused op does NOT write t (the engine's
        # steady state never reads the restored values from t)
        assert torch.equal(r, r2)
