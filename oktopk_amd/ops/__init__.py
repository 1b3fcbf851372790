"""Op dispatch: HIP/CDNA4 kernels on GPU, pure-torch reference on CPU.

On a CUDA(=ROCm) tensor the hand-written gfx950 extension
(oktopk_amd/ops/csrc, built in-tree as oktopk_amd._hip_ops) is REQUIRED —
a missing extension raises instead of silently falling back to eager torch,
so a GPU run can never pass on the slow path unnoticed.  Set
OKTOPK_FORCE_TORCH_OPS=1 to deliberately run the torch reference on GPU
(used by the numerics-oracle tests that compare both).
"""
from __future__ import annotations

import os
from typing import Tuple

import torch

from . import reference as _ref

_FORCE_TORCH = os.environ.get("OKTOPK_FORCE_TORCH_OPS", "0") == "1"
_hip = None
_hip_err: Exception | None = None


def _load_hip():
    global _hip, _hip_err
    if _hip is None and _hip_err is None:
        try:
            from oktopk_amd import _hip_ops  # in-tree built .so

            _hip = _hip_ops
        except ImportError as e:  # remembered; raised on first GPU use
            _hip_err = e
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _backend(t: torch.Tensor):
    if t.is_cuda and not _FORCE_TORCH:
        mod = _load_hip()
        if mod is None:
            raise RuntimeError(
                "oktopk_amd HIP extension (_hip_ops) is not built but a GPU tensor "
                "reached the ops layer. Build it with `python setup.py build_ext "
                "--inplace` (or __graft_entry__.build()). Original import error: "
                f"{_hip_err!r}"
            )
        return mod
    return _ref


# -- public API (signatures documented in ops/reference.py) ----------------

def kth_abs_value(t: torch.Tensor, k: int) -> float:
    return _backend(t).kth_abs_value(t, int(k))


def compact_gt(t: torch.Tensor, tau: float) -> Tuple[torch.Tensor, torch.Tensor]:
    return _backend(t).compact_gt(t, float(tau))


def count_gt(t: torch.Tensor, tau: float) -> int:
    return int(_backend(t).count_gt(t, float(tau)))


def count_multi_gt(t: torch.Tensor, taus) -> list:
    return [int(x) for x in _backend(t).count_multi_gt(t, [float(x) for x in taus])]


def compact_adaptive(t: torch.Tensor, taus, hi_limit: int):
    out = _backend(t).compact_adaptive(t, [float(x) for x in taus], int(hi_limit))
    idx, val, chosen, count = out
    return idx, val, int(chosen), int(count)


def compact_adaptive_ef(t: torch.Tensor, residual: torch.Tensor,
                        grad, taus, hi_limit: int):
    """Fused EF restore (+bf16 upcast when `grad` given) + adaptive-threshold
    compaction: t = float(grad)+residual (or t+=residual), residual = t, and
    select at the first candidate tau whose count fits hi_limit — one
    streaming pass instead of EF + count."""
    aligned = all(
        x is None or x.data_ptr() % 16 == 0 for x in (t, residual, grad)
    ) and (grad is None or grad.dtype == torch.bfloat16)
    if t.is_cuda and not aligned:  # rare: unaligned views fall back unfused
        if grad is not None:
            ef_restore_upcast_(t, residual, grad)
        else:
            ef_restore_snapshot_(t, residual)
        return compact_adaptive(t, taus, hi_limit)
    out = _backend(t).compact_adaptive_ef(
        t, residual, grad, [float(x) for x in taus], int(hi_limit))
    idx, val, chosen, count = out
    return idx, val, int(chosen), int(count)


def scatter_add_(dest: torch.Tensor, idx: torch.Tensor, val: torch.Tensor) -> torch.Tensor:
    return _backend(dest).scatter_add_(dest, idx, val)


def zero_at_(t: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    return _backend(t).zero_at_(t, idx)


def zero_at_masked_(t: torch.Tensor, idx: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    return _backend(t).zero_at_masked_(t, idx, mask)


def fill_sparse_scaled_(
    out: torch.Tensor, idx: torch.Tensor, val: torch.Tensor, scale: float
) -> torch.Tensor:
    return _backend(out).fill_sparse_scaled_(out, idx, val, float(scale))


def isin_sorted(a: torch.Tensor, b_sorted: torch.Tensor) -> torch.Tensor:
    return _backend(a).isin_sorted(a, b_sorted)


def ef_restore_snapshot_(t: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
    return _backend(t).ef_restore_snapshot_(t, residual)


def ef_restore_upcast_(t: torch.Tensor, residual: torch.Tensor, g: torch.Tensor) -> torch.Tensor:
    return _backend(t).ef_restore_upcast_(t, residual, g)


def fused_sgd_(param, grad, momentum_buf, lr, momentum, weight_decay, nesterov):
    return _backend(param).fused_sgd_(
        param, grad, momentum_buf, float(lr), float(momentum), float(weight_decay), bool(nesterov)
    )


def fused_adam_(param, grad, exp_avg, exp_avg_sq, lr, beta1, beta2, eps,
                weight_decay, gscale=None):
    return _backend(param).fused_adam_(
        param, grad, exp_avg, exp_avg_sq, float(lr), float(beta1), float(beta2),
        float(eps), float(weight_decay), gscale
    )


def fused_adam_mirror_(param, grad, exp_avg, exp_avg_sq, param_bf16, lr, beta1,
                       beta2, eps, weight_decay, gscale=None):
    """Adam step + bf16 weight-mirror write in one pass (GPU); CPU reference
    steps then casts.  `gscale` (optional 1-elem tensor) pre-scales the
    gradient read — the device-resident grad clip."""
    b = _backend(param)
    if hasattr(b, "fused_adam_mirror_"):
        return b.fused_adam_mirror_(
            param, grad, exp_avg, exp_avg_sq, param_bf16, float(lr), float(beta1),
            float(beta2), float(eps), float(weight_decay), gscale
        )
    b.fused_adam_(param, grad, exp_avg, exp_avg_sq, float(lr), float(beta1),
                  float(beta2), float(eps), float(weight_decay), gscale)
    param_bf16.copy_(param)


def scatter_gt_credit_(result, residual, idx, val, tau, scale=1.0) -> int:
    """Fused world-1 round-2 tail: where |val|>tau, result[idx]=val*scale
    and residual[idx]=0; returns the selected count.  `result` must be
    zeroed by the caller."""
    return int(_backend(result).scatter_gt_credit_(
        result, residual, idx, val, float(tau), float(scale)))


def grad_clip_scale(t: torch.Tensor, max_norm: float) -> torch.Tensor:
    """Device-resident clip factor: min(1, max/(||t||+1e-6)) as a 1-elem
    tensor, no host sync (feeds fused_adam_'s gscale)."""
    return _backend(t).grad_clip_scale(t, float(max_norm))


def sumsq_into_(acc: torch.Tensor, t: torch.Tensor) -> None:
    """acc (fp64[1], same device) += sum(t^2); no host sync on GPU."""
    return _backend(t).sumsq_into_(acc, t)


def l2norm(t: torch.Tensor) -> float:
    return float(_backend(t).l2norm(t))
