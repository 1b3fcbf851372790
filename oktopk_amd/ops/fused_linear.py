"""Fused Linear+GELU (hand-written MFMA kernel) with autograd.

The forward runs the CDNA4 kernel (ops/csrc/linear_gelu.hip): one launch for
GEMM + bias + erf-GELU, also materialising the pre-activation for backward.
Backward uses the saved pre-activation for the exact erf-GELU derivative and
rocBLAS matmuls for the two gradient GEMMs.

Parity: the reference's LinearActivation fused-gelu module
(/root/reference/BERT/bert/transformers/modeling.py:75).
"""
from __future__ import annotations

import os

import torch

_SQRT1_2 = 0.7071067811865476
_INV_SQRT_2PI = 0.3989422804014327


def fused_available(x: torch.Tensor, weight: torch.Tensor) -> bool:
    # opt-in: at BERT-base shapes (M=1024, 192 blocks) the correctness-tier
    # kernel underfills 256 CUs and loses ~1ms/step to hipBLASLt+gelu
    # (measured); enable for A/B with OKTOPK_FUSED_MLP=1
    if os.environ.get("OKTOPK_FUSED_MLP", "0") != "1":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16 and weight.dtype == torch.bfloat16):
        return False
    if weight.shape[1] % 32 != 0:
        return False
    from . import hip_available

    return hip_available()


class _FusedLinearGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        from oktopk_amd import _hip_ops

        shp = x.shape
        x2 = x.reshape(-1, shp[-1]).contiguous()
        b = bias if bias is not None else torch.empty(0, device=x.device)
        y, z = _hip_ops.linear_gelu(x2, weight.contiguous(), b, True, True)
        ctx.save_for_backward(x2, weight, z)
        ctx.in_shape = shp
        ctx.has_bias = bias is not None
        return y.reshape(*shp[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, gy):
        x2, w, z = ctx.saved_tensors
        gy2 = gy.reshape(-1, gy.shape[-1])
        zf = z.float()
        # d/dz gelu_erf(z) = Phi(z) + z * phi(z)
        gp = 0.5 * (1.0 + torch.erf(zf * _SQRT1_2)) + zf * torch.exp(-0.5 * zf * zf) * _INV_SQRT_2PI
        gpre = (gy2.float() * gp).to(gy2.dtype)
        gx = (gpre @ w).reshape(ctx.in_shape)
        gw = gpre.t() @ x2
        gb = gpre.float().sum(0).to(w.dtype) if ctx.has_bias else None
        return gx, gw, gb


def fused_linear_gelu(x: torch.Tensor, weight: torch.Tensor, bias=None) -> torch.Tensor:
    """y = gelu(x @ weight^T + bias), bf16, single MFMA kernel forward."""
    return _FusedLinearGelu.apply(x, weight, bias)


# ---------------------------------------------------------------------------
# ColsumLinear: nn.Linear whose BACKWARD computes the bias gradient with the
# colsum_bf16 kernel instead of torch's reduce_kernel chain (the r02-i
# profile showed ~60 such reductions/step at ~10 us each on BERT-base).
# The input/weight gradients are the same two GEMMs torch runs.
# ---------------------------------------------------------------------------

class _LinearColsumBias(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return torch.nn.functional.linear(x, w, b)

    @staticmethod
    def backward(ctx, gy):
        from oktopk_amd import _hip_ops

        x, w = ctx.saved_tensors
        gy_c = gy.contiguous()
        gx = gy_c @ w
        gy2 = gy_c.reshape(-1, gy_c.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        gw = gy2.t() @ x2
        gb = _hip_ops.colsum_bf16(gy_c)
        return gx, gw, gb


class ColsumLinear(torch.nn.Linear):
    """Drop-in nn.Linear (same state_dict keys); kernel bias-grad path
    behind OKTOPK_COLSUM_BIAS=1.  Measured NEUTRAL-to-negative end-to-end
    on BERT-base under hipGraph (9.27-9.31 vs 9.24-9.25 ms/step, same-box
    4-run A/B) — torch's bias reduce chain was not the bottleneck the
    aggregate profile suggested, so the default stays torch."""

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16
                and self.bias is not None
                and self.weight.dtype == torch.bfloat16
                and os.environ.get("OKTOPK_COLSUM_BIAS", "0") == "1"):
            from . import hip_available

            if hip_available():
                return _LinearColsumBias.apply(x, self.weight, self.bias)
        return torch.nn.functional.linear(x, self.weight, self.bias)
