"""Fused residual-add + LayerNorm with autograd (bf16 CDNA4 kernels).

Replaces `LN(x + r)` — two sites per BertLayer — collapsing add + moments +
apply (forward) and the three LN backward kernels into one kernel each way.
The gradient w.r.t. x and r is the same tensor (the add passes gradients
through unchanged).
"""
from __future__ import annotations

import os

import torch


def fused_ln_available(x: torch.Tensor) -> bool:
    # opt-in: under hipGraph capture the unfused add+LN launches are already
    # cheap and torch's LN kernels are well-tuned — the fused path measured
    # ~0.3 ms/step SLOWER on BERT-base (A/B in profiles/README.md); enable
    # for experimentation with OKTOPK_FUSED_LN=1
    if os.environ.get("OKTOPK_FUSED_LN", "0") != "1":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    h = x.shape[-1]
    if h % 128 != 0 or h > 2048:
        return False
    from . import hip_available

    return hip_available()


class _FusedAddLayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, r, weight, bias, eps):
        from oktopk_amd import _hip_ops

        y, s, mean, rstd = _hip_ops.add_ln_fwd(
            x.contiguous(), r.contiguous(), weight, bias, eps
        )
        ctx.save_for_backward(s, mean, rstd, weight)
        return y

    @staticmethod
    def backward(ctx, gy):
        from oktopk_amd import _hip_ops

        s, mean, rstd, weight = ctx.saved_tensors
        gx, dgamma, dbeta = _hip_ops.add_ln_bwd(gy, s, mean, rstd, weight)
        return gx, gx, dgamma.to(weight.dtype), dbeta.to(weight.dtype), None


def fused_add_layernorm(x, r, ln: torch.nn.LayerNorm) -> torch.Tensor:
    """y = LN(x + r) with ln's weight/bias/eps."""
    return _FusedAddLayerNorm.apply(x, r, ln.weight, ln.bias, ln.eps)
