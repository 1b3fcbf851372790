"""Fused self-attention with autograd — the production path since round 2.

_FlashAttention (default, any seq % 128 == 0 / hd 64): online-softmax
forward (ops/csrc/attention_fa.hip) saving only ctx + per-row LSE, and a
hand-written two-kernel backward (query-parallel dQ; key-parallel dK/dV)
that recomputes P from the LSE and REGENERATES the philox dropout mask —
nothing O(S^2) is ever materialised.  Beats torch SDPA (AOTriton) at
every measured shape (profiles/attn_ab_r02*).  The torch-recompute
backward is kept behind OKTOPK_ATTN_BWD_TORCH=1 as the slow oracle.

_FusedAttention (legacy, OKTOPK_ATTN_LEGACY=1, seq=128 only): the
round-1 single-pass kernel saving P and A, with a torch-bmm backward,
    dV = A^T gO,   dA = gO V^T,   dP = dA * mask/keep   (mask = [A != 0]),
    dS = P * (dP - rowsum(dP * P)),  dQ = dS K * scale,  dK = dS^T Q * scale.
"""
from __future__ import annotations

import math
import os

import torch


def fused_attn_available(x: torch.Tensor, num_heads: int, seq: int, dropout_p: float) -> bool:
    # DEFAULT ON at every seq % 128 == 0 (hd 64) since round 2: the flash
    # kernel pair (attention_fa.hip fwd + hand-written dQ/dKdV backward)
    # beats torch SDPA (AOTriton) isolated (70.6 vs 121.0 us fwd+bwd at
    # s128; 199 vs 213 at s512 with dropout after the cooperative LDS
    # dropout mask) AND end-to-end under hipGraph on every measured
    # shape: bert_base 10.30 -> 9.99 (s128), 12.77 -> 11.87 (s256),
    # 17.0 -> 16.1 (s512); bert_large s512 35.7 -> 34.3 (2-rep A/Bs,
    # profiles/README.md r02).  OKTOPK_FUSED_ATTN=0 forces SDPA.
    if os.environ.get("OKTOPK_FUSED_ATTN", "") == "0":
        return False
    if not (x.is_cuda and x.dtype == torch.bfloat16):
        return False
    hd = x.shape[-1] // (3 * num_heads)
    if seq % 128 != 0 or hd != 64:
        return False
    from . import hip_available

    return hip_available()


class _FusedAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, qkv, mask, num_heads, dropout_p, training):
        from oktopk_amd import _hip_ops

        need_grad = qkv.requires_grad
        outs = _hip_ops.attn_fwd(
            qkv,
            mask if mask is not None else torch.empty(0, device=qkv.device),
            num_heads,
            float(dropout_p),
            bool(training),
            need_grad,
        )
        if need_grad:
            out, p, a = outs
            ctx.save_for_backward(qkv, p, a)
            ctx.meta = (num_heads, dropout_p, training)
        else:
            out = outs[0]
        return out

    @staticmethod
    def backward(ctx, go):
        qkv, p, a = ctx.saved_tensors
        num_heads, dropout_p, training = ctx.meta
        b, s, h3 = qkv.shape
        h = h3 // 3
        hd = h // num_heads
        scale = 1.0 / math.sqrt(hd)
        # views of the projection buffer: (b*nh, s, hd)
        qkv5 = qkv.view(b, s, 3, num_heads, hd)
        q = qkv5[:, :, 0].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        k = qkv5[:, :, 1].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        v = qkv5[:, :, 2].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        go_h = go.view(b, s, num_heads, hd).permute(0, 2, 1, 3).reshape(
            b * num_heads, s, hd
        )
        dv = torch.bmm(a.transpose(1, 2), go_h)
        da = torch.bmm(go_h, v.transpose(1, 2))
        if training and dropout_p > 0:
            keep = 1.0 - dropout_p
            dp = da * (a != 0).to(da.dtype) / keep
        else:
            dp = da
        pf = p.float()
        dpf = dp.float()
        ds = (pf * (dpf - (dpf * pf).sum(dim=-1, keepdim=True))).to(qkv.dtype)
        dq = torch.bmm(ds, k) * scale
        dk = torch.bmm(ds.transpose(1, 2), q) * scale
        dqkv = torch.empty_like(qkv).view(b, s, 3, num_heads, hd)
        dqkv[:, :, 0] = dq.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        dqkv[:, :, 1] = dk.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        dqkv[:, :, 2] = dv.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        return dqkv.view(b, s, h3), None, None, None, None


class _FlashAttention(torch.autograd.Function):
    """Online-softmax forward (any seq % 128 == 0): the kernel materialises
    nothing O(S^2) — only ctx and the per-row LSE come back.  Backward
    recomputes P = exp(QK^T*scale + mask - lse) (already normalised, no
    softmax pass) and REGENERATES the dropout mask from the same philox
    counters the forward consumed (dropout_mask_mul_), then runs the bmm
    chain of _FusedAttention.backward."""

    @staticmethod
    def forward(ctx, qkv, mask, num_heads, dropout_p, training):
        from oktopk_amd import _hip_ops

        need_grad = qkv.requires_grad
        outs = _hip_ops.attn_fwd_fa(
            qkv,
            mask if mask is not None else torch.empty(0, device=qkv.device),
            num_heads,
            float(dropout_p),
            bool(training),
            need_grad,
        )
        if need_grad:
            out, lse, philox = outs
            ctx.save_for_backward(qkv, lse, philox, out)
            ctx.mask = mask
            ctx.meta = (num_heads, dropout_p, training)
        else:
            out = outs[0]
        return out

    @staticmethod
    def backward(ctx, go):
        from oktopk_amd import _hip_ops

        qkv, lse, philox, out = ctx.saved_tensors
        mask = ctx.mask
        num_heads, dropout_p, training = ctx.meta
        if os.environ.get("OKTOPK_ATTN_BWD_TORCH", "0") != "1":
            # hand-written flash backward (attn_bwd_dq/_dkv kernels):
            # D_i = gO_i . O_i, then dQ/dK/dV with P recomputed from lse
            b, s, h3 = qkv.shape
            go_c = go.contiguous()
            d = _hip_ops.attn_rowdot(go_c, out, num_heads)
            dqkv = _hip_ops.attn_bwd_fa(
                qkv, go_c, lse, d,
                mask if mask is not None else torch.empty(0, device=qkv.device),
                philox, num_heads, float(dropout_p), bool(training))
            return dqkv, None, None, None, None
        b, s, h3 = qkv.shape
        h = h3 // 3
        hd = h // num_heads
        scale = 1.0 / math.sqrt(hd)
        qkv5 = qkv.view(b, s, 3, num_heads, hd)
        q = qkv5[:, :, 0].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        k = qkv5[:, :, 1].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        v = qkv5[:, :, 2].permute(0, 2, 1, 3).reshape(b * num_heads, s, hd)
        go_h = go.view(b, s, num_heads, hd).permute(0, 2, 1, 3).reshape(
            b * num_heads, s, hd
        )
        # recompute normalised P from the saved LSE (one bmm + exp; the
        # softmax reduction never re-runs)
        sc = torch.baddbmm(
            torch.zeros(1, dtype=torch.float32, device=qkv.device),
            q.float(), k.transpose(1, 2).float(), alpha=scale,
        )
        if mask is not None:
            sc = sc + mask.reshape(b, 1, 1, s).float().expand(
                b, num_heads, 1, s).reshape(b * num_heads, 1, s)
        p = torch.exp(sc - lse.view(b * num_heads, s, 1)).to(qkv.dtype)
        del sc
        if training and dropout_p > 0:
            a = p.clone()
            _hip_ops.dropout_mask_mul_(a, philox, float(dropout_p))
        else:
            a = p
        dv = torch.bmm(a.transpose(1, 2), go_h)
        da = torch.bmm(go_h, v.transpose(1, 2))
        if training and dropout_p > 0:
            keep = 1.0 - dropout_p
            dp = da * (a != 0).to(da.dtype) / keep
        else:
            dp = da
        pf = p.float()
        dpf = dp.float()
        ds = (pf * (dpf - (dpf * pf).sum(dim=-1, keepdim=True))).to(qkv.dtype)
        dq = torch.bmm(ds, k) * scale
        dk = torch.bmm(ds.transpose(1, 2), q) * scale
        dqkv = torch.empty_like(qkv).view(b, s, 3, num_heads, hd)
        dqkv[:, :, 0] = dq.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        dqkv[:, :, 1] = dk.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        dqkv[:, :, 2] = dv.view(b, num_heads, s, hd).permute(0, 2, 1, 3)
        return dqkv.view(b, s, h3), None, None, None, None


def fused_attention(qkv, mask, num_heads, dropout_p, training):
    """ctx = dropout(softmax(QK^T*scale + mask)) @ V from the fused qkv
    projection buffer [b, s, 3h]; returns [b, s, h].  Dispatch: the flash
    (online-softmax) kernel for any seq % 128 == 0 unless
    OKTOPK_ATTN_LEGACY=1 forces the seq-128 single-pass kernel."""
    s = qkv.shape[1]
    if os.environ.get("OKTOPK_ATTN_LEGACY", "0") == "1" and s == 128:
        return _FusedAttention.apply(qkv, mask, num_heads, dropout_p, training)
    return _FlashAttention.apply(qkv, mask, num_heads, dropout_p, training)
