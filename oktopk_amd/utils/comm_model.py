"""alpha-beta communication cost models (reference VGG/utils.py:62-134).

Used for predicted-vs-measured sanity checks of collective times; constants
default to MI355X xGMI (7 point-to-point links x ~153 GB/s per GPU) instead
of the reference's Cray Aries numbers.
"""
from __future__ import annotations

# Per-link xGMI: ~153 GB/s peak, assume ~80% achievable; latency per hop.
XGMI_LINK_BW = 153e9 * 0.8  # B/s
XGMI_ALPHA = 8e-6  # s per message


def alpha_beta_time(nbytes: float, alpha: float = XGMI_ALPHA, beta_bw: float = XGMI_LINK_BW) -> float:
    """Point-to-point transfer time for one message."""
    return alpha + nbytes / beta_bw


def predict_allreduce_time(nbytes: float, p: int, alpha: float = XGMI_ALPHA,
                           beta_bw: float = XGMI_LINK_BW, channels: int = 7) -> float:
    """Ring allreduce: 2(p-1)/p of the data crosses each link; with
    `channels` concurrent rings the per-link share divides (RCCL uses
    multiple channels over the 7 xGMI links)."""
    if p <= 1:
        return 0.0
    vol = 2.0 * (p - 1) / p * nbytes
    return 2 * (p - 1) * alpha + vol / (beta_bw * min(channels, p))


def predict_sparse_allgather_time(k_elems: int, p: int, bytes_per_elem: int = 8,
                                  alpha: float = XGMI_ALPHA,
                                  beta_bw: float = XGMI_LINK_BW) -> float:
    """Allgather of ~k (idx,val) pairs split across p owners (Ok-Topk
    round 2): each rank receives ~k pairs total."""
    if p <= 1:
        return 0.0
    nbytes = k_elems * bytes_per_elem
    return (p - 1) * alpha + nbytes / beta_bw


def predict_oktopk_volume(k: int, p: int) -> int:
    """Ok-Topk per-rank communication volume bound: < 6k values
    (README.md:2 of the reference; round 1 <= 2*2k spread by ownership,
    round 2 <= 2*2k received)."""
    return 6 * k
