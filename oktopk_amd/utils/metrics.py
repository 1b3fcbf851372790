"""Evaluation metrics: top-k accuracy (VGG/CIFAR), WER/CER via greedy CTC
decoding (LSTM/AN4), perplexity (PTB).

Reference: VGG/dl_trainer.py:709-784 (test loop with top-1 / WER / ppl),
LSTM/decoder.py GreedyDecoder (WER through python-Levenshtein; here edit
distance is implemented directly — no external dependency).
"""
from __future__ import annotations

import math
from typing import List, Sequence

import torch


def accuracy_topk(output: torch.Tensor, target: torch.Tensor, topk=(1,)) -> List[float]:
    maxk = max(topk)
    _, pred = output.topk(maxk, 1, True, True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    for k in topk:
        c = correct[:k].reshape(-1).float().sum(0).item()
        res.append(c * 100.0 / target.size(0))
    return res


def edit_distance(a: Sequence, b: Sequence) -> int:
    """Levenshtein distance (replaces the python-Levenshtein dependency)."""
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[j - 1] + 1, prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


def wer(hyp: str, ref: str) -> float:
    h, r = hyp.split(), ref.split()
    if not r:
        return 0.0 if not h else 1.0
    return edit_distance(h, r) / len(r)


def cer(hyp: str, ref: str) -> float:
    if not ref:
        return 0.0 if not hyp else 1.0
    return edit_distance(list(hyp), list(ref)) / len(ref)


def perplexity(mean_nll: float) -> float:
    return float(math.exp(min(mean_nll, 50.0)))


class GreedyDecoder:
    """Greedy (best-path) CTC decoder (reference LSTM/decoder.py:GreedyDecoder):
    argmax per frame, collapse repeats, strip blanks."""

    def __init__(self, labels: str, blank_index: int = 0):
        self.labels = labels
        self.blank = blank_index

    def decode(self, logits: torch.Tensor) -> List[str]:
        """logits: (T, N, C) -> list of N decoded strings."""
        best = logits.argmax(dim=-1)  # (T, N)
        out = []
        for n in range(best.size(1)):
            seq = best[:, n].tolist()
            chars = []
            prev = None
            for s in seq:
                if s != self.blank and s != prev:
                    if s < len(self.labels):
                        chars.append(self.labels[s])
                prev = s
            out.append("".join(chars))
        return out
