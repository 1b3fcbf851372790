"""DeepSpeech-style speech model (reference: LSTM/models/lstm_models.py:148,
built by LSTM/models/lstman4.py:7 with rnn_hidden_size=800, nb_layers=5,
bidirectional): 2x Conv2d frontend + 5x800 bidirectional LSTM (sum of
directions) + per-timestep FC -> CTC logits."""
from __future__ import annotations

import math

import torch
import torch.nn as nn


class BatchRNN(nn.Module):
    """One LSTM layer with BN on input; bidirectional variants sum the two
    directions (reference LSTM/models/lstm_models.py:84-116 BatchRNN; the
    an4 recipe builds it UNIdirectional, LSTM/models/lstman4.py:8)."""

    def __init__(self, input_size: int, hidden_size: int, batch_norm: bool = True,
                 bidirectional: bool = False):
        super().__init__()
        self.batch_norm = (
            nn.BatchNorm1d(input_size) if batch_norm else None
        )
        self.bidirectional = bidirectional
        self.rnn = nn.LSTM(input_size, hidden_size, bidirectional=bidirectional,
                           bias=True)
        self.hidden_size = hidden_size

    def forward(self, x: torch.Tensor) -> torch.Tensor:  # x: (T, N, F)
        if self.batch_norm is not None:
            t, n = x.size(0), x.size(1)
            x = self.batch_norm(x.view(t * n, -1)).view(t, n, -1)
        x, _ = self.rnn(x)
        if self.bidirectional:
            # sum forward/backward directions (reference BatchRNN)
            x = x.view(x.size(0), x.size(1), 2, -1).sum(2)
        return x


class DeepSpeech(nn.Module):
    def __init__(
        self,
        rnn_hidden_size: int = 800,
        nb_layers: int = 5,
        num_classes: int = 29,
        sample_rate: int = 16000,
        window_size: float = 0.02,
        bidirectional: bool = False,
    ):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv2d(1, 32, kernel_size=(41, 11), stride=(2, 2), padding=(20, 5)),
            nn.BatchNorm2d(32),
            nn.Hardtanh(0, 20, inplace=True),
            nn.Conv2d(32, 32, kernel_size=(21, 11), stride=(2, 1), padding=(10, 5)),
            nn.BatchNorm2d(32),
            nn.Hardtanh(0, 20, inplace=True),
        )
        freq = int(math.floor((sample_rate * window_size) / 2) + 1)  # 161
        freq = (freq + 2 * 20 - 41) // 2 + 1
        freq = (freq + 2 * 10 - 21) // 2 + 1
        rnn_in = freq * 32
        rnns = [BatchRNN(rnn_in, rnn_hidden_size, batch_norm=False,
                         bidirectional=bidirectional)]
        for _ in range(nb_layers - 1):
            rnns.append(BatchRNN(rnn_hidden_size, rnn_hidden_size,
                                 bidirectional=bidirectional))
        self.rnns = nn.Sequential(*rnns)
        self.fc = nn.Sequential(
            nn.BatchNorm1d(rnn_hidden_size),
            nn.Linear(rnn_hidden_size, num_classes, bias=False),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """x: (N, 1, freq, time) spectrogram -> (T, N, num_classes) logits."""
        x = self.conv(x)
        n, c, f, t = x.size()
        x = x.view(n, c * f, t).permute(2, 0, 1).contiguous()  # (T, N, F)
        x = self.rnns(x)
        t_, n_ = x.size(0), x.size(1)
        x = self.fc(x.view(t_ * n_, -1)).view(t_, n_, -1)
        return x


def deepspeech_an4(**kw) -> DeepSpeech:
    kw.setdefault("rnn_hidden_size", 800)
    kw.setdefault("nb_layers", 5)
    return DeepSpeech(**kw)
