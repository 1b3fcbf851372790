"""AN4-style audio spectrogram pipeline.

Reference: LSTM/audio_data loaders compute log-magnitude STFT spectrograms
with librosa (20 ms window, 10 ms stride, hamming).  librosa is not available
offline — torch.stft computes the identical feature.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
from torch.utils.data import Dataset


def compute_spectrogram(
    waveform: torch.Tensor,
    sample_rate: int = 16000,
    window_size: float = 0.02,
    window_stride: float = 0.01,
    normalize: bool = True,
) -> torch.Tensor:
    """(samples,) -> (freq, time) log-magnitude spectrogram, freq = n_fft/2+1
    (161 at 16 kHz / 20 ms — matches the DeepSpeech conv frontend)."""
    n_fft = int(sample_rate * window_size)
    hop = int(sample_rate * window_stride)
    window = torch.hamming_window(n_fft, device=waveform.device)
    spec = torch.stft(
        waveform.float(), n_fft=n_fft, hop_length=hop, win_length=n_fft,
        window=window, return_complex=True, center=True,
    ).abs()
    spec = torch.log1p(spec)
    if normalize:
        spec = (spec - spec.mean()) / (spec.std() + 1e-6)
    return spec


class SpectrogramDataset(Dataset):
    """Dataset over (waveform, transcript) pairs; synthesises waveforms when
    no audio files exist (offline CI) — shapes match AN4 utterances."""

    LABELS = "_'abcdefghijklmnopqrstuvwxyz "

    def __init__(
        self,
        items: Optional[List[Tuple[torch.Tensor, str]]] = None,
        n_synthetic: int = 0,
        sample_rate: int = 16000,
        seed: int = 0,
    ):
        self.sample_rate = sample_rate
        if items is None:
            g = torch.Generator().manual_seed(seed)
            items = []
            for i in range(n_synthetic):
                dur = int(sample_rate * (1.0 + (i % 5) * 0.25))
                wave = torch.randn(dur, generator=g) * 0.1
                text = "synthetic utterance " + str(i)
                items.append((wave, text))
        self.items = items
        self.label2id = {c: i for i, c in enumerate(self.LABELS)}

    def __len__(self) -> int:
        return len(self.items)

    def __getitem__(self, idx: int):
        wave, text = self.items[idx]
        spec = compute_spectrogram(wave, self.sample_rate)
        target = torch.tensor(
            [self.label2id[c] for c in text.lower() if c in self.label2id],
            dtype=torch.long,
        )
        return spec, target

    @staticmethod
    def collate(batch):
        """Pad to the longest utterance; returns (N,1,F,T), targets, lens."""
        specs = [b[0] for b in batch]
        targets = [b[1] for b in batch]
        fmax = specs[0].size(0)
        tmax = max(s.size(1) for s in specs)
        x = torch.zeros(len(batch), 1, fmax, tmax)
        in_lens = torch.zeros(len(batch), dtype=torch.int32)
        for i, s in enumerate(specs):
            x[i, 0, :, : s.size(1)] = s
            in_lens[i] = s.size(1)
        tgt_lens = torch.tensor([t.numel() for t in targets], dtype=torch.int32)
        flat_targets = torch.cat(targets) if targets else torch.empty(0, dtype=torch.long)
        return x, flat_targets, in_lens, tgt_lens
