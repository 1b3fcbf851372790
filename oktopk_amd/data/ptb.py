"""PTB-style word-LM reader (reference VGG/ptb_reader.py): builds a word
vocab from train text, converts splits to id streams, batchifies into
(seq_len, batch) blocks."""
from __future__ import annotations

import collections
import os
from typing import Dict, List

import torch


class PTBReader:
    def __init__(self, data_dir: str, train: str = "ptb.train.txt",
                 valid: str = "ptb.valid.txt", test: str = "ptb.test.txt"):
        self.word2id: Dict[str, int] = {}
        self.train_ids = self._load(os.path.join(data_dir, train), build=True)
        self.valid_ids = self._load(os.path.join(data_dir, valid))
        self.test_ids = self._load(os.path.join(data_dir, test))

    @property
    def vocab_size(self) -> int:
        return len(self.word2id)

    def _load(self, path: str, build: bool = False) -> torch.Tensor:
        if not os.path.exists(path):
            return torch.empty(0, dtype=torch.long)
        words: List[str] = []
        with open(path, "r", encoding="utf-8") as f:
            for line in f:
                words.extend(line.split() + ["<eos>"])
        if build:
            counter = collections.Counter(words)
            for w, _ in counter.most_common():
                self.word2id.setdefault(w, len(self.word2id))
        unk = self.word2id.setdefault("<unk>", len(self.word2id))
        return torch.tensor([self.word2id.get(w, unk) for w in words], dtype=torch.long)


def ptb_batchify(ids: torch.Tensor, batch_size: int, seq_len: int,
                 rank: int = 0, world: int = 1):
    """Yield (input, target) of shape (seq_len, batch) — rank-sharded
    contiguous streams (DistributedSampler equivalent for LM streams)."""
    per = ids.numel() // world
    shard = ids[rank * per : (rank + 1) * per]
    nbatch = shard.numel() // batch_size
    shard = shard[: nbatch * batch_size].view(batch_size, -1).t().contiguous()
    for i in range(0, shard.size(0) - 1 - seq_len, seq_len):
        yield shard[i : i + seq_len], shard[i + 1 : i + 1 + seq_len]
