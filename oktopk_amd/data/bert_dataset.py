"""BERT MLM+NSP pretraining dataset.

Capability parity with the reference BERTDataset
(/root/reference/BERT/bert/main_bert.py:257-461 and BERT/bert/dataset.py:93
BERTDatasetPartitioned): sentence-pair sampling with 50% random next
sentence, 15% masked-LM masking (80% [MASK] / 10% random / 10% keep),
padding/truncation to max_seq_length, per-rank partitioning.

Works from any plain-text corpus (one sentence per line, blank line between
documents); no network needed.
"""
from __future__ import annotations

import random
from typing import Dict, Sequence

import torch
from torch.utils.data import Dataset

from .tokenization import WordPieceTokenizer

SPECIALS = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]"]


def build_vocab_from_corpus(lines: Sequence[str], max_size: int = 30000) -> Dict[str, int]:
    """Word-level vocab builder for offline corpora (the reference ships a
    pretrained WordPiece vocab file; absent one, whole words are pieces)."""
    import collections

    counter = collections.Counter()
    for ln in lines:
        counter.update(ln.lower().split())
    vocab = {t: i for i, t in enumerate(SPECIALS)}
    for w, _ in counter.most_common(max_size - len(vocab)):
        vocab.setdefault(w, len(vocab))
    return vocab


class BertPretrainingDataset(Dataset):
    def __init__(
        self,
        documents: Sequence[Sequence[str]],
        tokenizer: WordPieceTokenizer,
        max_seq_length: int = 128,
        mlm_prob: float = 0.15,
        seed: int = 12345,
        rank: int = 0,
        world: int = 1,
    ):
        self.tok = tokenizer
        self.max_len = max_seq_length
        self.mlm_prob = mlm_prob
        self.rng = random.Random(seed + rank)
        # flatten documents into (doc_idx, sent_idx) pairs, rank-sharded
        self.docs = [
            [self.tok.convert_tokens_to_ids(self.tok.tokenize(s)) for s in doc]
            for doc in documents
        ]
        self.pairs = []
        for d, doc in enumerate(self.docs):
            for s in range(len(doc) - 1):
                self.pairs.append((d, s))
        self.pairs = self.pairs[rank::world]
        self.mask_id = self.tok.vocab["[MASK]"]
        self.cls_id = self.tok.vocab["[CLS]"]
        self.sep_id = self.tok.vocab["[SEP]"]
        self.pad_id = self.tok.vocab["[PAD]"]
        self.vocab_size = len(self.tok.vocab)

    def __len__(self) -> int:
        return len(self.pairs)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        d, s = self.pairs[idx]
        a = self.docs[d][s]
        # 50%: true next sentence; 50%: random sentence from another doc
        if self.rng.random() < 0.5 or len(self.docs) < 2:
            b = self.docs[d][s + 1]
            is_next = 0
        else:
            rd = self.rng.randrange(len(self.docs))
            while rd == d:
                rd = self.rng.randrange(len(self.docs))
            rb = self.docs[rd]
            b = rb[self.rng.randrange(len(rb))] if rb else []
            is_next = 1
        max_tokens = self.max_len - 3
        a = list(a[: max_tokens // 2])
        b = list(b[: max_tokens - len(a)])
        ids = [self.cls_id] + a + [self.sep_id] + b + [self.sep_id]
        type_ids = [0] * (len(a) + 2) + [1] * (len(b) + 1)
        labels = [-1] * len(ids)
        # MLM masking (reference main_bert.py:497-532)
        for i in range(1, len(ids)):
            if ids[i] in (self.sep_id, self.cls_id):
                continue
            if self.rng.random() < self.mlm_prob:
                labels[i] = ids[i]
                r = self.rng.random()
                if r < 0.8:
                    ids[i] = self.mask_id
                elif r < 0.9:
                    ids[i] = self.rng.randrange(len(SPECIALS), self.vocab_size)
        attn = [1] * len(ids)
        pad = self.max_len - len(ids)
        ids += [self.pad_id] * pad
        type_ids += [0] * pad
        attn += [0] * pad
        labels += [-1] * pad
        return {
            "input_ids": torch.tensor(ids[: self.max_len], dtype=torch.long),
            "token_type_ids": torch.tensor(type_ids[: self.max_len], dtype=torch.long),
            "attention_mask": torch.tensor(attn[: self.max_len], dtype=torch.long),
            "masked_lm_labels": torch.tensor(labels[: self.max_len], dtype=torch.long),
            "next_sentence_label": torch.tensor(is_next, dtype=torch.long),
        }
