"""WordPiece tokenizer (self-contained).

Capability parity with the reference's vendored BERT tokenizer
(/root/reference/BERT/bert/transformers/tokenization.py): basic whitespace +
punctuation splitting, lowercasing, greedy longest-match-first WordPiece with
'##' continuation pieces, vocab load from file, convert tokens<->ids.
"""
from __future__ import annotations

import collections
import unicodedata
from typing import Dict, List, Optional


def load_vocab(vocab_file: str) -> Dict[str, int]:
    vocab = collections.OrderedDict()
    with open(vocab_file, "r", encoding="utf-8") as f:
        for i, line in enumerate(f):
            tok = line.rstrip("\n")
            if tok:
                vocab[tok] = i
    return vocab


def _is_whitespace(ch: str) -> bool:
    return ch in " \t\n\r" or unicodedata.category(ch) == "Zs"


def _is_control(ch: str) -> bool:
    if ch in ("\t", "\n", "\r"):
        return False
    return unicodedata.category(ch).startswith("C")


def _is_punct(ch: str) -> bool:
    cp = ord(ch)
    if (33 <= cp <= 47) or (58 <= cp <= 64) or (91 <= cp <= 96) or (123 <= cp <= 126):
        return True
    return unicodedata.category(ch).startswith("P")


class BasicTokenizer:
    """Whitespace cleanup, lowercasing, accent stripping, punctuation split."""

    def __init__(self, do_lower_case: bool = True):
        self.do_lower_case = do_lower_case

    def tokenize(self, text: str) -> List[str]:
        text = "".join(
            " " if _is_whitespace(c) else c
            for c in text
            if not (_is_control(c) or ord(c) == 0 or ord(c) == 0xFFFD)
        )
        tokens = []
        for tok in text.strip().split():
            if self.do_lower_case:
                tok = tok.lower()
                tok = "".join(
                    c for c in unicodedata.normalize("NFD", tok)
                    if unicodedata.category(c) != "Mn"
                )
            tokens.extend(self._split_punct(tok))
        return [t for t in tokens if t]

    @staticmethod
    def _split_punct(tok: str) -> List[str]:
        out, cur = [], []
        for c in tok:
            if _is_punct(c):
                if cur:
                    out.append("".join(cur))
                    cur = []
                out.append(c)
            else:
                cur.append(c)
        if cur:
            out.append("".join(cur))
        return out


class WordPieceTokenizer:
    def __init__(
        self,
        vocab: Optional[Dict[str, int]] = None,
        vocab_file: Optional[str] = None,
        do_lower_case: bool = True,
        unk_token: str = "[UNK]",
        max_chars_per_word: int = 100,
    ):
        if vocab is None and vocab_file is not None:
            vocab = load_vocab(vocab_file)
        if vocab is None:
            raise ValueError("need vocab or vocab_file")
        self.vocab = dict(vocab)
        self.inv_vocab = {v: k for k, v in self.vocab.items()}
        self.basic = BasicTokenizer(do_lower_case)
        self.unk = unk_token
        self.max_chars = max_chars_per_word

    def tokenize(self, text: str) -> List[str]:
        pieces: List[str] = []
        for word in self.basic.tokenize(text):
            pieces.extend(self._wordpiece(word))
        return pieces

    def _wordpiece(self, word: str) -> List[str]:
        if len(word) > self.max_chars:
            return [self.unk]
        out: List[str] = []
        start = 0
        while start < len(word):
            end = len(word)
            cur = None
            while start < end:
                sub = word[start:end]
                if start > 0:
                    sub = "##" + sub
                if sub in self.vocab:
                    cur = sub
                    break
                end -= 1
            if cur is None:
                return [self.unk]
            out.append(cur)
            start = end
        return out

    def convert_tokens_to_ids(self, tokens: List[str]) -> List[int]:
        unk_id = self.vocab.get(self.unk, 0)
        return [self.vocab.get(t, unk_id) for t in tokens]

    def convert_ids_to_tokens(self, ids: List[int]) -> List[str]:
        return [self.inv_vocab.get(i, self.unk) for i in ids]
