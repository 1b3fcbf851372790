"""Data layer tests: tokenizer, PTB reader, BERT dataset, audio, vision."""
import torch

from oktopk_amd.data import (
    BasicTokenizer,
    BertPretrainingDataset,
    PTBReader,
    SpectrogramDataset,
    WordPieceTokenizer,
    cifar_like_dataset,
    compute_spectrogram,
    ptb_batchify,
)
from oktopk_amd.data.bert_dataset import build_vocab_from_corpus


def test_basic_tokenizer():
    t = BasicTokenizer()
    assert t.tokenize("Hello, World!  ") == ["hello", ",", "world", "!"]


def test_wordpiece_greedy():
    vocab = {t: i for i, t in enumerate(
        ["[UNK]", "[PAD]", "un", "##aff", "##able", "hello", "runn", "##ing"]
    )}
    tok = WordPieceTokenizer(vocab=vocab)
    assert tok.tokenize("unaffable") == ["un", "##aff", "##able"]
    assert tok.tokenize("hello running") == ["hello", "runn", "##ing"]
    assert tok.tokenize("xyzzy") == ["[UNK]"]
    ids = tok.convert_tokens_to_ids(["hello", "nope"])
    assert ids == [5, 0]
    assert tok.convert_ids_to_tokens([5]) == ["hello"]


def test_ptb_reader(tmp_path):
    (tmp_path / "ptb.train.txt").write_text("the cat sat\nthe dog ran\n")
    (tmp_path / "ptb.valid.txt").write_text("the cat ran\n")
    r = PTBReader(str(tmp_path), test="missing.txt")
    assert r.vocab_size >= 6
    assert r.train_ids.numel() == 8  # 6 words + 2 <eos>
    batches = list(ptb_batchify(torch.arange(100), batch_size=2, seq_len=10))
    x, y = batches[0]
    assert x.shape == (10, 2)
    assert torch.equal(y[0], x[1])  # shifted by one


def test_bert_dataset_masking():
    docs = [
        ["the quick brown fox jumps", "over the lazy dog", "again and again"],
        ["completely different document", "with other sentences here"],
    ]
    vocab = build_vocab_from_corpus([s for d in docs for s in d])
    tok = WordPieceTokenizer(vocab=vocab)
    ds = BertPretrainingDataset(docs, tok, max_seq_length=32, seed=0)
    assert len(ds) == 3
    item = ds[0]
    assert item["input_ids"].shape == (32,)
    assert item["attention_mask"].sum() > 4
    assert int(item["next_sentence_label"]) in (0, 1)
    # labels only at masked positions
    lab = item["masked_lm_labels"]
    assert (lab[item["attention_mask"] == 0] == -1).all()


def test_spectrogram_shape():
    wave = torch.randn(16000)
    spec = compute_spectrogram(wave)
    assert spec.size(0) == 161  # matches DeepSpeech conv frontend expectation


def test_spectrogram_dataset_collate():
    ds = SpectrogramDataset(n_synthetic=4)
    batch = [ds[i] for i in range(4)]
    x, targets, in_lens, tgt_lens = SpectrogramDataset.collate(batch)
    assert x.dim() == 4 and x.size(1) == 1 and x.size(2) == 161
    assert tgt_lens.sum() == targets.numel()
    # feed through the real model
    from oktopk_amd import models

    m = models.create_net("lstman4", rnn_hidden_size=32)
    out = m(x)
    assert out.size(1) == 4


def test_cifar_synthetic_fallback(tmp_path):
    ds = cifar_like_dataset(str(tmp_path), n_synthetic=16)
    x, y = ds[0]
    assert x.shape == (3, 32, 32) and 0 <= int(y) < 10
