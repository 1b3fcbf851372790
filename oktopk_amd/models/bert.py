"""BERT encoder for MLM+NSP pretraining, written fresh for this framework.

Capability parity with the reference's HF fork
(/root/reference/BERT/bert/transformers/modeling.py:59-1159): embeddings,
post-LN transformer encoder, pooler, tied MLM head + NSP head, and the
bert-base / bert-large configs the reference trains
(BERT/bert/configs/bert_config_bert-base-uncased.json).

Attention is explicit matmul + softmax (the reference's BertSelfAttention is
the same vanilla form, modeling.py:288); the matmuls run on MFMA via
rocBLAS/hipBLASLt under bf16 autocast.  Stage partitioning for the pipeline
runtime slices `encoder.layer`.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

# torch SDPA (flash attention) in place of explicit matmul+softmax — default
# ON: measured 11.69 -> 10.38 ms/step on BERT-base seq128 and 20.4 -> 17.4 at
# seq512 (profiles/README.md r01-n).  OKTOPK_SDPA=0 restores the explicit
# form (reference parity shape, modeling.py:288).
_USE_SDPA = os.environ.get("OKTOPK_SDPA", "1") != "0"


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    hidden_dropout_prob: float = 0.1
    attention_probs_dropout_prob: float = 0.1
    layer_norm_eps: float = 1e-12
    initializer_range: float = 0.02


class BertEmbeddings(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position_embeddings = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size, cfg.hidden_size)
        self.ln = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.dropout = nn.Dropout(cfg.hidden_dropout_prob)

    def forward(self, input_ids, token_type_ids=None):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        x = (
            self.word_embeddings(input_ids)
            + self.position_embeddings(pos)
            + self.token_type_embeddings(token_type_ids)
        )
        return self.dropout(self.ln(x))


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.nh = cfg.num_attention_heads
        self.hd = cfg.hidden_size // cfg.num_attention_heads
        from ..ops.fused_linear import ColsumLinear

        self.qkv = ColsumLinear(cfg.hidden_size, 3 * cfg.hidden_size)
        self.out = ColsumLinear(cfg.hidden_size, cfg.hidden_size)
        self.attn_drop = nn.Dropout(cfg.attention_probs_dropout_prob)

    def forward(self, x, attn_mask=None):
        b, s, h = x.shape
        qkv_flat = self.qkv(x)
        # hand-written CDNA4 fused attention (QK^T+softmax+dropout+PV in one
        # MFMA kernel) on GPU/bf16 at the reference shape; eager otherwise
        from ..ops.fused_attn import fused_attention, fused_attn_available

        if fused_attn_available(qkv_flat, self.nh, s, self.attn_drop.p):
            ctx = fused_attention(qkv_flat, attn_mask, self.nh,
                                  self.attn_drop.p, self.training)
            return self.out(ctx)
        qkv = qkv_flat.view(b, s, 3, self.nh, self.hd).permute(2, 0, 3, 1, 4)
        q, k, v = qkv[0], qkv[1], qkv[2]  # (b, nh, s, hd)
        if _USE_SDPA and x.is_cuda:
            # torch flash-attention path (opt-in A/B vs explicit matmuls)
            ctx = F.scaled_dot_product_attention(
                q, k, v, attn_mask=attn_mask,
                dropout_p=self.attn_drop.p if self.training else 0.0)
            ctx = ctx.transpose(1, 2).reshape(b, s, h)
            return self.out(ctx)
        scores = torch.matmul(q, k.transpose(-1, -2)) / math.sqrt(self.hd)
        if attn_mask is not None:
            scores = scores + attn_mask
        probs = self.attn_drop(F.softmax(scores, dim=-1))
        ctx = torch.matmul(probs, v)  # (b, nh, s, hd)
        ctx = ctx.transpose(1, 2).reshape(b, s, h)
        return self.out(ctx)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        from ..ops.fused_linear import ColsumLinear

        self.fc1 = ColsumLinear(cfg.hidden_size, cfg.intermediate_size)
        self.fc2 = ColsumLinear(cfg.intermediate_size, cfg.hidden_size)
        self.ln2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.drop = nn.Dropout(cfg.hidden_dropout_prob)

    def forward(self, x, attn_mask=None):
        a = self.drop(self.attn(x, attn_mask))
        x = self._add_ln(x, a, self.ln1)
        h = self._mlp_act(x)
        o = self.drop(self.fc2(h))
        x = self._add_ln(x, o, self.ln2)
        return x

    def _add_ln(self, x, r, ln):
        # fused residual+LayerNorm CDNA4 kernel on GPU/bf16
        # (OKTOPK_NO_FUSED_LN=1 opts out); torch path otherwise
        from ..ops.fused_ln import fused_add_layernorm, fused_ln_available

        if fused_ln_available(x):
            return fused_add_layernorm(x, r, ln)
        return ln(x + r)

    def _mlp_act(self, x):
        # hand-written MFMA fused GEMM+bias+GELU on GPU/bf16 (reference
        # LinearActivation fused-gelu, modeling.py:75); torch path otherwise
        from ..ops.fused_linear import fused_available, fused_linear_gelu

        if fused_available(x, self.fc1.weight):
            return fused_linear_gelu(x, self.fc1.weight, self.fc1.bias)
        return F.gelu(self.fc1(x))


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = BertEmbeddings(cfg)
        self.layer = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.pooler = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, self.cfg.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()
        elif isinstance(m, nn.LayerNorm):
            m.weight.data.fill_(1.0)
            m.bias.data.zero_()

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        mask = None
        if attention_mask is not None:
            # (b, s) {0,1} -> additive (b, 1, 1, s), in the compute dtype
            wdt = self.embeddings.word_embeddings.weight.dtype
            mask = (1.0 - attention_mask[:, None, None, :].to(wdt)) * -10000.0
        x = self.embeddings(input_ids, token_type_ids)
        for lyr in self.layer:
            x = lyr(x, mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled


class BertForPreTraining(nn.Module):
    """MLM (tied decoder) + NSP heads, reference BertForPreTraining
    (modeling.py heads + BertPreTrainingHeads)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.bert = BertModel(cfg)
        self.transform = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.transform_ln = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps)
        self.decoder_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.nsp = nn.Linear(cfg.hidden_size, 2)

    def forward(
        self,
        input_ids,
        token_type_ids=None,
        attention_mask=None,
        masked_lm_labels=None,
        next_sentence_label=None,
    ):
        seq, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        h = self.transform_ln(F.gelu(self.transform(seq)))
        # tied with word embeddings (reference ties decoder to embedding matrix)
        logits = F.linear(h, self.bert.embeddings.word_embeddings.weight, self.decoder_bias)
        nsp_logits = self.nsp(pooled)
        if masked_lm_labels is None:
            return logits, nsp_logits
        mlm_loss = F.cross_entropy(
            logits.view(-1, logits.size(-1)), masked_lm_labels.view(-1), ignore_index=-1
        )
        loss = mlm_loss
        if next_sentence_label is not None:
            loss = loss + F.cross_entropy(nsp_logits.view(-1, 2), next_sentence_label.view(-1))
        return loss


def bert_base(**kw) -> BertForPreTraining:
    return BertForPreTraining(BertConfig(**kw))


def bert_large(**kw) -> BertForPreTraining:
    cfg = dict(hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
               intermediate_size=4096)
    cfg.update(kw)
    return BertForPreTraining(BertConfig(**cfg))
