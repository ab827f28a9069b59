"""Masked text compact transformers (reference: models/cifar10/cctnets/
text/{cct.py, transformer.py} + utils/{embedder.py, tokenizer.py
TextTokenizer, transformers.py MaskedAttention/MaskedTransformerClassifier}).

Own single-file design in the style of :mod:`cct`: word embedding with a
padding mask, an optional 1-D conv tokenizer (kernel over the sequence,
max-pooled, mask downsampled alongside), SDPA attention with a key-padding
mask, and masked sequence pooling.  The reference's simulator never uses
these; they exist for zoo parity and run on CPU/GPU like
any other model here.

Factory names match the reference registry: ``text_cct_2/4/6``,
``text_transformer_2/4/6``.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .cct import DropPath


class Embedder(nn.Module):
    """Token ids -> word vectors; padding id contributes a masked slot."""

    def __init__(self, vocab_size: int = 10000, word_embedding_dim: int = 300,
                 padding_idx: int = 0, **_):
        super().__init__()
        self.emb = nn.Embedding(vocab_size, word_embedding_dim,
                                padding_idx=padding_idx)
        self.padding_idx = padding_idx

    def forward(self, ids: torch.Tensor, mask: Optional[torch.Tensor] = None):
        if mask is None:
            mask = ids != self.padding_idx
        return self.emb(ids), mask


class TextTokenizer(nn.Module):
    """1-D conv over the sequence + max-pool; the mask pools alongside so
    a downsampled position is valid iff any source position was."""

    def __init__(self, in_dim: int, out_dim: int, kernel_size: int = 2,
                 stride: int = 1, padding: int = 1, pool_kernel: int = 2,
                 pool_stride: int = 2, pool_padding: int = 1):
        super().__init__()
        self.conv = nn.Conv1d(in_dim, out_dim, kernel_size, stride=stride,
                              padding=padding, bias=False)
        self.pool = nn.MaxPool1d(pool_kernel, stride=pool_stride,
                                 padding=pool_padding)

    def seq_len(self, seq_len: int) -> int:
        with torch.no_grad():
            x = torch.zeros(1, seq_len, self.conv.in_channels)
            return self.forward(x)[0].shape[1]

    def forward(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None):
        # x: [B, N, E] -> conv over N
        y = self.pool(F.relu(self.conv(x.transpose(1, 2)))).transpose(1, 2)
        if mask is not None:
            # propagate validity through the SAME geometry with a ones
            # kernel (never the learned weights — near-zero weights would
            # silently invalidate real positions): a downsampled slot is
            # valid iff its receptive field saw any valid token
            ones = torch.ones(1, 1, self.conv.kernel_size[0],
                              device=mask.device)
            m = F.conv1d(mask.float().unsqueeze(1), ones,
                         stride=self.conv.stride[0],
                         padding=self.conv.padding[0])
            m = self.pool(m).squeeze(1)
            mask = (m > 0)[:, :y.shape[1]]
        return y, mask


class MaskedEncoderLayer(nn.Module):
    """Pre-norm encoder layer (cct.EncoderLayer topology) with a
    key-padding mask applied inside SDPA."""

    def __init__(self, dim: int, heads: int, mlp_ratio: float,
                 dropout: float, attn_dropout: float, drop_path: float):
        super().__init__()
        self.pre_norm = nn.LayerNorm(dim)
        self.heads = heads
        self.qkv = nn.Linear(dim, dim * 3, bias=False)
        self.attn_dropout = attn_dropout
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(dropout)
        self.norm1 = nn.LayerNorm(dim)
        hidden = int(dim * mlp_ratio)
        self.linear1 = nn.Linear(dim, hidden)
        self.dropout1 = nn.Dropout(dropout)
        self.linear2 = nn.Linear(hidden, dim)
        self.dropout2 = nn.Dropout(dropout)
        self.drop_path = DropPath(drop_path)

    def _attn(self, x, mask):
        B, N, D = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.heads, D // self.heads)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)
        attn_mask = None
        if mask is not None:
            attn_mask = mask[:, None, None, :]  # broadcast over heads/query
        o = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask,
            dropout_p=self.attn_dropout if self.training else 0.0)
        o = o.transpose(1, 2).reshape(B, N, D)
        return self.proj_drop(self.proj(o))

    def forward(self, x, mask=None):
        x = x + self.drop_path(self._attn(self.pre_norm(x), mask))
        x = self.norm1(x)
        h = self.linear2(self.dropout1(F.gelu(self.linear1(x))))
        return x + self.drop_path(self.dropout2(h))


class MaskedTextClassifier(nn.Module):
    """Encoder stack + masked seq-pool head over valid positions."""

    def __init__(self, seq_len: int, dim: int, num_layers: int, heads: int,
                 mlp_ratio: float, num_classes: int, dropout: float = 0.0,
                 attn_dropout: float = 0.1, stochastic_depth: float = 0.1):
        super().__init__()
        self.pos_emb = nn.Parameter(torch.empty(1, seq_len, dim))
        nn.init.trunc_normal_(self.pos_emb, std=0.2)
        self.dropout = nn.Dropout(dropout)
        dpr = torch.linspace(0, stochastic_depth, num_layers).tolist()
        self.layers = nn.ModuleList([
            MaskedEncoderLayer(dim, heads, mlp_ratio, dropout, attn_dropout,
                               p)
            for p in dpr])
        self.norm = nn.LayerNorm(dim)
        self.seq_pool = nn.Linear(dim, 1)
        self.fc = nn.Linear(dim, num_classes)

    def forward(self, x, mask=None):
        n = x.shape[1]
        x = self.dropout(x + self.pos_emb[:, :n])
        for layer in self.layers:
            x = layer(x, mask)
        x = self.norm(x)
        w = self.seq_pool(x).squeeze(-1)  # [B, N]
        if mask is not None:
            w = w.masked_fill(~mask[:, :n], float("-inf"))
        pooled = torch.einsum("bn,bnd->bd", w.softmax(dim=1), x)
        return self.fc(pooled)


class TextCCT(nn.Module):
    def __init__(self, seq_len: int = 64, word_embedding_dim: int = 300,
                 embedding_dim: int = 128, num_layers: int = 2,
                 num_heads: int = 2, mlp_ratio: float = 1.0,
                 num_classes: int = 2, vocab_size: int = 10000,
                 kernel_size: int = 2, use_tokenizer: bool = True, **kw):
        super().__init__()
        self.embedder = Embedder(vocab_size, word_embedding_dim)
        self.tokenizer = None
        dim = word_embedding_dim
        eff_len = seq_len
        if use_tokenizer:
            self.tokenizer = TextTokenizer(word_embedding_dim, embedding_dim,
                                           kernel_size=kernel_size)
            dim = embedding_dim
            eff_len = self.tokenizer.seq_len(seq_len)
        self.classifier = MaskedTextClassifier(
            eff_len, dim, num_layers, num_heads, mlp_ratio, num_classes, **kw)

    def forward(self, ids, mask=None):
        x, mask = self.embedder(ids, mask)
        if self.tokenizer is not None:
            x, mask = self.tokenizer(x, mask)
        return self.classifier(x, mask)


def _cct(layers, heads, ratio, dim, **kw):
    return TextCCT(num_layers=layers, num_heads=heads, mlp_ratio=ratio,
                   embedding_dim=dim, **kw)


def text_cct_2(**kw):
    return _cct(2, 2, 1, 128, **kw)


def text_cct_4(**kw):
    return _cct(4, 2, 1, 128, **kw)


def text_cct_6(**kw):
    return _cct(6, 4, 2, 256, **kw)


def text_transformer_2(**kw):
    return TextCCT(num_layers=2, num_heads=2, mlp_ratio=1,
                   use_tokenizer=False, **kw)


def text_transformer_4(**kw):
    return TextCCT(num_layers=4, num_heads=2, mlp_ratio=1,
                   use_tokenizer=False, **kw)


def text_transformer_6(**kw):
    return TextCCT(num_layers=6, num_heads=4, mlp_ratio=2,
                   use_tokenizer=False, **kw)
