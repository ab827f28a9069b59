"""Compact Convolutional Transformer (CCT), written for this framework.

Architecture parity with the reference's vendored Compact-Transformers
(reference: src/blades/models/cifar10/cctnets/cct.py:121-123 `cct_2`,
utils/tokenizer.py:6-50, utils/transformers.py:134-233): conv tokenizer
(n 3x3 convs, ReLU + maxpool) -> pre-norm transformer encoder with
stochastic depth -> sequence pooling (softmax attention over tokens) ->
linear head.  ``cct_2_3x2_32`` (2 layers, dim 128, 2 heads, mlp_ratio 1,
2-conv tokenizer) is the model the reference uses for CIFAR-10, d ≈ 284k.

Implementation choices are our own: a single file, fused
``scaled_dot_product_attention`` for the encoder (maps to the ROCm
flash-attention path on MI355X instead of 5 separate matmul/softmax
kernels), and no pretrained-URL machinery (no network in this
environment).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class DropPath(nn.Module):
    """Stochastic depth per sample."""

    def __init__(self, p: float = 0.0):
        super().__init__()
        self.p = p

    def forward(self, x):
        if self.p == 0.0 or not self.training:
            return x
        keep = 1.0 - self.p
        mask = x.new_empty(x.shape[0], *([1] * (x.ndim - 1))).bernoulli_(keep)
        return x * mask / keep


class Tokenizer(nn.Module):
    def __init__(self, n_conv_layers: int = 2, in_channels: int = 3,
                 embedding_dim: int = 128, in_planes: int = 64,
                 kernel_size: int = 3, stride: int = 1, padding: int = 1,
                 pool_kernel: int = 3, pool_stride: int = 2, pool_padding: int = 1):
        super().__init__()
        chans = [in_channels] + [in_planes] * (n_conv_layers - 1) + [embedding_dim]
        layers = []
        for i in range(n_conv_layers):
            layers += [
                nn.Conv2d(chans[i], chans[i + 1], kernel_size, stride=stride,
                          padding=padding, bias=False),
                nn.ReLU(inplace=True),
                nn.MaxPool2d(pool_kernel, stride=pool_stride, padding=pool_padding),
            ]
        self.conv_layers = nn.Sequential(*layers)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Conv2d):
            nn.init.kaiming_normal_(m.weight)

    def sequence_length(self, in_channels=3, height=32, width=32) -> int:
        with torch.no_grad():
            return self.forward(torch.zeros(1, in_channels, height, width)).shape[1]

    def forward(self, x):
        # [B,C,H,W] -> [B, H'*W', D]
        return self.conv_layers(x).flatten(2, 3).transpose(-2, -1)


class EncoderLayer(nn.Module):
    """Pre-norm transformer encoder layer matching the reference topology:
    x = x + drop_path(attn(pre_norm(x))); x = norm1(x);
    x = x + drop_path(mlp(x)).
    """

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

    def _attn(self, x):
        B, N, D = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.heads, D // self.heads)
        q, k, v = qkv.permute(2, 0, 3, 1, 4).unbind(0)  # [B, h, N, d]
        o = F.scaled_dot_product_attention(
            q, k, v, dropout_p=self.attn_dropout if self.training else 0.0)
        o = o.transpose(1, 2).reshape(B, N, D)
        return self.proj_drop(self.proj(o))

    def forward(self, x):
        x = x + self.drop_path(self._attn(self.pre_norm(x)))
        x = self.norm1(x)
        h = self.linear2(self.dropout1(F.gelu(self.linear1(x))))
        return x + self.drop_path(self.dropout2(h))


class PatchEmbed(nn.Module):
    """Non-overlapping patch embedding (ViT-Lite / CVT tokenizer)."""

    def __init__(self, in_channels: int = 3, embedding_dim: int = 128,
                 patch_size: int = 4):
        super().__init__()
        self.proj = nn.Conv2d(in_channels, embedding_dim, patch_size,
                              stride=patch_size)

    def sequence_length(self, in_channels=3, height=32, width=32) -> int:
        with torch.no_grad():
            return self.forward(torch.zeros(1, in_channels, height, width)).shape[1]

    def forward(self, x):
        return self.proj(x).flatten(2, 3).transpose(-2, -1)


class CCT(nn.Module):
    """Compact transformer family.

    ``tokenizer='conv'`` + ``pool='seq'``  -> CCT (default);
    ``tokenizer='patch'`` + ``pool='seq'`` -> CVT;
    ``tokenizer='patch'`` + ``pool='cls'`` -> ViT-Lite.
    (reference zoo: cctnets/cct.py, cvt.py, vit.py)
    """

    def __init__(self, img_size: int = 32, embedding_dim: int = 128,
                 num_layers: int = 2, num_heads: int = 2, mlp_ratio: float = 1.0,
                 n_conv_layers: int = 2, num_classes: int = 10,
                 dropout: float = 0.0, attn_dropout: float = 0.1,
                 stochastic_depth: float = 0.1, in_channels: int = 3,
                 positional_embedding: str = "learnable",
                 tokenizer: str = "conv", pool: str = "seq",
                 patch_size: int = 4):
        super().__init__()
        if tokenizer == "conv":
            self.tokenizer = Tokenizer(n_conv_layers=n_conv_layers,
                                       in_channels=in_channels,
                                       embedding_dim=embedding_dim)
        else:
            self.tokenizer = PatchEmbed(in_channels=in_channels,
                                        embedding_dim=embedding_dim,
                                        patch_size=patch_size)
        self.pool = pool
        seq_len = self.tokenizer.sequence_length(in_channels, img_size, img_size)
        if pool == "cls":
            self.class_emb = nn.Parameter(torch.zeros(1, 1, embedding_dim))
            seq_len += 1
        self.seq_len = seq_len

        if positional_embedding == "learnable":
            self.positional_emb = nn.Parameter(torch.zeros(1, seq_len, embedding_dim))
            nn.init.trunc_normal_(self.positional_emb, std=0.2)
        elif positional_embedding == "sine":
            self.register_buffer("positional_emb",
                                 self._sinusoidal(seq_len, embedding_dim))
        else:
            self.positional_emb = None

        self.dropout = nn.Dropout(dropout)
        dpr = torch.linspace(0, stochastic_depth, num_layers).tolist()
        self.blocks = nn.ModuleList([
            EncoderLayer(embedding_dim, num_heads, mlp_ratio,
                         dropout, attn_dropout, dpr[i])
            for i in range(num_layers)
        ])
        self.norm = nn.LayerNorm(embedding_dim)
        self.attention_pool = nn.Linear(embedding_dim, 1)
        self.fc = nn.Linear(embedding_dim, num_classes)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, nn.Linear):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.LayerNorm):
            nn.init.zeros_(m.bias)
            nn.init.ones_(m.weight)

    @staticmethod
    def _sinusoidal(n: int, dim: int) -> torch.Tensor:
        pos = torch.arange(n, dtype=torch.float32).unsqueeze(1)
        i = torch.arange(dim, dtype=torch.float32).unsqueeze(0)
        angle = pos / torch.pow(10000.0, 2 * (i // 2) / dim)
        pe = torch.where(i.long() % 2 == 0, torch.sin(angle), torch.cos(angle))
        return pe.unsqueeze(0)

    def forward(self, x):
        x = self.tokenizer(x)
        if self.pool == "cls":
            cls = self.class_emb.expand(x.shape[0], -1, -1)
            x = torch.cat([cls, x], dim=1)
        if self.positional_emb is not None:
            x = x + self.positional_emb
        x = self.dropout(x)
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        if self.pool == "cls":
            x = x[:, 0]
        else:
            # sequence pooling: softmax(Wx) over tokens, weighted sum
            w = F.softmax(self.attention_pool(x), dim=1)  # [B, N, 1]
            x = torch.matmul(w.transpose(-1, -2), x).squeeze(-2)
        return self.fc(x)


def cct_2_3x2_32(num_classes: int = 10, img_size: int = 32, **kw) -> CCT:
    return CCT(img_size=img_size, embedding_dim=128, num_layers=2, num_heads=2,
               mlp_ratio=1.0, n_conv_layers=2, num_classes=num_classes, **kw)


def cvt_2_4_32(num_classes: int = 10, img_size: int = 32, **kw) -> CCT:
    """Compact Vision Transformer: patch embedding + sequence pooling."""
    return CCT(img_size=img_size, embedding_dim=128, num_layers=2, num_heads=2,
               mlp_ratio=1.0, num_classes=num_classes, tokenizer="patch",
               pool="seq", patch_size=4, **kw)


def vit_lite_2_4_32(num_classes: int = 10, img_size: int = 32, **kw) -> CCT:
    """ViT-Lite: patch embedding + class token."""
    return CCT(img_size=img_size, embedding_dim=128, num_layers=2, num_heads=2,
               mlp_ratio=1.0, num_classes=num_classes, tokenizer="patch",
               pool="cls", patch_size=4, **kw)


class CCTNet(nn.Module):
    """Reference-name wrapper (reference: src/blades/models/cifar10/cct.py:7-16)."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.model = cct_2_3x2_32(num_classes=num_classes)

    def forward(self, x):
        return self.model(x)


def create_model():
    return CCTNet(), nn.CrossEntropyLoss()


# ------------------------------------------------------------- checkpoints
def resize_positional_embedding(pe: torch.Tensor,
                                target_len: int) -> torch.Tensor:
    """Bilinearly rescale a [1, N, D] learnable positional embedding to a
    new sequence length (square token grids), so checkpoints trained at
    one image size load at another (the capability of the reference's
    pretrained-URL path, cctnets/utils/helpers.py resize_pos_embed)."""
    import math

    n, d = pe.shape[1], pe.shape[2]
    if n == target_len:
        return pe
    gs_old = int(math.sqrt(n))
    gs_new = int(math.sqrt(target_len))
    if gs_old * gs_old != n or gs_new * gs_new != target_len:
        raise ValueError(f"non-square token grids: {n} -> {target_len}")
    grid = pe.reshape(1, gs_old, gs_old, d).permute(0, 3, 1, 2)
    grid = F.interpolate(grid, size=(gs_new, gs_new), mode="bilinear",
                         align_corners=False)
    return grid.permute(0, 2, 3, 1).reshape(1, target_len, d)


def load_pretrained(model: "CCT", path: str, strict: bool = True) -> "CCT":
    """Load a LOCAL state-dict file into a CCT-family model, resizing the
    positional embedding and dropping a mismatched classifier head.

    This environment has no egress, so the reference's
    load_state_dict_from_url path (cctnets/cct.py:106-117) maps to
    loading an already-downloaded file; the adaptation semantics
    (pos-embed resize, head reset on class-count change) are preserved.
    """
    sd = torch.load(path, map_location="cpu", weights_only=True)
    own = model.state_dict()
    pe_key = "positional_emb"
    if pe_key in sd and pe_key in own and sd[pe_key].shape != own[pe_key].shape:
        sd[pe_key] = resize_positional_embedding(sd[pe_key],
                                                 own[pe_key].shape[1])
    for key in ("fc.weight", "fc.bias"):
        if key in sd and key in own and sd[key].shape != own[key].shape:
            sd[key] = own[key]  # class count changed: keep the fresh head
    model.load_state_dict(sd, strict=strict)
    return model
