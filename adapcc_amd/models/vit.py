"""ViT for the DDP workload (reference: models/vit/train_vit.py used
vit-pytorch's ViT-base on synthetic data; self-contained here)."""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.fused import FusedLayerNorm


@dataclass
class ViTConfig:
    image_size: int = 224
    patch_size: int = 16
    num_classes: int = 1000
    dim: int = 768
    depth: int = 12
    heads: int = 12
    mlp_dim: int = 3072

    @classmethod
    def base(cls) -> "ViTConfig":
        return cls()

    @classmethod
    def tiny(cls) -> "ViTConfig":
        return cls(image_size=64, patch_size=8, num_classes=10, dim=128,
                   depth=2, heads=4, mlp_dim=256)


class EncoderBlock(nn.Module):
    def __init__(self, dim: int, heads: int, mlp_dim: int):
        super().__init__()
        self.ln1 = FusedLayerNorm(dim)
        self.qkv = nn.Linear(dim, 3 * dim)
        self.proj = nn.Linear(dim, dim)
        self.heads = heads
        self.ln2 = FusedLayerNorm(dim)
        self.mlp = nn.Sequential(
            nn.Linear(dim, mlp_dim), nn.GELU(), nn.Linear(mlp_dim, dim)
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, T, C = x.shape
        h = self.ln1(x)
        q, k, v = self.qkv(h).split(C, dim=2)
        hd = C // self.heads
        q = q.view(B, T, self.heads, hd).transpose(1, 2)
        k = k.view(B, T, self.heads, hd).transpose(1, 2)
        v = v.view(B, T, self.heads, hd).transpose(1, 2)
        a = F.scaled_dot_product_attention(q, k, v)
        a = a.transpose(1, 2).contiguous().view(B, T, C)
        x = x + self.proj(a)
        x = x + self.mlp(self.ln2(x))
        return x


class ViT(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.cfg = cfg
        n_patches = (cfg.image_size // cfg.patch_size) ** 2
        self.patch = nn.Conv2d(3, cfg.dim, kernel_size=cfg.patch_size,
                               stride=cfg.patch_size)
        self.cls_token = nn.Parameter(torch.zeros(1, 1, cfg.dim))
        self.pos = nn.Parameter(torch.zeros(1, n_patches + 1, cfg.dim))
        self.blocks = nn.ModuleList(
            EncoderBlock(cfg.dim, cfg.heads, cfg.mlp_dim)
            for _ in range(cfg.depth)
        )
        self.ln = FusedLayerNorm(cfg.dim)
        self.head = nn.Linear(cfg.dim, cfg.num_classes)
        nn.init.trunc_normal_(self.pos, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)

    def forward(self, images: torch.Tensor) -> torch.Tensor:
        x = self.patch(images).flatten(2).transpose(1, 2)  # [B, P, D]
        cls = self.cls_token.expand(x.shape[0], -1, -1)
        x = torch.cat([cls, x], dim=1) + self.pos
        for blk in self.blocks:
            x = blk(x)
        return self.head(self.ln(x[:, 0]))
