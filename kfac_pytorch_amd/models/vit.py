"""Vision Transformer for the K-FAC model zoo (beyond the reference's
CNN-only ImageNet list -- every parameterized layer is a ``Linear`` or
an ungrouped ``Conv2d`` patchifier, so the whole model is K-FAC-hooked
with the existing machinery; attention projections are separate
Linear modules exactly like the BERT shape model, models/bert.py)."""

from __future__ import annotations

import torch
import torch.nn as nn

__all__ = ["VisionTransformer", "vit_tiny", "vit_small"]


class MHSA(nn.Module):
    def __init__(self, dim: int, heads: int):
        super().__init__()
        self.heads = heads
        self.dk = dim // heads
        self.q = nn.Linear(dim, dim)
        self.k = nn.Linear(dim, dim)
        self.v = nn.Linear(dim, dim)
        self.o = nn.Linear(dim, dim)

    def forward(self, x):
        B, N, D = x.shape
        h, dk = self.heads, self.dk

        def split(t):
            return t.view(B, N, h, dk).transpose(1, 2)

        q, k, v = split(self.q(x)), split(self.k(x)), split(self.v(x))
        att = torch.softmax(q @ k.transpose(-2, -1) / dk ** 0.5, dim=-1)
        y = (att @ v).transpose(1, 2).reshape(B, N, D)
        return self.o(y)


class Block(nn.Module):
    def __init__(self, dim: int, heads: int, mlp_ratio: int = 4):
        super().__init__()
        self.n1 = nn.LayerNorm(dim)
        self.attn = MHSA(dim, heads)
        self.n2 = nn.LayerNorm(dim)
        self.fc1 = nn.Linear(dim, dim * mlp_ratio)
        self.fc2 = nn.Linear(dim * mlp_ratio, dim)

    def forward(self, x):
        x = x + self.attn(self.n1(x))
        return x + self.fc2(torch.nn.functional.gelu(
            self.fc1(self.n2(x))))


class VisionTransformer(nn.Module):
    def __init__(self, image_size: int = 224, patch: int = 16,
                 dim: int = 192, depth: int = 12, heads: int = 3,
                 num_classes: int = 1000):
        super().__init__()
        n = (image_size // patch) ** 2
        self.patchify = nn.Conv2d(3, dim, patch, stride=patch)
        self.cls = nn.Parameter(torch.zeros(1, 1, dim))
        self.pos = nn.Parameter(torch.zeros(1, n + 1, dim) * 0.0)
        nn.init.trunc_normal_(self.pos, std=0.02)
        nn.init.trunc_normal_(self.cls, std=0.02)
        self.blocks = nn.ModuleList(
            [Block(dim, heads) for _ in range(depth)])
        self.norm = nn.LayerNorm(dim)
        self.head = nn.Linear(dim, num_classes)

    def forward(self, x):
        x = self.patchify(x).flatten(2).transpose(1, 2)
        x = torch.cat([self.cls.expand(x.size(0), -1, -1), x], dim=1)
        x = x + self.pos
        for b in self.blocks:
            x = b(x)
        return self.head(self.norm(x)[:, 0])


def vit_tiny(num_classes: int = 1000, image_size: int = 224):
    return VisionTransformer(image_size=image_size, dim=192, depth=12,
                             heads=3, num_classes=num_classes)


def vit_small(num_classes: int = 1000, image_size: int = 224):
    return VisionTransformer(image_size=image_size, dim=384, depth=12,
                             heads=6, num_classes=num_classes)
