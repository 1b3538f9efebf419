"""ViT-L/16 image classifier for the heterogeneous batch (BASELINE
config 4).

Standard pre-LN ViT: patch embed (conv), cls token, learned positions,
bidirectional attention.  Sequence length 197 is not a multiple of 64;
the flash wrapper zero-pads to 256 and masks the padded keys in-kernel
(ragged-T support), so ViT rides the fused MFMA attention path too.
Classification loss is plain CE over 1000 classes.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from saturn_amd.ops.functional import FusedLayerNorm, full_attention


@dataclass
class ViTConfig:
    img_size: int = 224
    patch: int = 16
    n_embd: int = 1024
    n_head: int = 16
    n_layer: int = 24
    n_classes: int = 1000


class ViTBlock(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.ln_1 = FusedLayerNorm(cfg.n_embd)
        self.qkv = nn.Linear(cfg.n_embd, 3 * cfg.n_embd)
        self.proj = nn.Linear(cfg.n_embd, cfg.n_embd)
        self.ln_2 = FusedLayerNorm(cfg.n_embd)
        self.mlp = nn.Sequential(
            nn.Linear(cfg.n_embd, 4 * cfg.n_embd),
            nn.GELU(approximate="tanh"),
            nn.Linear(4 * cfg.n_embd, cfg.n_embd),
        )
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head

    def forward(self, x):
        B, T, E = x.shape
        h = self.ln_1(x)
        q, k, v = self.qkv(h).split(E, dim=-1)
        q = q.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        k = k.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        v = v.view(B, T, self.n_head, self.head_dim).transpose(1, 2)
        o = full_attention(q, k, v).transpose(1, 2).reshape(B, T, E)
        x = x + self.proj(o)
        x = x + self.mlp(self.ln_2(x))
        return x


class ViTForImageClassification(nn.Module):
    def __init__(self, cfg: ViTConfig):
        super().__init__()
        self.cfg = cfg
        n_patches = (cfg.img_size // cfg.patch) ** 2
        self.patch_embed = nn.Conv2d(
            3, cfg.n_embd, kernel_size=cfg.patch, stride=cfg.patch
        )
        self.cls_token = nn.Parameter(torch.zeros(1, 1, cfg.n_embd))
        self.pos_embed = nn.Parameter(
            torch.zeros(1, n_patches + 1, cfg.n_embd)
        )
        self.h = nn.ModuleList(ViTBlock(cfg) for _ in range(cfg.n_layer))
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        self.head = nn.Linear(cfg.n_embd, cfg.n_classes)
        nn.init.normal_(self.pos_embed, std=0.02)
        nn.init.normal_(self.cls_token, std=0.02)

    def forward(self, pixels):
        B = pixels.shape[0]
        x = self.patch_embed(pixels).flatten(2).transpose(1, 2)
        cls = self.cls_token.expand(B, -1, -1).to(x.dtype)
        x = torch.cat([cls, x], dim=1) + self.pos_embed.to(x.dtype)
        for block in self.h:
            x = block(x)
        return self.head(self.ln_f(x[:, 0]))


def vit_loss(logits, labels):
    return torch.nn.functional.cross_entropy(logits.float(), labels)


def get_vit_model(kwargs=None):
    kwargs = kwargs or {}
    cfg = ViTConfig(
        n_layer=kwargs.get("n_layer", 24),
        n_embd=kwargs.get("n_embd", 1024),
        n_head=kwargs.get("n_head", 16),
        img_size=kwargs.get("img_size", 224),
    )
    torch.manual_seed(kwargs.get("seed", 0))
    return ViTForImageClassification(cfg)


class SyntheticImages(torch.utils.data.Dataset):
    def __init__(self, n, img_size=224, n_classes=1000, seed=1):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(n, 3, img_size, img_size, generator=g)
        self.y = torch.randint(0, n_classes, (n,), generator=g)

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def make_image_dataloader(batch_size=32, img_size=224, n_batches=16):
    def get_dataloader():
        return torch.utils.data.DataLoader(
            SyntheticImages(batch_size * n_batches, img_size),
            batch_size=batch_size,
            shuffle=False,
        )

    return get_dataloader
