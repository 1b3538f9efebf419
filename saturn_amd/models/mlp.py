"""Tiny MLP + synthetic regression data: CPU plumbing workload.

This is BASELINE.json config 1's model ("2-job MLP lr-sweep on CPU/gloo"),
and the model every no-GPU orchestration test uses.  Factories are
top-level functions (picklable across the gang spawn boundary).
"""

from __future__ import annotations

import torch
import torch.nn as nn


class MLP(nn.Sequential):
    def __init__(self, in_dim: int = 32, hidden: int = 64, out_dim: int = 8, depth: int = 2):
        layers = []
        d = in_dim
        for _ in range(depth):
            layers += [nn.Linear(d, hidden), nn.ReLU()]
            d = hidden
        layers.append(nn.Linear(d, out_dim))
        super().__init__(*layers)


def get_mlp_model(kwargs=None):
    kwargs = kwargs or {}
    torch.manual_seed(kwargs.get("seed", 0))
    return MLP(
        in_dim=kwargs.get("in_dim", 32),
        hidden=kwargs.get("hidden", 64),
        out_dim=kwargs.get("out_dim", 8),
        depth=kwargs.get("depth", 2),
    )


class SyntheticRegression(torch.utils.data.Dataset):
    def __init__(self, n: int = 64, in_dim: int = 32, out_dim: int = 8, seed: int = 1):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(n, in_dim, generator=g)
        w = torch.randn(in_dim, out_dim, generator=g)
        self.y = self.x @ w + 0.01 * torch.randn(n, out_dim, generator=g)

    def __len__(self) -> int:
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def get_mlp_dataloader():
    return torch.utils.data.DataLoader(
        SyntheticRegression(), batch_size=8, shuffle=False
    )


def mse_loss(output, target):
    return torch.nn.functional.mse_loss(output.float(), target.float())
