"""2-layer MLP — the CPU/gloo plumbing smoke model (contract config 1)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class MLP(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        m = cfg.model
        self.fc1 = nn.Linear(m.in_dim if m.in_dim else cfg.data.vector_dim, m.hidden_dim)
        self.fc2 = nn.Linear(m.hidden_dim, m.out_dim)

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        h = F.relu(self.fc1(x))
        out = self.fc2(h)
        return F.mse_loss(out.float(), y.float())
