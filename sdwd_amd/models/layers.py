"""Parameter-holding wrappers around the fused ops (ops/__init__.py)."""
from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class FusedGroupNorm(nn.Module):
    """GroupNorm with an optionally fused SiLU (one HIP kernel on GPU)."""

    def __init__(self, channels: int, groups: int = 32, silu: bool = False,
                 eps: float = 1e-5):
        super().__init__()
        self.groups = groups
        self.silu = silu
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.group_norm_silu(
            x, self.weight, self.bias, self.groups, self.eps, self.silu
        )


class FusedLayerNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class GEGLU(nn.Module):
    def __init__(self, dim_in: int, dim_out: int):
        super().__init__()
        self.proj = nn.Linear(dim_in, dim_out * 2)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.geglu(self.proj(x))
