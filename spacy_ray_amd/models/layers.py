"""Building-block layers (torch modules over the ops API)."""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from spacy_ray_amd.ops import api as ops


def glorot_uniform_(w: torch.Tensor) -> torch.Tensor:
    """Thinc-style Glorot/Xavier uniform init."""
    fan_in, fan_out = w.shape[-1], w.shape[0]
    limit = math.sqrt(6.0 / (fan_in + fan_out))
    with torch.no_grad():
        w.uniform_(-limit, limit)
    return w


class Maxout(nn.Module):
    """Linear (nI -> P*nO, pieces-major) followed by max over P pieces — the
    Thinc Maxout layer (SURVEY.md §2.5 maxout_fwd/bwd).  Weight rows are laid
    out pieces-major so each piece is a contiguous nO block in the GEMM
    output (coalesced maxout kernel)."""

    def __init__(self, nI: int, nO: int, pieces: int = 3, normalize: bool = False):
        super().__init__()
        self.nI, self.nO, self.pieces = nI, nO, pieces
        self.weight = nn.Parameter(glorot_uniform_(torch.empty(pieces * nO, nI)))
        self.bias = nn.Parameter(torch.zeros(pieces * nO))
        self.norm = LayerNorm(nO) if normalize else None

    def forward(self, X: torch.Tensor) -> torch.Tensor:
        Y = ops.linear_cdw(X, self.weight, self.bias)
        Y = ops.maxout(Y.view(*Y.shape[:-1], self.pieces, self.nO))
        if self.norm is not None:
            Y = self.norm(Y)
        return Y


class LayerNorm(nn.Module):
    def __init__(self, width: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(width))
        self.bias = nn.Parameter(torch.zeros(width))
        self.eps = eps

    def forward(self, X: torch.Tensor) -> torch.Tensor:
        return ops.layernorm(X, self.weight, self.bias, self.eps)


class Linear(nn.Module):
    def __init__(self, nI: int, nO: int, bias: bool = True):
        super().__init__()
        self.weight = nn.Parameter(glorot_uniform_(torch.empty(nO, nI)))
        self.bias = nn.Parameter(torch.zeros(nO)) if bias else None

    def forward(self, X: torch.Tensor) -> torch.Tensor:
        return torch.nn.functional.linear(X, self.weight, self.bias)
