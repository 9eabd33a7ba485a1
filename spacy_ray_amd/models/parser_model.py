"""Transition-parser scoring model (parser + NER share it).

MI355X-first split of spaCy's parser_model.pyx:
  * per batch: ONE GEMM precomputes lower(tok2vec) for every token ->
    [T, nF, H*P] (hipBLASLt on GPU), plus a learned pad row per feature slot;
  * per transition step: fused gather(nF rows)+sum+bias+maxout(P=2) kernel
    (ops.parser_step_score) then the small upper GEMM -> action scores,
    while the C++ transition system advances states on the CPU.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from spacy_ray_amd.ops import api as ops
from .layers import glorot_uniform_


class TransitionModel(nn.Module):
    PIECES = 2  # maxout pieces in the hidden layer (spaCy default)

    def __init__(self, width: int, hidden_width: int = 64, nF: int = 13):
        super().__init__()
        self.width = width
        self.hidden_width = hidden_width
        self.nF = nF
        HP = hidden_width * self.PIECES
        self.lower_W = nn.Parameter(glorot_uniform_(torch.empty(nF * HP, width)))
        self.lower_b = nn.Parameter(torch.zeros(HP))
        self.pad = nn.Parameter(torch.randn(nF, HP) * 0.05)
        self.upper: Optional[nn.Linear] = None
        self.n_actions: Optional[int] = None

    def initialize_output(self, n_actions: int) -> None:
        self.n_actions = n_actions
        # zero-init upper (spaCy initializes the output layer to zeros so
        # early training is driven by the oracle, not random scores)
        self.upper = nn.Linear(self.hidden_width, n_actions)
        nn.init.zeros_(self.upper.weight)
        nn.init.zeros_(self.upper.bias)

    def precompute(self, tok2vec: torch.Tensor) -> torch.Tensor:
        """tok2vec [T, W] -> [T+1, nF, H*P]; row T is the learned pad."""
        T = tok2vec.shape[0]
        HP = self.hidden_width * self.PIECES
        pre = torch.nn.functional.linear(tok2vec, self.lower_W).view(T, self.nF, HP)
        return torch.cat([pre, self.pad.unsqueeze(0).to(pre.dtype)], dim=0)

    def score(self, precomputed: torch.Tensor, feats: torch.Tensor) -> torch.Tensor:
        """feats [S, nF] int (missing already remapped to row T) -> [S, nA]."""
        hidden = ops.parser_step_score(precomputed, feats, self.lower_b)
        return self.upper(hidden)  # maxout is the nonlinearity; no extra relu
