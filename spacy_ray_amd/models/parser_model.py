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


class _PrecomputePad(torch.autograd.Function):
    """tok2vec [T,W] -> [T+1, nF, HP] with the learned pad as row T,
    WITHOUT torch.cat: the cat copied the whole [T, nF*HP] GEMM output
    (3.3 GB at 1M words) just to append one row — 1.8 ms/call x 2 pipes in
    the r2 profile.  The GEMM writes straight into rows [0, T) of the
    preallocated buffer; backward splits dOut into the three grads."""

    @staticmethod
    def forward(ctx, X, lower_W, pad):
        T = X.shape[0]
        nF, HP = pad.shape
        out = X.new_empty(T + 1, nF, HP)
        torch.mm(X, lower_W.t(), out=out[:T].view(T, nF * HP))
        out[T] = pad.to(out.dtype)
        ctx.save_for_backward(X, lower_W)
        return out

    @staticmethod
    def backward(ctx, dOut):
        from spacy_ray_amd.ops.api import mm_dw_chunked

        X, lower_W = ctx.saved_tensors
        T = X.shape[0]
        d2 = dOut[:T].reshape(T, -1)
        dX = d2.mm(lower_W)
        dW = mm_dw_chunked(d2, X)
        dPad = dOut[T]
        return dX, dW, dPad


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
        return _PrecomputePad.apply(tok2vec, self.lower_W, self.pad)

    def score(self, precomputed: torch.Tensor, feats: torch.Tensor) -> torch.Tensor:
        """feats [S, nF] int (missing already remapped to row T) -> [S, nA]."""
        hidden = ops.parser_step_score(precomputed, feats, self.lower_b)
        return self.upper(hidden)  # maxout is the nonlinearity; no extra relu
