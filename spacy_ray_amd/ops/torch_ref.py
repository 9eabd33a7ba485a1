"""Pure-torch reference implementations of the hot ops.

These define the semantics that the hand-written gfx950 HIP kernels in
ops/kernels/ must reproduce (kernel list: SURVEY.md §2.5; numerics tests
compare HIP output against these in fp32).  On CPU they ARE the compute
path; on GPU they serve as the test reference and are NOT used in
production (ops/api.py fails loudly if the HIP extension is missing on a
GPU device).
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch


def doc_boundary_masks(lengths: torch.Tensor, total: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """(is_start, is_end) bool masks of shape [total] from doc lengths."""
    device = lengths.device
    starts = torch.zeros(total, dtype=torch.bool, device=device)
    ends = torch.zeros(total, dtype=torch.bool, device=device)
    offs = torch.cumsum(lengths, 0)
    starts[offs - lengths] = True
    ends[offs - 1] = True
    return starts, ends


def seq2col(X: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    """Window-1 column concat with doc boundaries: out[i] = [prev, self, next]
    where prev/next are zero across doc boundaries.  [T, W] -> [T, 3W].
    (Contract of thinc seq2col)"""
    T, W = X.shape
    is_start, is_end = doc_boundary_masks(lengths, T)
    prev = torch.cat([X.new_zeros(1, W), X[:-1]], dim=0)
    prev = prev.masked_fill(is_start.unsqueeze(1), 0)
    nxt = torch.cat([X[1:], X.new_zeros(1, W)], dim=0)
    nxt = nxt.masked_fill(is_end.unsqueeze(1), 0)
    return torch.cat([prev, X, nxt], dim=1)


def seq2col_backward(dY: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    T, W3 = dY.shape
    W = W3 // 3
    is_start, is_end = doc_boundary_masks(lengths, T)
    d_prev, d_self, d_next = dY[:, :W], dY[:, W : 2 * W], dY[:, 2 * W :]
    dX = d_self.clone()
    # prev-slot of token i+1 came from token i (unless i+1 is a doc start)
    contrib = d_prev.masked_fill(is_start.unsqueeze(1), 0)
    dX[:-1] += contrib[1:]
    contrib = d_next.masked_fill(is_end.unsqueeze(1), 0)
    dX[1:] += contrib[:-1]
    return dX


def maxout(X: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Pieces-major maxout: [..., P, W] -> ([..., W], argmax [..., W] uint8).

    Pieces-major (vs thinc's pieces-last) so that on GPU each piece is a
    contiguous W-row — coalesced loads and vectorized max in the HIP kernel."""
    best, which = X.max(dim=-2)
    return best, which.to(torch.uint8)


def maxout_backward(dY: torch.Tensor, which: torch.Tensor, P: int) -> torch.Tensor:
    shape = list(dY.shape)
    shape.insert(-1, P)
    dX = dY.new_zeros(shape)
    dX.scatter_(-2, which.long().unsqueeze(-2), dY.unsqueeze(-2))
    return dX


def hashembed_rows_cpu(ids_u64: np.ndarray, seed: int, nrows: int) -> np.ndarray:
    from spacy_ray_amd import _srx_cpu

    return _srx_cpu.hashembed_rows(ids_u64.reshape(-1), seed, nrows)


def hashembed_forward(table: torch.Tensor, rows: torch.Tensor) -> torch.Tensor:
    """table [R, W], rows [T, 4] int -> [T, W]: sum of the 4 hashed rows
    (Thinc HashEmbed contract)."""
    return table[rows.long()].sum(dim=1)


def hashembed_backward(dY: torch.Tensor, rows: torch.Tensor, nrows: int) -> torch.Tensor:
    dT = dY.new_zeros(nrows, dY.shape[1])
    flat_rows = rows.long().reshape(-1)
    dT.index_add_(0, flat_rows, dY.repeat_interleave(4, dim=0))
    return dT


def layernorm(X: torch.Tensor, g: torch.Tensor, b: torch.Tensor, eps: float = 1e-5):
    mu = X.mean(dim=-1, keepdim=True)
    var = X.var(dim=-1, unbiased=False, keepdim=True)
    xhat = (X - mu) / torch.sqrt(var + eps)
    return xhat * g + b


def softmax_ce(scores: torch.Tensor, target: torch.Tensor, mask: Optional[torch.Tensor] = None):
    """Fused softmax + cross-entropy grad: returns (loss_sum, d_scores) where
    d_scores = softmax(scores) - target.  `mask` (bool, same shape) marks
    valid logits; invalid get -inf before softmax and zero grad."""
    s = scores.float()
    if mask is not None:
        s = s.masked_fill(~mask, float("-inf"))
    logp = torch.log_softmax(s, dim=-1)
    probs = logp.exp()
    loss = -(target * logp.masked_fill(target == 0, 0)).sum()
    d = probs - target
    if mask is not None:
        d = d.masked_fill(~mask, 0)
    return loss, d


def parser_step_score(
    precomputed: torch.Tensor,  # [T+1, nF, P*H] pieces-major (row T = pad)
    feats: torch.Tensor,        # [S, nF] int64 indices into [0, T]; T = missing
    bias: torch.Tensor,         # [P*H]
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-step parser state scorer input: gather nF precomputed rows per
    state, sum, + bias, maxout over P (pieces-major).  Returns
    (hidden [S, H], which [S, H]).
    (Contract of spaCy parser_model.pyx precompute_hiddens)"""
    S, nF = feats.shape
    HP = precomputed.shape[-1]
    slot = torch.arange(nF, device=feats.device).unsqueeze(0)
    summed = precomputed[feats.long(), slot].sum(dim=1) + bias  # [S, P*H]
    P = 2
    H = HP // P
    hidden, which = maxout(summed.view(S, P, H))
    return hidden, which


def reduce_sum_ragged(X: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    seg = torch.repeat_interleave(
        torch.arange(lengths.shape[0], device=X.device), lengths
    )
    out = X.new_zeros(lengths.shape[0], X.shape[1])
    out.index_add_(0, seg, X)
    return out


def reduce_mean_ragged(X: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
    s = reduce_sum_ragged(X, lengths)
    return s / lengths.clamp(min=1).unsqueeze(1).to(s.dtype)


def reduce_max_ragged(X: torch.Tensor, lengths: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    n_docs = lengths.shape[0]
    out = X.new_full((n_docs, X.shape[1]), float("-inf"))
    which = torch.zeros(n_docs, X.shape[1], dtype=torch.long, device=X.device)
    off = 0
    for i, n in enumerate(lengths.tolist()):
        if n:
            vals, idx = X[off : off + n].max(dim=0)
            out[i] = vals
            which[i] = idx + off
        off += n
    return out, which


# ---- activations (Thinc's elementwise surface: mish/swish/gelu/
# clipped_linear — SURVEY.md §2.2 N1); fp32 torch compositions used as the
# CPU path and the GPU numerics reference.
def act_forward(X: torch.Tensor, op: int, slope: float = 1.0,
                offset: float = 0.0, lo: float = float("-inf"),
                hi: float = float("inf")) -> torch.Tensor:
    if op == 0:  # mish
        return X * torch.tanh(torch.nn.functional.softplus(X))
    if op == 1:  # swish / silu
        return X * torch.sigmoid(X)
    if op == 2:  # gelu (erf form, matching thinc's gaussian gelu)
        return 0.5 * X * (1.0 + torch.erf(X * 0.7071067811865475))
    return torch.clamp(slope * X + offset, min=lo, max=hi)


def act_backward(dY: torch.Tensor, X: torch.Tensor, op: int,
                 slope: float = 1.0, offset: float = 0.0,
                 lo: float = float("-inf"),
                 hi: float = float("inf")) -> torch.Tensor:
    if op == 0:
        sp = torch.nn.functional.softplus(X)
        t = torch.tanh(sp)
        sig = torch.sigmoid(X)
        return dY * (t + X * sig * (1 - t * t))
    if op == 1:
        sig = torch.sigmoid(X)
        return dY * (sig * (1 + X * (1 - sig)))
    if op == 2:
        phi = 0.5 * (1.0 + torch.erf(X * 0.7071067811865475))
        pdf = 0.3989422804014327 * torch.exp(-0.5 * X * X)
        return dY * (phi + X * pdf)
    pre = slope * X + offset
    return dY * torch.where((pre > lo) & (pre < hi),
                            torch.full_like(X, slope),
                            torch.zeros_like(X))
