"""Op dispatch: gfx950 HIP kernels on GPU, torch reference on CPU.

Every op is a torch.autograd.Function so the whole pipeline trains through
plain autograd.  On a CUDA(ROCm) device the HIP extension `_srx_hip` is
REQUIRED — a missing extension raises instead of silently falling back to
eager torch (the silent-fallback failure mode called out in the build
contract).  Set SRX_ALLOW_TORCH_FALLBACK=1 only for bring-up/debugging.
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from . import torch_ref as ref

_HIP = None
_HIP_TRIED = False


def hip_ext():
    global _HIP, _HIP_TRIED
    if not _HIP_TRIED:
        _HIP_TRIED = True
        try:
            from spacy_ray_amd import _srx_hip  # noqa: F401
            _HIP = _srx_hip
        except ImportError:
            _HIP = None
    return _HIP


def _want_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if hip_ext() is not None:
        return True
    if os.environ.get("SRX_ALLOW_TORCH_FALLBACK") == "1":
        return False
    raise RuntimeError(
        "spacy_ray_amd._srx_hip (gfx950 HIP kernels) is not built but an op "
        "ran on a GPU tensor. Build with `python setup.py build_ext --inplace` "
        "(or set SRX_ALLOW_TORCH_FALLBACK=1 for debugging only)."
    )


# --------------------------------------------------------------- seq2col
def boundary_masks_u8(lengths: torch.Tensor, total: int):
    """(is_start, is_end) uint8 device tensors — computed once per batch
    WITHOUT any device sync (no nonzero/masked_select)."""
    device = lengths.device
    starts = torch.zeros(total, dtype=torch.uint8, device=device)
    ends = torch.zeros(total, dtype=torch.uint8, device=device)
    l = lengths.long()
    offs = l.cumsum(0)
    one = torch.ones_like(l, dtype=torch.uint8)
    starts.scatter_(0, (offs - l).clamp_(0, max(total - 1, 0)), one)
    ends.scatter_(0, (offs - 1).clamp_(0, max(total - 1, 0)), one)
    return starts, ends


class _Seq2Col(torch.autograd.Function):
    @staticmethod
    def forward(ctx, X: torch.Tensor, lengths: torch.Tensor,
                starts: Optional[torch.Tensor], ends: Optional[torch.Tensor]) -> torch.Tensor:
        if starts is None:
            starts, ends = boundary_masks_u8(lengths, X.shape[0])
        ctx.save_for_backward(lengths, starts, ends)
        if _want_hip(X):
            return hip_ext().seq2col_fwd(X.contiguous(), starts, ends)
        return ref.seq2col(X, lengths)

    @staticmethod
    def backward(ctx, dY: torch.Tensor):
        lengths, starts, ends = ctx.saved_tensors
        if _want_hip(dY):
            return hip_ext().seq2col_bwd(dY.contiguous(), starts, ends), None, None, None
        return ref.seq2col_backward(dY, lengths), None, None, None


def seq2col(X: torch.Tensor, lengths: torch.Tensor,
            starts: Optional[torch.Tensor] = None,
            ends: Optional[torch.Tensor] = None) -> torch.Tensor:
    return _Seq2Col.apply(X, lengths, starts, ends)


# ---------------------------------------------------------------- maxout
class _Maxout(torch.autograd.Function):
    """Pieces-major maxout: [..., P, W] -> [..., W] (see ref.maxout)."""

    @staticmethod
    def forward(ctx, X: torch.Tensor) -> torch.Tensor:
        if _want_hip(X):
            Y, which = hip_ext().maxout_fwd(X.contiguous())
        else:
            Y, which = ref.maxout(X)
        ctx.save_for_backward(which)
        ctx.P = X.shape[-2]
        return Y

    @staticmethod
    def backward(ctx, dY: torch.Tensor):
        (which,) = ctx.saved_tensors
        if _want_hip(dY):
            return hip_ext().maxout_bwd(dY.contiguous(), which, ctx.P)
        return ref.maxout_backward(dY, which, ctx.P)


def maxout(X: torch.Tensor) -> torch.Tensor:
    """[..., P, W] -> [..., W]"""
    return _Maxout.apply(X)


# ------------------------------------------------------------- hashembed
class _HashEmbed(torch.autograd.Function):
    """table [R, W], ids [T] int64(bit-cast uint64), seed -> [T, W].

    GPU path: fused murmur3-x64-128 hash + 4-row gather-sum HIP kernel.
    CPU path: rows via the C++ murmur, then torch gather (same bits)."""

    @staticmethod
    def forward(ctx, table: torch.Tensor, ids: torch.Tensor, seed: int) -> torch.Tensor:
        nrows = table.shape[0]
        if _want_hip(table):
            Y, rows = hip_ext().hashembed_fwd(table, ids.contiguous(), seed)
        else:
            rows_np = ref.hashembed_rows_cpu(
                ids.cpu().numpy().view("uint64"), seed, nrows
            )
            rows = torch.from_numpy(rows_np).to(ids.device)
            Y = ref.hashembed_forward(table, rows)
        ctx.save_for_backward(rows)
        ctx.nrows = nrows
        return Y

    @staticmethod
    def backward(ctx, dY: torch.Tensor):
        (rows,) = ctx.saved_tensors
        if _want_hip(dY):
            dT = hip_ext().hashembed_bwd(dY.contiguous(), rows, ctx.nrows)
        else:
            dT = ref.hashembed_backward(dY, rows, ctx.nrows)
        return dT, None, None


def hashembed(table: torch.Tensor, ids: torch.Tensor, seed: int) -> torch.Tensor:
    return _HashEmbed.apply(table, ids, seed)


# ------------------------------------------------------------- layernorm
class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, X, g, b, eps: float):
        if _want_hip(X):
            Y, mu, rstd = hip_ext().layernorm_fwd(X.contiguous(), g, b, eps)
            ctx.save_for_backward(X, g, mu, rstd)
            return Y
        # CPU: compose from torch (autograd-visible pieces not needed; manual)
        mu = X.mean(dim=-1, keepdim=True)
        var = X.var(dim=-1, unbiased=False, keepdim=True)
        rstd = torch.rsqrt(var + eps)
        ctx.save_for_backward(X, g, mu, rstd)
        return (X - mu) * rstd * g + b

    @staticmethod
    def backward(ctx, dY):
        X, g, mu, rstd = ctx.saved_tensors
        if _want_hip(dY):
            dX, dg, db = hip_ext().layernorm_bwd(dY.contiguous(), X, g, mu, rstd)
            return dX, dg, db, None
        xhat = (X - mu) * rstd
        dg = (dY * xhat).sum(dim=tuple(range(dY.dim() - 1)))
        db = dY.sum(dim=tuple(range(dY.dim() - 1)))
        dxhat = dY * g
        W = X.shape[-1]
        dX = rstd * (
            dxhat
            - dxhat.mean(dim=-1, keepdim=True)
            - xhat * (dxhat * xhat).mean(dim=-1, keepdim=True)
        )
        return dX, dg, db, None


def layernorm(X, g, b, eps: float = 1e-5):
    return _LayerNorm.apply(X, g, b, eps)


# ------------------------------------------------------ parser step score
class _ParserStepScore(torch.autograd.Function):
    """Gather nF precomputed rows per state, sum, +bias, maxout(P=2).

    precomputed [T+1, nF, H*P] (row T zero-pad for missing features),
    feats [S, nF] int32/int64 in [0, T].  Output hidden [S, H]."""

    @staticmethod
    def forward(ctx, precomputed, feats, bias):
        S, nF = feats.shape
        HP = precomputed.shape[-1]
        P = 2
        H = HP // P
        if _want_hip(precomputed):
            hidden, which = hip_ext().parser_step_fwd(precomputed, feats.contiguous(), bias)
        else:
            slot = torch.arange(nF, device=feats.device).unsqueeze(0)
            summed = precomputed[feats.long(), slot].sum(dim=1) + bias
            hidden, which = ref.maxout(summed.view(S, P, H))
        ctx.save_for_backward(feats, which)
        ctx.shape = (precomputed.shape[0], nF, HP)
        return hidden

    @staticmethod
    def backward(ctx, dHidden):
        feats, which = ctx.saved_tensors
        T1, nF, HP = ctx.shape
        P = 2
        if _want_hip(dHidden):
            dPre, dBias = hip_ext().parser_step_bwd(dHidden.contiguous(), feats, which, T1, nF, HP)
            return dPre, None, dBias
        dSummed = ref.maxout_backward(dHidden, which, P).reshape(dHidden.shape[0], HP)  # [S, P*H]
        dBias = dSummed.sum(dim=0)
        dPre = dHidden.new_zeros(T1, nF, HP)
        # scatter-add into the gathered rows
        flat = dPre.view(T1 * nF, HP)
        idx = (feats.long() * nF + torch.arange(nF, device=feats.device)).reshape(-1)
        flat.index_add_(0, idx, dSummed.repeat_interleave(nF, dim=0))
        return dPre, None, dBias


def parser_step_score(precomputed, feats, bias):
    return _ParserStepScore.apply(precomputed, feats, bias)


class _ParserStepScoreAccum(torch.autograd.Function):
    """Step scorer whose backward ACCUMULATES dPre into one persistent fp32
    buffer instead of materializing a fresh [T+1, nF, HP] gradient per
    transition step (the per-step allocation + autograd summation was the
    dominant host cost — see the two-phase scheme in pipes._step_loop).
    `precomputed` is passed detached; only `bias` is differentiable here."""

    @staticmethod
    def forward(ctx, precomputed, feats, bias, dPre32):
        S, nF = feats.shape
        HP = precomputed.shape[-1]
        P = 2
        H = HP // P
        if _want_hip(precomputed):
            hidden, which = hip_ext().parser_step_fwd(precomputed, feats.contiguous(), bias)
        else:
            slot = torch.arange(nF, device=feats.device).unsqueeze(0)
            summed = precomputed[feats.long(), slot].sum(dim=1) + bias
            hidden, which = ref.maxout(summed.view(S, P, H))
        ctx.save_for_backward(feats, which)
        # dPre32 is a side accumulator (mutated across steps) — stash it on
        # ctx directly so autograd's saved-tensor version check doesn't trip.
        ctx.dPre32 = dPre32
        ctx.HP = HP
        return hidden

    @staticmethod
    def backward(ctx, dHidden):
        feats, which = ctx.saved_tensors
        dPre32 = ctx.dPre32
        HP = ctx.HP
        P = 2
        if _want_hip(dHidden):
            dBias = hip_ext().parser_step_bwd_into(dHidden.contiguous(), feats, which, dPre32)
        else:
            S = dHidden.shape[0]
            nF = feats.shape[1]
            dSummed = ref.maxout_backward(dHidden, which, P).reshape(S, HP)
            dBias = dSummed.sum(dim=0)
            flat = dPre32.view(-1, HP)
            idx = (feats.long() * nF + torch.arange(nF, device=feats.device)).reshape(-1)
            flat.index_add_(0, idx, dSummed.float().repeat_interleave(nF, dim=0))
        return None, None, dBias.to(dHidden.dtype), None


def parser_step_score_accum(precomputed_detached, feats, bias, dPre32):
    return _ParserStepScoreAccum.apply(precomputed_detached, feats, bias, dPre32)


# ------------------------------------------------------- ragged reductions
def reduce_mean_ragged(X, lengths):
    if _want_hip(X):
        return hip_ext().reduce_mean_ragged(X.contiguous(), lengths)
    return ref.reduce_mean_ragged(X, lengths)
