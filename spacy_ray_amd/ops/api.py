"""Op dispatch: gfx950 HIP kernels on GPU, torch reference on CPU.

Every op is a torch.autograd.Function so the whole pipeline trains through
plain autograd.  On a CUDA(ROCm) device the HIP extension `_srx_hip` is
REQUIRED — a missing extension raises instead of silently falling back to
eager torch (the silent-fallback failure mode called out in the build
contract).  Set SRX_ALLOW_TORCH_FALLBACK=1 only for bring-up/debugging.
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from . import torch_ref as ref

_HIP = None
_HIP_TRIED = False

FIXED_SCALE = 16777216.0  # 2^24 (mirrors srx_common.hip.h SRX_FIXED_SCALE)


def deterministic() -> bool:
    """SRX_DETERMINISTIC=1: every atomic float accumulation runs in
    fixed-point int64 (bit-identical across runs) and sorts are stable —
    closes SURVEY §5.2 / VERDICT r1 item 7.  Costs ~10-20% on the scatter
    backwards."""
    return os.environ.get("SRX_DETERMINISTIC") == "1"


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
            hip = hip_ext()
            T = dY.shape[0]
            det = deterministic()
            if T >= 4096 or det:
                # Zipf-hot rows serialize plain atomics (1.5 ms/call at
                # T=128k measured) — sort by destination row + chunked
                # segmented reduction instead (SURVEY.md §7 hard-part 2).
                # Deterministic mode: stable sort + int64 fixed-point sink.
                dst = rows.reshape(-1)
                order = torch.argsort(dst, stable=True) if det else torch.argsort(dst)
                dst_sorted = dst[order].contiguous().int()
                src = (order // 4).int()
                acc_dt = torch.int64 if det else torch.float32
                dT32 = torch.zeros(ctx.nrows, dY.shape[1], dtype=acc_dt,
                                   device=dY.device)
                hip.seg_scatter_add(dst_sorted, src, dY.contiguous(), dT32)
                if det:
                    dT32 = dT32.to(torch.float32) / FIXED_SCALE
                dT = dT32.to(dY.dtype)
            else:
                dT = hip.hashembed_bwd(dY.contiguous(), rows, ctx.nrows)
        else:
            dT = ref.hashembed_backward(dY, rows, ctx.nrows)
        return dT, None, None


def hashembed(table: torch.Tensor, ids: torch.Tensor, seed: int) -> torch.Tensor:
    return _HashEmbed.apply(table, ids, seed)


class _MultiHashEmbed(torch.autograd.Function):
    """All 4 attr tables in one autograd node producing the concatenated
    [T, 4W] matrix: forward writes each table's output straight into its
    column block (no torch.cat — a 0.77 GB copy per step at 1M words);
    backward runs the sorted segmented reduction per table on COLUMN VIEWS
    of dX (strided seg_scatter source — no .contiguous() copies)."""

    @staticmethod
    def forward(ctx, ids4, seeds, *tables):
        hip = hip_ext()
        T = ids4.shape[0]
        W = tables[0].shape[1]
        X = tables[0].new_empty(T, W * len(tables))
        rows_all = []
        for i, table in enumerate(tables):
            _, rows = hip.hashembed_fwd(table, ids4[:, i].contiguous(),
                                        int(seeds[i]), X, i * W)
            rows_all.append(rows)
        ctx.save_for_backward(*rows_all)
        ctx.nrows = [t.shape[0] for t in tables]
        ctx.W = W
        return X

    @staticmethod
    def backward(ctx, dX):
        hip = hip_ext()
        rows_all = ctx.saved_tensors
        W = ctx.W
        det = deterministic()
        grads = []
        for i, rows in enumerate(rows_all):
            dY = dX[:, i * W : (i + 1) * W]  # strided view, no copy
            dst = rows.reshape(-1)
            order = torch.argsort(dst, stable=True) if det else torch.argsort(dst)
            dst_sorted = dst[order].contiguous().int()
            src = (order // 4).int()
            acc_dt = torch.int64 if det else torch.float32
            dT32 = torch.zeros(ctx.nrows[i], W, dtype=acc_dt, device=dX.device)
            hip.seg_scatter_add(dst_sorted, src, dY, dT32)
            if det:
                dT32 = dT32.to(torch.float32) / FIXED_SCALE
            grads.append(dT32.to(dX.dtype))
        return (None, None, *grads)


def multi_hashembed(ids4: torch.Tensor, seeds, tables) -> torch.Tensor:
    """GPU fast path for MultiHashEmbed; CPU falls back to per-table
    hashembed + cat."""
    if ids4.is_cuda and _want_hip(tables[0]):
        return _MultiHashEmbed.apply(ids4, list(seeds), *tables)
    outs = [hashembed(t, ids4[:, i].contiguous(), int(seeds[i]))
            for i, t in enumerate(tables)]
    return torch.cat(outs, dim=1)


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
            dX, dg, db = hip_ext().layernorm_bwd(dY.contiguous(), X, g, mu, rstd,
                                                 deterministic())
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
    """Step scorer whose backward ACCUMULATES dPre instead of materializing
    a fresh [T+1, nF, HP] gradient per transition step (the per-step
    allocation + autograd summation was the dominant host cost — see the
    two-phase scheme in pipes).  `precomputed` is passed detached; only
    `bias` is differentiable here.

    GPU: backward computes the compact dSummed [S, HP] (no atomics) and
    defers the scatter — finish_task sorts ALL steps' (token,f)
    destinations once and runs the chunked segmented reduction.
    CPU / fallback: scatter directly into the fp32 accumulator."""

    @staticmethod
    def forward(ctx, precomputed, feats, bias, dPre32, entries):
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
        # side accumulators (mutated across steps) — stashed on ctx directly
        # so autograd's saved-tensor version check doesn't trip.
        ctx.dPre32 = dPre32
        ctx.entries = entries
        ctx.HP = HP
        return hidden

    @staticmethod
    def backward(ctx, dHidden):
        feats, which = ctx.saved_tensors
        HP = ctx.HP
        P = 2
        if _want_hip(dHidden):
            hip = hip_ext()
            dSummed = hip.maxout_bwd(dHidden.contiguous(), which, P).view(-1, HP)
            dBias = dSummed.float().sum(dim=0)
            if ctx.entries is not None:
                ctx.entries.append((feats, dSummed))
            else:
                dPre32 = ctx.dPre32
                hip.parser_step_bwd_into(dHidden.contiguous(), feats, which, dPre32)
        else:
            S = dHidden.shape[0]
            nF = feats.shape[1]
            dSummed = ref.maxout_backward(dHidden, which, P).reshape(S, HP)
            dBias = dSummed.sum(dim=0)
            flat = ctx.dPre32.view(-1, HP)
            idx = (feats.long() * nF + torch.arange(nF, device=feats.device)).reshape(-1)
            flat.index_add_(0, idx, dSummed.float().repeat_interleave(nF, dim=0))
        return None, None, dBias.to(dHidden.dtype), None, None


def parser_step_score_accum(precomputed_detached, feats, bias, dPre32, entries=None):
    return _ParserStepScoreAccum.apply(precomputed_detached, feats, bias, dPre32, entries)


def mm_dw_chunked(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a [K, M], b [K, N] -> a^T @ b, for weight-gradient GEMMs with tiny
    M x N and huge K: hipBLASLt's tile grid is (M/64)*(N/64) workgroups —
    e.g. [1M,288]^T @ [1M,384] runs ~30 workgroups on 256 CUs (~2 ms
    measured).  Splitting K into 32 batched GEMMs + one fp32 sum fills the
    chip; partials are rounded to bf16 once each, summed in fp32."""
    K, M = a.shape
    N = b.shape[1]
    if (not a.is_cuda) or K < (1 << 18) or (M // 64 + 1) * (N // 64 + 1) >= 512:
        return a.t().mm(b)
    n = 32
    ck = K // n
    main = torch.bmm(a[: ck * n].view(n, ck, M).transpose(1, 2),
                     b[: ck * n].view(n, ck, N))
    out = main.sum(0, dtype=torch.float32)
    if ck * n < K:
        out += a[ck * n :].t().mm(b[ck * n :]).float()
    return out.to(a.dtype)


class _LinearCDW(torch.autograd.Function):
    """F.linear whose backward computes dW via mm_dw_chunked (the M*N-small,
    K-huge weight-grad GEMM) and db with fp32 accumulation."""

    @staticmethod
    def forward(ctx, X, W, b):
        ctx.save_for_backward(X, W)
        ctx.has_bias = b is not None
        return torch.nn.functional.linear(X, W, b)

    @staticmethod
    def backward(ctx, dY):
        X, W = ctx.saved_tensors
        dY = dY.contiguous()
        dW = mm_dw_chunked(dY, X)
        dX = dY.mm(W)
        db = dY.sum(0, dtype=torch.float32).to(dY.dtype) if ctx.has_bias else None
        return dX, dW, db


def linear_cdw(X, W, b=None):
    if X.is_cuda and X.dim() == 2:
        return _LinearCDW.apply(X, W, b)
    return torch.nn.functional.linear(X, W, b)


class _TransitionLoopLoss(torch.autograd.Function):
    """Batched loss + phase-1 backward for the C++-owned transition loop
    (srx_steploop.hip): the loop returns arenas over ALL transition steps;
    forward runs ONE fused CE kernel (softmax-over-valid vs uniform-min-cost
    target, spaCy's parser loss contract), backward is 3 large GEMMs + one
    maxout scatter + one dPre scatter — replacing ~121 per-step autograd
    nodes (each with 2 tiny GEMMs + reductions) per pipe per batch
    (VERDICT r1 items 1 & 4).

    Inputs needing grad: pre [T+1,nF,HP] (the grad-connected precompute),
    lower_b [HP], upperW [A,H], upperB [A].  The arena tensors are data.
    Returns the SUMMED loss (caller divides by n_states, so the upstream
    grad carries the normalization)."""

    @staticmethod
    def forward(ctx, pre, lower_b, upperW, upperB, scores, gold, valid,
                feats, which, hidden, doc_off, doc_lens, cap_mult, maxlen,
                doc_total):
        hip = hip_ext()
        loss_count, dScores, colsum = hip.transition_ce(scores, gold, valid,
                                                        deterministic())
        ctx.save_for_backward(dScores, colsum, feats, which, hidden, upperW)
        ctx.pre_shape = tuple(pre.shape)
        ctx.pre_dtype = pre.dtype
        ctx.doc_layout = (doc_off, doc_lens, cap_mult, maxlen, doc_total)
        return loss_count[0]

    @staticmethod
    def backward(ctx, g):
        dScores, colsum, feats, which, hidden, upperW = ctx.saved_tensors
        hip = hip_ext()
        T1, nF, HP = ctx.pre_shape
        dS = dScores * g.to(dScores.dtype)
        dUpperW = mm_dw_chunked(dS, hidden)
        # colsum came fused out of transition_ce (unscaled)
        dUpperB = (colsum * g.float()).to(dS.dtype)
        dHidden = dS.mm(upperW)
        dSummed = hip.maxout_bwd(dHidden.contiguous(), which, 2).view(-1, HP)
        doc_off, doc_lens, cap_mult, maxlen, doc_total = ctx.doc_layout
        # Doc-major atomic-free scatter: measured SLOWER than the direct
        # packed-bf16 atomic kernel on MI355X (12.8-23.6 ms vs 5.3 at 1.9M
        # rows — the per-(doc,slot) LDS accumulation re-reads dSummed up to
        # nF times and thrashes L2); kept opt-in for long-doc workloads
        # where per-destination contention would bite the atomics.
        if (doc_off is not None and not deterministic()
                and os.environ.get("SRX_DPRE_DOCMAJOR", "0") == "1"
                and HP % 64 == 0 and 0 < maxlen and maxlen * HP * 4 <= 65536
                and hasattr(hip, "dpre_scatter_docmajor")):
            # GPU-state-machine arenas are doc-major: (token, slot) rows are
            # exclusive to one (doc, slot) stream, so the scatter runs
            # atomic-free (LDS per-doc accumulation + plain stores) into an
            # UNINITIALIZED buffer — the direct kernel below is bound by
            # packed-atomic op rate, not bandwidth.
            dPre = torch.empty(T1, nF, HP, dtype=ctx.pre_dtype,
                               device=dS.device)
            # the kernel writes every (token, slot) row of the DOCS; batch
            # PADDING tokens [doc_total, T1-1) get no contributions — zero
            # them explicitly (the pad row T1-1 is set from dPad below)
            if doc_total < T1 - 1:
                dPre[doc_total:T1 - 1].zero_()
            dBias32, dPad32 = hip.dpre_scatter_docmajor(
                dSummed, feats, dPre, doc_off, doc_lens, T1 - 1, cap_mult,
                maxlen)
        else:
            # dPre accumulates directly in the compute dtype (bf16: packed
            # atomics, no fp32 buffer + convert); the Zipf-hot pad row and
            # the bias column-sum are register-accumulated into fp32 side
            # buffers.  Deterministic mode: int64 fixed-point end to end.
            acc_dt = torch.int64 if deterministic() else ctx.pre_dtype
            dPre = torch.zeros(T1, nF, HP, dtype=acc_dt, device=dS.device)
            dBias32, dPad32 = hip.dpre_scatter(dSummed, feats, dPre, T1 - 1)
            if acc_dt == torch.int64:
                dPre = (dPre.to(torch.float32) / FIXED_SCALE).to(ctx.pre_dtype)
        dPre[T1 - 1] = dPad32.to(dPre.dtype)
        return (dPre, dBias32.to(dS.dtype), dUpperW, dUpperB,
                None, None, None, None, None, None, None, None, None, None,
                None)


def transition_loop_loss(pre, lower_b, upperW, upperB, scores, gold, valid,
                         feats, which, hidden, doc_off=None, doc_lens=None,
                         cap_mult=1, maxlen=0, doc_total=0):
    return _TransitionLoopLoss.apply(pre, lower_b, upperW, upperB, scores,
                                     gold, valid, feats, which, hidden,
                                     doc_off, doc_lens, cap_mult, maxlen,
                                     doc_total)


class _InjectGrad(torch.autograd.Function):
    """Hand a precomputed gradient to a tensor through autograd: forward is
    a zero scalar (the value is never used — losses are logged separately);
    backward returns `grad_buf` cast to the tensor's dtype.  Replaces the
    (pre.float() * dPre32).sum() surrogate, which cost three full passes
    over the [T+1,nF,HP] fp32 buffer per pipe per step."""

    @staticmethod
    def forward(ctx, x, grad_buf):
        ctx.grad_buf = grad_buf
        ctx.dtype = x.dtype
        return x.new_zeros(())

    @staticmethod
    def backward(ctx, g):
        # g is the upstream scalar (1.0 from a plain loss sum); never read it
        # on the host — a float(g) here would device-sync every step
        return ctx.grad_buf.to(ctx.dtype) * g, None


def inject_grad(x: torch.Tensor, grad_buf: torch.Tensor) -> torch.Tensor:
    return _InjectGrad.apply(x, grad_buf)


def parser_scatter_entries(entries, dPre32) -> None:
    """Batched dPre scatter: sort all steps' (token*nF+f) destinations once,
    then one chunked segmented reduction into the fp32 accumulator."""
    if not entries:
        return
    hip = hip_ext()
    device = dPre32.device
    nF = entries[0][0].shape[1]
    HP = dPre32.shape[-1]
    feats_all = torch.cat([f for f, _ in entries], dim=0)          # [SS, nF] i64
    dS_all = torch.cat([d for _, d in entries], dim=0).contiguous()  # [SS, HP]
    SS = feats_all.shape[0]
    slot = torch.arange(nF, device=device)
    dest = (feats_all * nF + slot).reshape(-1).int()
    src = torch.arange(SS, device=device, dtype=torch.int32).repeat_interleave(nF)
    order = torch.argsort(dest)
    hip.seg_scatter_add(dest[order].contiguous(), src[order].contiguous(),
                        dS_all, dPre32.view(-1, HP))


# ---------------------------------------------------- fused MWE layer (MFMA)
class _MWELayer(torch.autograd.Function):
    """One MaxoutWindowEncoder block as a single hand-written MFMA kernel:
    Y = X + LN(maxout_3(seq2col(X) @ W^T + bias)).  Forward is the fused
    gfx950 kernel (srx_mwe.hip.h); backward composes existing kernels:
    layernorm_bwd on the saved maxout output -> maxout scatter -> the two
    GEMM backwards (hipBLASLt) with seq2col recomputed -> seq2col_bwd."""

    @staticmethod
    def forward(ctx, X, weight, bias, g, b, starts, ends, dropmask, eps):
        Y, Mout, which, mu, rstd = hip_ext().mwe_layer_fwd(
            X.contiguous(), weight, bias, g, b, starts, ends, dropmask, eps
        )
        ctx.save_for_backward(X, weight, g, which, Mout, mu, rstd, starts, ends,
                              *( [dropmask] if dropmask is not None else [] ))
        ctx.has_mask = dropmask is not None
        return Y

    @staticmethod
    def backward(ctx, dY):
        if ctx.has_mask:
            X, weight, g, which, Mout, mu, rstd, starts, ends, dropmask = ctx.saved_tensors
        else:
            X, weight, g, which, Mout, mu, rstd, starts, ends = ctx.saved_tensors
            dropmask = None
        hip = hip_ext()
        dY = dY.contiguous()
        # stage 1 in ONE kernel: dropout mask x LN backward x maxout scatter
        # + the dg/db/dbias column sums (was 3 kernels + 2 reduce passes)
        dPre, dg32, db32, dbias32 = hip.mwe_bwd_stage1(
            dY, dropmask, Mout, g, mu, rstd, which, deterministic())
        # GEMM backwards (pre = X3 @ W^T)
        X3 = hip.seq2col_fwd(X, starts, ends)
        dW = mm_dw_chunked(dPre, X3)
        dX3 = dPre.mm(weight)
        # seq2col backward with the residual dY folded in
        dX = hip.seq2col_bwd(dX3.contiguous(), starts, ends, dY)
        return (dX, dW, dbias32.to(dY.dtype), dg32.to(dY.dtype),
                db32.to(dY.dtype), None, None, None, None)


def mwe_layer(X, weight, bias, g, b, starts, ends, dropmask=None, eps: float = 1e-5):
    return _MWELayer.apply(X, weight, bias, g, b, starts, ends, dropmask, eps)


def mwe_layer_available(X: torch.Tensor, width: int, pieces: int) -> bool:
    return (
        X.is_cuda
        and X.dtype == torch.bfloat16
        and pieces == 3
        and width in (96, 128)
        and X.shape[0] % 64 == 0
        and hip_ext() is not None
    )


class _AddLayerNorm(torch.autograd.Function):
    """LN(X + R) with the residual add fused into the layernorm kernels'
    data passes (saves a full elementwise pass each way on the roberta
    blocks' `LayerNorm(hidden + input)` pattern).  The sum S is
    materialized once for the backward (the LN backward needs its input);
    dX = dR = LN-input grad."""

    @staticmethod
    def forward(ctx, X, R, g, b, eps: float):
        S = X + R
        if _want_hip(S):
            Y, mu, rstd = hip_ext().layernorm_fwd(S.contiguous(), g, b, eps)
            ctx.save_for_backward(S, g, mu, rstd)
            ctx.eps = eps
            return Y
        mu = S.mean(dim=-1, keepdim=True)
        var = S.var(dim=-1, unbiased=False, keepdim=True)
        rstd = torch.rsqrt(var + eps)
        ctx.save_for_backward(S, g, mu, rstd)
        return (S - mu) * rstd * g + b

    @staticmethod
    def backward(ctx, dY):
        S, g, mu, rstd = ctx.saved_tensors
        if _want_hip(dY):
            dS, dg, db = hip_ext().layernorm_bwd(dY.contiguous(), S, g, mu, rstd,
                                                 deterministic())
            return dS, dS, dg, db, None
        xhat = (S - mu) * rstd
        dg = (dY * xhat).sum(dim=tuple(range(dY.dim() - 1)))
        db = dY.sum(dim=tuple(range(dY.dim() - 1)))
        dxhat = dY * g
        dS = rstd * (
            dxhat
            - dxhat.mean(dim=-1, keepdim=True)
            - xhat * (dxhat * xhat).mean(dim=-1, keepdim=True)
        )
        return dS, dS, dg, db, None


def add_layernorm(X, R, g, b, eps: float = 1e-5):
    return _AddLayerNorm.apply(X, R, g, b, eps)


# ------------------------------------------------------------ dropout mask
_dropout_calls = 0


def dropout_mask_like(X: torch.Tensor, p: float) -> torch.Tensor:
    """Scaled keep-mask in X's dtype via the Philox kernel (SURVEY §2.5
    dropout_mask).  Reproducible under torch.manual_seed: the Philox
    (seed, offset) derive from torch's initial seed + a call counter."""
    global _dropout_calls
    _dropout_calls += 1
    if X.is_cuda and _want_hip(X):
        return hip_ext().dropout_mask(X, p, torch.initial_seed() & 0x7FFFFFFF,
                                      _dropout_calls)
    keep = 1.0 - p
    return (torch.rand_like(X, dtype=torch.float32) < keep).to(X.dtype) / keep


# ----------------------------------------------------------- softmax + CE
class _SoftmaxCE(torch.autograd.Function):
    """Fused softmax + cross-entropy (SURVEY.md §2.5 softmax_ce_fwd/bwd):
    forward returns the summed NLL over rows with gold >= 0; backward hands
    (softmax - onehot) * grad_out to the scores."""

    @staticmethod
    def forward(ctx, scores: torch.Tensor, gold: torch.Tensor):
        loss_count, dScores = hip_ext().softmax_ce(scores.contiguous(), gold)
        ctx.save_for_backward(dScores)
        return loss_count[0]

    @staticmethod
    def backward(ctx, grad_out):
        (dScores,) = ctx.saved_tensors
        return dScores * grad_out.to(dScores.dtype), None


def softmax_ce_loss(scores: torch.Tensor, gold: torch.Tensor) -> torch.Tensor:
    """Summed cross-entropy; rows with gold < 0 are ignored.  GPU: fused
    kernel.  CPU: torch cross_entropy."""
    if scores.is_cuda and _want_hip(scores):
        return _SoftmaxCE.apply(scores, gold)
    return torch.nn.functional.cross_entropy(
        scores.float(), gold, ignore_index=-1, reduction="sum"
    )


# ------------------------------------------------------- ragged reductions
def _offsets_and_docof(lengths: torch.Tensor):
    l = lengths.to(torch.int32)
    offsets = torch.zeros(l.shape[0] + 1, dtype=torch.int32, device=l.device)
    offsets[1:] = torch.cumsum(l, 0)
    doc_of = torch.repeat_interleave(
        torch.arange(l.shape[0], device=l.device, dtype=torch.int32), lengths.long()
    )
    return offsets, doc_of


class _ReduceRagged(torch.autograd.Function):
    @staticmethod
    def forward(ctx, X, lengths, mode: int):
        if _want_hip(X):
            offsets, doc_of = _offsets_and_docof(lengths)
            out, argmax = hip_ext().reduce_ragged(X.contiguous(), offsets, mode)
            ctx.save_for_backward(lengths, offsets, doc_of, argmax)
        else:
            if mode == 0:
                out = ref.reduce_sum_ragged(X, lengths)
            elif mode == 1:
                out = ref.reduce_mean_ragged(X, lengths)
            else:
                out, argmax_l = ref.reduce_max_ragged(X, lengths)
                ctx.argmax_cpu = argmax_l
            ctx.save_for_backward(lengths)
        ctx.mode = mode
        ctx.T = X.shape[0]
        return out

    @staticmethod
    def backward(ctx, dY):
        mode, T = ctx.mode, ctx.T
        if _want_hip(dY):
            lengths, offsets, doc_of, argmax = ctx.saved_tensors
            if mode == 2:
                dX = hip_ext().reduce_max_bwd(dY.contiguous(), argmax, T)
            else:
                dX = hip_ext().reduce_ragged_bwd(dY.contiguous(), doc_of, offsets, T, mode)
            return dX, None, None
        (lengths,) = ctx.saved_tensors
        if mode == 2:
            dX = dY.new_zeros(T, dY.shape[1])
            which = ctx.argmax_cpu
            dX.scatter_(0, which, dY)
            return dX, None, None
        seg = torch.repeat_interleave(
            torch.arange(lengths.shape[0], device=dY.device), lengths.long()
        )
        scale = dY
        if mode == 1:
            scale = dY / lengths.clamp(min=1).unsqueeze(1).to(dY.dtype)
        return scale[seg], None, None


def reduce_sum_ragged(X, lengths):
    return _ReduceRagged.apply(X, lengths, 0)


def reduce_mean_ragged(X, lengths):
    return _ReduceRagged.apply(X, lengths, 1)


def reduce_max_ragged(X, lengths):
    return _ReduceRagged.apply(X, lengths, 2)


# ---- elementwise activations (Thinc kernel-surface parity: mish, swish,
# gelu, clipped_linear and its relu/hard_* family — SURVEY.md §2.2 N1).
ACT_MISH, ACT_SWISH, ACT_GELU, ACT_CLIPPED_LINEAR = 0, 1, 2, 3


class _Activation(torch.autograd.Function):
    @staticmethod
    def forward(ctx, X, op, slope, offset, lo, hi):
        ctx.save_for_backward(X)
        ctx.act = (op, slope, offset, lo, hi)
        if _want_hip(X):
            return hip_ext().act_fwd(X, op, slope, offset, lo, hi)
        return ref.act_forward(X, op, slope, offset, lo, hi)

    @staticmethod
    def backward(ctx, dY):
        (X,) = ctx.saved_tensors
        op, slope, offset, lo, hi = ctx.act
        if _want_hip(X):
            dX = hip_ext().act_bwd(dY, X, op, slope, offset, lo, hi)
        else:
            dX = ref.act_backward(dY, X, op, slope, offset, lo, hi)
        return dX, None, None, None, None, None


def _act(X, op, slope=1.0, offset=0.0, lo=float("-inf"), hi=float("inf")):
    return _Activation.apply(X, op, slope, offset, lo, hi)


def mish(X: torch.Tensor) -> torch.Tensor:
    return _act(X, ACT_MISH)


def swish(X: torch.Tensor) -> torch.Tensor:
    return _act(X, ACT_SWISH)


def gelu(X: torch.Tensor) -> torch.Tensor:
    return _act(X, ACT_GELU)


def clipped_linear(X, slope=1.0, offset=0.0, min_val=float("-inf"),
                   max_val=float("inf")):
    return _act(X, ACT_CLIPPED_LINEAR, slope, offset, min_val, max_val)


def relu(X: torch.Tensor) -> torch.Tensor:
    return clipped_linear(X, 1.0, 0.0, 0.0, float("inf"))


def hard_sigmoid(X: torch.Tensor) -> torch.Tensor:
    return clipped_linear(X, 0.2, 0.5, 0.0, 1.0)


def hard_tanh(X: torch.Tensor) -> torch.Tensor:
    return clipped_linear(X, 1.0, 0.0, -1.0, 1.0)


# ---- windowed attention: two hipBLASLt bmm GEMMs + the fused masked
# softmax(+dropout) kernels (srx_attn.hip.h).  Replaces SDPA inside the
# transformer's short windows: aotriton's flash backward measured 5.7x
# its forward at L<=96 and the eager math path costs ~6 elementwise
# kernels per call (profiles/trf262k_r2_kernel_stats.csv).
class _WindowAttention(torch.autograd.Function):
    """Two execution paths:
    * FUSED (bf16, D=64, L<=96): the flash-style MFMA kernels
      (attn_fused_fwd/bwd) — Q/K/V/dO tiles and the S x S probabilities
      stay in LDS, only lse is saved, the backward recomputes P.
    * bmm fallback (fp32, D!=64 or L>96): hipBLASLt bmm GEMMs + the
      fused masked-softmax kernels (S/P materialized in HBM)."""

    @staticmethod
    def forward(ctx, q, k, v, lens, scale, drop_p):
        hip = hip_ext()
        B, H, L, D = q.shape
        qf = q.reshape(B * H, L, D)
        kf = k.reshape(B * H, L, D)
        vf = v.reshape(B * H, L, D)
        seed = (int(torch.randint(0, 2**62, (1,), device="cpu").item())
                if drop_p > 0 else 0)
        fused = (L <= 96 and D == 64 and q.dtype == torch.bfloat16
                 and hasattr(hip, "attn_fused_fwd"))
        ctx.attn_meta = (H, scale, drop_p, seed, fused)
        if fused:
            O, lse = hip.attn_fused_fwd(qf, kf, vf, lens, H, scale, drop_p,
                                        seed)
            ctx.save_for_backward(qf, kf, vf, lens, lse)
            return O.view(B, H, L, D)
        S = torch.bmm(qf, kf.transpose(1, 2))
        P, lse = hip.attn_softmax_fwd(S, lens, H, scale, drop_p, seed)
        O = torch.bmm(P, vf)
        ctx.save_for_backward(S, lse, P, qf, kf, vf, lens)
        return O.view(B, H, L, D)

    @staticmethod
    def backward(ctx, dO):
        H, scale, drop_p, seed, fused = ctx.attn_meta
        hip = hip_ext()
        if fused:
            qf, kf, vf, lens, lse = ctx.saved_tensors
            BH, L, D = qf.shape
            dOf = dO.reshape(BH, L, D)
            if not dOf.is_contiguous():
                dOf = dOf.contiguous()
            dQ, dK, dV = hip.attn_fused_bwd(qf, kf, vf, dOf, lse, lens, H,
                                            scale, drop_p, seed)
            shp = dO.shape
            return (dQ.view(shp), dK.view(shp), dV.view(shp),
                    None, None, None)
        S, lse, P, qf, kf, vf, lens = ctx.saved_tensors
        BH, L, D = qf.shape
        dOf = dO.reshape(BH, L, D)
        if not dOf.is_contiguous():
            dOf = dOf.contiguous()
        dPt = torch.bmm(dOf, vf.transpose(1, 2))
        dS = hip.attn_softmax_bwd(S, lse, dPt, lens, H, scale, drop_p, seed)
        dQ = torch.bmm(dS, kf)
        dK = torch.bmm(dS.transpose(1, 2), qf)
        dV = torch.bmm(P.transpose(1, 2), dOf)
        shp = dO.shape
        return (dQ.view(shp), dK.view(shp), dV.view(shp), None, None, None)


def window_attention(q, k, v, lens, scale: float, drop_p: float = 0.0):
    """q/k/v: [B, H, L, D]; lens: [B] int32 valid prefix per window.
    Returns [B, H, L, D]."""
    return _WindowAttention.apply(q, k, v, lens, scale, drop_p)
