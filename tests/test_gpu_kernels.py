"""GPU kernel parity: each gfx950 HIP kernel vs the plain-torch fp32
reference in ops/torch_ref.py (same op, same inputs).  bf16 variants are
checked against the fp32 reference with bf16-scale tolerances."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from spacy_ray_amd import _srx_hip
from spacy_ray_amd.ops import torch_ref as ref

need_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")

DEV = "cuda:0"


def _tol(dtype):
    return dict(atol=2e-2, rtol=2e-2) if dtype == torch.bfloat16 else dict(atol=1e-5, rtol=1e-5)


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("W", [96, 100])
def test_seq2col_gpu(dtype, W):
    torch.manual_seed(0)
    lengths = torch.tensor([5, 1, 9, 3], device=DEV)
    T = int(lengths.sum())
    X = torch.randn(T, W, device=DEV, dtype=dtype)
    from spacy_ray_amd.ops.api import boundary_masks_u8

    starts, ends = boundary_masks_u8(lengths, T)
    Y = _srx_hip.seq2col_fwd(X, starts, ends)
    Yr = ref.seq2col(X.float(), lengths)
    assert torch.allclose(Y.float(), Yr, **_tol(dtype))
    dY = torch.randn(T, 3 * W, device=DEV, dtype=dtype)
    dX = _srx_hip.seq2col_bwd(dY, starts, ends)
    dXr = ref.seq2col_backward(dY.float(), lengths)
    assert torch.allclose(dX.float(), dXr, **_tol(dtype))


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(17, 3, 96), (5, 2, 128), (9, 3, 50)])
def test_maxout_gpu(dtype, shape):
    torch.manual_seed(1)
    X = torch.randn(*shape, device=DEV, dtype=dtype)
    Y, which = _srx_hip.maxout_fwd(X)
    Yr, whichr = ref.maxout(X.float())
    assert torch.allclose(Y.float(), Yr, **_tol(dtype))
    assert (which.cpu() == whichr.cpu()).float().mean() > 0.99  # fp ties may differ
    dY = torch.randn_like(Y)
    dX = _srx_hip.maxout_bwd(dY, which, shape[-2])
    dXr = ref.maxout_backward(dY.float(), which, shape[-2])
    assert torch.allclose(dX.float(), dXr, **_tol(dtype))


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("W", [96, 300])
def test_layernorm_gpu(dtype, W):
    torch.manual_seed(2)
    N = 257
    X = torch.randn(N, W, device=DEV, dtype=dtype)
    g = torch.randn(W, device=DEV, dtype=dtype)
    b = torch.randn(W, device=DEV, dtype=dtype)
    Y, mu, rstd = _srx_hip.layernorm_fwd(X, g, b, 1e-5)
    Yr = torch.nn.functional.layer_norm(X.float(), (W,), g.float(), b.float(), 1e-5)
    assert torch.allclose(Y.float(), Yr, **_tol(dtype))
    dY = torch.randn_like(X)
    dX, dg, db = _srx_hip.layernorm_bwd(dY, X, g, mu, rstd)
    X2 = X.float().requires_grad_(True)
    g2 = g.float().requires_grad_(True)
    b2 = b.float().requires_grad_(True)
    Y2 = torch.nn.functional.layer_norm(X2, (W,), g2, b2, 1e-5)
    Y2.backward(dY.float())
    tol = dict(atol=5e-2, rtol=5e-2) if dtype == torch.bfloat16 else dict(atol=1e-3, rtol=1e-3)
    assert torch.allclose(dX.float(), X2.grad, **tol)
    assert torch.allclose(dg.float(), g2.grad, **tol)
    assert torch.allclose(db.float(), b2.grad, **tol)


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_hashembed_gpu_matches_cpu_hash(dtype):
    """GPU murmur must be bit-identical to the C++ CPU murmur."""
    torch.manual_seed(3)
    R, W, T = 500, 96, 1000
    table = torch.randn(R, W, device=DEV, dtype=dtype)
    ids_np = (np.arange(T, dtype=np.uint64) * np.uint64(0x9E3779B97F4A7C15)) + np.uint64(13)
    ids = torch.from_numpy(ids_np.view(np.int64)).to(DEV)
    Y, rows = _srx_hip.hashembed_fwd(table, ids, 7)
    rows_cpu = ref.hashembed_rows_cpu(ids_np, 7, R)
    assert (rows.cpu().numpy() == rows_cpu).all(), "GPU murmur != CPU murmur"
    Yr = ref.hashembed_forward(table.float(), torch.from_numpy(rows_cpu).long().to(DEV))
    assert torch.allclose(Y.float(), Yr, **_tol(dtype))
    dY = torch.randn(T, W, device=DEV, dtype=dtype)
    dT = _srx_hip.hashembed_bwd(dY, rows, R)
    dTr = ref.hashembed_backward(dY.float(), rows.long(), R)
    tol = dict(atol=1e-1, rtol=5e-2) if dtype == torch.bfloat16 else dict(atol=1e-3, rtol=1e-4)
    assert torch.allclose(dT.float(), dTr, **tol)


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("H", [64, 80])
def test_parser_step_gpu(dtype, H):
    torch.manual_seed(4)
    T, nF, S = 50, 13, 37
    P = 2
    pre = torch.randn(T + 1, nF, P * H, device=DEV, dtype=dtype)
    bias = torch.randn(P * H, device=DEV, dtype=dtype)
    feats = torch.randint(0, T + 1, (S, nF), device=DEV)
    hidden, which = _srx_hip.parser_step_fwd(pre, feats, bias)
    hr, whichr = ref.parser_step_score(pre.float(), feats, bias.float())
    assert torch.allclose(hidden.float(), hr, **_tol(dtype))
    dH = torch.randn(S, H, device=DEV, dtype=dtype)
    dPre, dBias = _srx_hip.parser_step_bwd(dH, feats, which, T + 1, nF, 2 * H)
    # reference backward via autograd
    pre2 = pre.float().requires_grad_(True)
    bias2 = bias.float().requires_grad_(True)
    slot = torch.arange(nF, device=DEV).unsqueeze(0)
    summed = pre2[feats.long(), slot].sum(dim=1) + bias2
    h2 = summed.view(S, P, H).max(dim=-2).values
    h2.backward(dH.float())
    tol = dict(atol=5e-2, rtol=5e-2) if dtype == torch.bfloat16 else dict(atol=1e-3, rtol=1e-4)
    assert torch.allclose(dPre.float(), pre2.grad, **tol)
    assert torch.allclose(dBias.float(), bias2.grad, **tol)


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_action_select_gpu(dtype):
    torch.manual_seed(6)
    S, A = 123, 82
    scores = torch.randn(S, A, device=DEV, dtype=dtype)
    is_gold = (torch.rand(S, A, device=DEV) < 0.05).to(torch.uint8)
    valid = ((torch.rand(S, A, device=DEV) < 0.4).to(torch.uint8) | is_gold)
    is_gold[0] = 0  # state with no gold -> falls back to valid
    acts = _srx_hip.action_select(scores, is_gold, valid).cpu().numpy()
    s_np = scores.float().cpu().numpy()
    g_np = is_gold.cpu().numpy()
    v_np = valid.cpu().numpy()
    choose = np.where(g_np > 0, s_np, -1e30)
    fallback = np.where(v_np > 0, s_np, -1e30)
    any_gold = (g_np > 0).any(axis=1, keepdims=True)
    expect = np.where(any_gold, choose, fallback).argmax(axis=1)
    assert (acts == expect).all()


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_seg_scatter_add_gpu(dtype):
    """Sorted chunked segmented reduction == plain index_add (incl. a
    Zipf-hot destination)."""
    torch.manual_seed(9)
    M, Ns, No, W = 5000, 700, 50, 96
    dst = torch.randint(0, No, (M,), device=DEV, dtype=torch.int32)
    dst[: M // 3] = 7  # hot row
    src = torch.randint(0, Ns, (M,), device=DEV, dtype=torch.int32)
    SRC = torch.randn(Ns, W, device=DEV, dtype=dtype)
    order = torch.argsort(dst)
    OUT = torch.zeros(No, W, device=DEV)
    _srx_hip.seg_scatter_add(dst[order].contiguous(), src[order].contiguous(), SRC, OUT)
    expect = torch.zeros(No, W, device=DEV)
    expect.index_add_(0, dst.long(), SRC[src.long()].float())
    tol = dict(atol=2e-1, rtol=1e-2) if dtype == torch.bfloat16 else dict(atol=1e-3, rtol=1e-4)
    assert torch.allclose(OUT, expect, **tol), (OUT - expect).abs().max()


@need_gpu
def test_batched_parser_scatter_matches_atomic_path():
    """The deferred sort+segmented dPre scatter == the per-step atomic
    scatter kernel over several steps."""
    from spacy_ray_amd.ops.api import parser_scatter_entries

    torch.manual_seed(10)
    T, nF, H = 300, 13, 64
    HP = 2 * H
    dtype = torch.bfloat16
    dPre_atomic = torch.zeros(T + 1, nF, HP, device=DEV)
    dPre_batched = torch.zeros(T + 1, nF, HP, device=DEV)
    entries = []
    for step in range(5):
        S = 40 + step * 7
        feats = torch.randint(0, T + 1, (S, nF), device=DEV)
        dHidden = torch.randn(S, H, device=DEV, dtype=dtype)
        which = (torch.rand(S, H, device=DEV) < 0.5).to(torch.uint8)
        _srx_hip.parser_step_bwd_into(dHidden, feats, which, dPre_atomic)
        dSummed = _srx_hip.maxout_bwd(dHidden, which, 2).view(S, HP)
        entries.append((feats, dSummed))
    parser_scatter_entries(entries, dPre_batched)
    assert torch.allclose(dPre_atomic, dPre_batched, atol=5e-2, rtol=1e-2), (
        (dPre_atomic - dPre_batched).abs().max()
    )


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_softmax_ce_gpu(dtype):
    torch.manual_seed(7)
    N, C = 513, 50
    scores = torch.randn(N, C, device=DEV, dtype=dtype)
    gold = torch.randint(0, C, (N,), device=DEV)
    gold[::7] = -1
    loss_count, d = _srx_hip.softmax_ce(scores, gold)
    s32 = scores.float()
    lr = torch.nn.functional.cross_entropy(s32, gold, ignore_index=-1, reduction="sum")
    tol = dict(atol=5e-2, rtol=1e-2) if dtype == torch.bfloat16 else dict(atol=1e-2, rtol=1e-4)
    assert torch.allclose(loss_count[0], lr, **tol)
    assert int(loss_count[1]) == int((gold >= 0).sum())
    s2 = s32.requires_grad_(True)
    l2 = torch.nn.functional.cross_entropy(s2, gold, ignore_index=-1, reduction="sum")
    l2.backward()
    assert torch.allclose(d.float(), s2.grad, **_tol(dtype))


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("mode", [0, 1, 2])
def test_reduce_ragged_gpu(dtype, mode):
    torch.manual_seed(8)
    lengths = torch.tensor([3, 1, 7, 5], device=DEV)
    T, W = 16, 96
    X = torch.randn(T, W, device=DEV, dtype=dtype)
    from spacy_ray_amd.ops.api import _offsets_and_docof

    offsets, doc_of = _offsets_and_docof(lengths)
    out, argmax = _srx_hip.reduce_ragged(X, offsets, mode)
    if mode == 0:
        expect = ref.reduce_sum_ragged(X.float(), lengths)
    elif mode == 1:
        expect = ref.reduce_mean_ragged(X.float(), lengths)
    else:
        expect, which = ref.reduce_max_ragged(X.float(), lengths)
    assert torch.allclose(out.float(), expect, **_tol(dtype))
    dY = torch.randn(4, W, device=DEV, dtype=dtype)
    if mode == 2:
        dX = _srx_hip.reduce_max_bwd(dY, argmax, T)
        dXr = torch.zeros(T, W, device=DEV)
        dXr.scatter_(0, which, dY.float())
        assert torch.allclose(dX.float(), dXr, **_tol(dtype))
    else:
        dX = _srx_hip.reduce_ragged_bwd(dY, doc_of, offsets, T, mode)
        seg = torch.repeat_interleave(torch.arange(4, device=DEV), lengths)
        scale = dY.float() / lengths.clamp(min=1).unsqueeze(1).float() if mode == 1 else dY.float()
        assert torch.allclose(dX.float(), scale[seg], **_tol(dtype))


@need_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_adam_step_gpu(dtype):
    torch.manual_seed(5)
    n = 10_000
    grad = torch.randn(n, device=DEV, dtype=dtype)
    master = torch.randn(n, device=DEV, dtype=torch.float32)
    m = torch.zeros(n, device=DEV, dtype=torch.float32)
    v = torch.zeros(n, device=DEV, dtype=torch.float32)
    param_out = torch.zeros(n, device=DEV, dtype=dtype)
    lr, b1, b2, eps, wd = 0.01, 0.9, 0.999, 1e-8, 0.01
    ref_p = master.clone()
    g32 = grad.float() * 0.5
    ref_p.mul_(1 - lr * wd)
    mr = (1 - b1) * g32
    vr = (1 - b2) * g32 * g32
    denom = (vr / (1 - b2)).sqrt() + eps
    ref_p -= (lr / (1 - b1)) * (mr / denom)
    scale = torch.tensor([0.5], device=DEV)
    _srx_hip.adam_step(grad, master, m, v, param_out, scale, lr, b1, b2, eps, wd,
                       1 - b1, 1 - b2)
    assert torch.allclose(master, ref_p, atol=1e-6)
    assert torch.allclose(m, mr, atol=1e-6)
    assert torch.allclose(v, vr, atol=1e-6)
    assert torch.allclose(param_out.float(), ref_p.float().to(dtype).float(), atol=1e-2)


@need_gpu
def test_dropout_mask_philox_statistics_and_reproducibility():
    """Philox dropout mask (SURVEY §2.5 dropout_mask): keep fraction ~= 1-p,
    kept entries scaled by 1/(1-p), same (seed, offset) -> identical mask."""
    from spacy_ray_amd.ops.api import hip_ext

    hip = hip_ext()
    X = torch.empty(1 << 20, device="cuda", dtype=torch.bfloat16)
    p = 0.3
    m1 = hip.dropout_mask(X, p, 1234, 7)
    m2 = hip.dropout_mask(X, p, 1234, 7)
    m3 = hip.dropout_mask(X, p, 1234, 8)
    assert torch.equal(m1, m2)
    assert not torch.equal(m1, m3)
    keep_frac = float((m1 > 0).float().mean())
    assert abs(keep_frac - (1 - p)) < 0.01, keep_frac
    kept = m1[m1 > 0].float()
    assert torch.allclose(kept, torch.full_like(kept, 1 / (1 - p)), atol=1e-2)
    # the mean of the mask is ~1 (unbiased dropout scaling)
    assert abs(float(m1.float().mean()) - 1.0) < 0.02


@need_gpu
def test_activation_kernels_match_fp32_ref():
    """act_fwd/act_bwd HIP kernels vs the fp32 torch reference (mish,
    swish, gelu, clipped_linear), bf16 and fp32, vector and tail paths."""
    from spacy_ray_amd.ops import api, torch_ref as ref

    hip = api.hip_ext()
    torch.manual_seed(4)
    cases = [(0, 1.0, 0.0, float("-inf"), float("inf")),
             (1, 1.0, 0.0, float("-inf"), float("inf")),
             (2, 1.0, 0.0, float("-inf"), float("inf")),
             (3, 0.2, 0.5, 0.0, 1.0)]
    for n in (4096, 4099):  # vector path and scalar-tail path
        for dt, tol in ((torch.float32, 1e-5), (torch.bfloat16, 2e-2)):
            X = torch.randn(n, device="cuda", dtype=dt)
            dY = torch.randn(n, device="cuda", dtype=dt)
            for op, slope, offset, lo, hi in cases:
                y = hip.act_fwd(X, op, slope, offset, lo, hi).float()
                y_ref = ref.act_forward(X.float(), op, slope, offset, lo, hi)
                assert torch.allclose(y, y_ref, atol=tol, rtol=tol), (op, dt)
                dx = hip.act_bwd(dY, X, op, slope, offset, lo, hi).float()
                dx_ref = ref.act_backward(dY.float(), X.float(), op, slope,
                                          offset, lo, hi)
                assert torch.allclose(dx, dx_ref, atol=tol, rtol=tol), (op, dt)
