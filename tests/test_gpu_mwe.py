"""Fused MWE-layer MFMA kernel vs the composed reference (GPU)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu
need_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")
DEV = "cuda:0"


def _composed_ref(X32, Wt32, bias32, g32, b32, lengths, eps, mask32=None):
    from spacy_ray_amd.ops import torch_ref as ref

    X3 = ref.seq2col(X32, lengths)
    pre = X3 @ Wt32.t() + bias32
    T, W = X32.shape
    m, which = pre.view(T, 3, W).max(dim=1)
    ln = torch.nn.functional.layer_norm(m, (W,), g32, b32, eps)
    if mask32 is not None:
        ln = ln * mask32
    return X32 + ln, m


@need_gpu
@pytest.mark.parametrize("W", [96, 128])
@pytest.mark.parametrize("with_mask", [False, True])
def test_mwe_layer_fused_matches_composed(W, with_mask):
    from spacy_ray_amd import _srx_hip
    from spacy_ray_amd.ops.api import boundary_masks_u8, mwe_layer

    torch.manual_seed(0)
    lengths = torch.tensor([30, 2, 64, 17, 15], device=DEV)
    T = 128
    assert int(lengths.sum()) == T
    X = (torch.randn(T, W, device=DEV) * 0.5).to(torch.bfloat16)
    Wt = (torch.randn(3 * W, 3 * W, device=DEV) * (1.0 / np.sqrt(3 * W))).to(torch.bfloat16)
    bias = torch.randn(3 * W, device=DEV).to(torch.bfloat16) * 0.1
    g = (1 + 0.1 * torch.randn(W, device=DEV)).to(torch.bfloat16)
    b = (0.1 * torch.randn(W, device=DEV)).to(torch.bfloat16)
    starts, ends = boundary_masks_u8(lengths, T)
    mask = None
    if with_mask:
        mask = ((torch.rand(T, W, device=DEV) < 0.9).to(torch.bfloat16) / 0.9)

    # fused forward (no grad path first)
    Y, Mout, which, mu, rstd = _srx_hip.mwe_layer_fwd(
        X, Wt, bias, g, b, starts, ends, mask, 1e-5
    )
    Yr, Mr = _composed_ref(
        X.float(), Wt.float(), bias.float(), g.float(), b.float(),
        lengths, 1e-5, mask.float() if mask is not None else None,
    )
    assert torch.allclose(Mout.float(), Mr, atol=5e-2, rtol=5e-2), (
        (Mout.float() - Mr).abs().max()
    )
    assert torch.allclose(Y.float(), Yr, atol=6e-2, rtol=6e-2), (
        (Y.float() - Yr).abs().max()
    )

    # autograd through the fused op vs composed fp32 autograd
    Xa = X.clone().requires_grad_(True)
    Wa = Wt.clone().requires_grad_(True)
    ba = bias.clone().requires_grad_(True)
    ga = g.clone().requires_grad_(True)
    bb = b.clone().requires_grad_(True)
    Yf = mwe_layer(Xa, Wa, ba, ga, bb, starts, ends, mask, 1e-5)
    dY = torch.randn_like(Yf)
    Yf.backward(dY)

    X2 = X.float().requires_grad_(True)
    W2 = Wt.float().requires_grad_(True)
    b2 = bias.float().requires_grad_(True)
    g2 = g.float().requires_grad_(True)
    bb2 = b.float().requires_grad_(True)
    Y2, _ = _composed_ref(X2, W2, b2, g2, bb2, lengths, 1e-5,
                          mask.float() if mask is not None else None)
    Y2.backward(dY.float())

    for got, want, name in [
        (Xa.grad, X2.grad, "dX"),
        (Wa.grad, W2.grad, "dW"),
        (ba.grad, b2.grad, "dbias"),
        (ga.grad, g2.grad, "dg"),
        (bb.grad, bb2.grad, "db"),
    ]:
        assert torch.allclose(got.float(), want, atol=2e-1, rtol=1e-1), (
            name, (got.float() - want).abs().max(), want.abs().max()
        )


@need_gpu
def test_srx_embedding_backward_matches_torch():
    from spacy_ray_amd.models.transformer import _SrxEmbedding

    torch.manual_seed(3)
    R, W, N = 500, 64, 4000
    emb = _SrxEmbedding(R, W, padding_idx=1).to(DEV).to(torch.bfloat16)
    ids = torch.randint(0, R, (N,), device=DEV)
    ids[::5] = 1    # padding rows must get no grad
    ids[::3] = 7    # hot row
    Y = emb(ids)
    dY = torch.randn_like(Y)
    Y.backward(dY)
    ref_emb = torch.nn.Embedding(R, W, padding_idx=1).to(DEV)
    with torch.no_grad():
        ref_emb.weight.copy_(emb.weight.float())
    Y2 = ref_emb(ids)
    Y2.backward(dY.float())
    assert torch.allclose(emb.weight.grad.float(), ref_emb.weight.grad,
                          atol=2e-1, rtol=5e-2)
    assert emb.weight.grad[1].abs().max() == 0  # padding_idx zeroed


@need_gpu
def test_srx_layernorm_module_matches_torch():
    from spacy_ray_amd.models.transformer import _SrxLayerNorm

    torch.manual_seed(4)
    ln = _SrxLayerNorm(768, eps=1e-5).to(DEV).to(torch.bfloat16)
    X = torch.randn(257, 768, device=DEV, dtype=torch.bfloat16, requires_grad=True)
    Y = ln(X)
    Yr = torch.nn.functional.layer_norm(X.float(), (768,), ln.weight.float(),
                                        ln.bias.float(), 1e-5)
    assert torch.allclose(Y.float(), Yr, atol=5e-2, rtol=5e-2)
    Y.sum().backward()
    assert torch.isfinite(X.grad.float()).all()


@need_gpu
def test_mwe_used_in_encoder_forward():
    """The CNN encoder must actually route through the fused kernel on GPU
    (bf16, W=96, padded T)."""
    from spacy_ray_amd.models.tok2vec import MaxoutWindowEncoder
    from spacy_ray_amd.ops import api

    enc = MaxoutWindowEncoder(96, depth=2).to(DEV).to(torch.bfloat16)
    T = 128
    X = torch.randn(T, 96, device=DEV, dtype=torch.bfloat16)
    lengths = torch.tensor([64, 64], device=DEV)
    assert api.mwe_layer_available(X, 96, 3)
    Y = enc(X, lengths)
    assert Y.shape == (T, 96)
    assert torch.isfinite(Y.float()).all()
