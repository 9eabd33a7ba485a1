"""Op semantics + custom-backward correctness on CPU.

Custom autograd backwards (seq2col, maxout, hashembed, layernorm,
parser_step_score) are checked against autograd-composed references and
numerical expectations in fp64/fp32."""
import numpy as np
import pytest
import torch

from spacy_ray_amd.ops import api as ops
from spacy_ray_amd.ops import torch_ref as ref


def test_seq2col_boundaries():
    X = torch.arange(12, dtype=torch.float32).reshape(4, 3)
    lengths = torch.tensor([2, 2])
    Y = ref.seq2col(X, lengths)
    assert Y.shape == (4, 9)
    # token 0: prev = 0 (doc start), self = X0, next = X1
    assert (Y[0, :3] == 0).all() and (Y[0, 3:6] == X[0]).all() and (Y[0, 6:] == X[1]).all()
    # token 1: next = 0 (doc end)
    assert (Y[1, 6:] == 0).all()
    # token 2 (doc 2 start): prev = 0
    assert (Y[2, :3] == 0).all()


def test_seq2col_backward_matches_autograd():
    torch.manual_seed(0)
    X = torch.randn(11, 5, requires_grad=True)
    lengths = torch.tensor([4, 1, 6])
    Y = ops.seq2col(X, lengths)
    g = torch.randn_like(Y)
    Y.backward(g)
    X2 = X.detach().clone().requires_grad_(True)
    Y2 = ref.seq2col(X2, lengths)  # composed of autograd-visible torch ops
    Y2.backward(g)
    assert torch.allclose(X.grad, X2.grad, atol=1e-6)


def test_maxout_backward_matches_autograd():
    torch.manual_seed(1)
    X = torch.randn(7, 3, 4, requires_grad=True)  # pieces-major [T, P, W]
    Y = ops.maxout(X)
    assert Y.shape == (7, 4)
    g = torch.randn_like(Y)
    Y.backward(g)
    X2 = X.detach().clone().requires_grad_(True)
    Y2 = X2.max(dim=-2).values
    Y2.backward(g)
    assert torch.allclose(X.grad, X2.grad)


def test_hashembed_forward_backward():
    torch.manual_seed(2)
    table = torch.randn(50, 8, requires_grad=True)
    ids = torch.from_numpy(
        (np.arange(13, dtype=np.uint64) * np.uint64(0x9E3779B97F4A7C15)).view(np.int64)
    )
    Y = ops.hashembed(table, ids, seed=3)
    assert Y.shape == (13, 8)
    # forward equals manual 4-row gather-sum
    rows = ref.hashembed_rows_cpu(ids.numpy().view("uint64"), 3, 50)
    rows_t = torch.from_numpy(rows).long()
    expect = table[rows_t].sum(dim=1)
    assert torch.allclose(Y, expect)
    g = torch.randn_like(Y)
    Y.backward(g)
    t2 = table.detach().clone().requires_grad_(True)
    expect2 = t2[rows_t].sum(dim=1)
    expect2.backward(g)
    assert torch.allclose(table.grad, t2.grad, atol=1e-6)


def test_layernorm_backward_matches_autograd():
    torch.manual_seed(3)
    X = torch.randn(9, 16, requires_grad=True, dtype=torch.float64)
    g_ = torch.randn(16, requires_grad=True, dtype=torch.float64)
    b_ = torch.randn(16, requires_grad=True, dtype=torch.float64)
    Y = ops.layernorm(X, g_, b_, 1e-5)
    gout = torch.randn_like(Y)
    Y.backward(gout)
    X2 = X.detach().clone().requires_grad_(True)
    g2 = g_.detach().clone().requires_grad_(True)
    b2 = b_.detach().clone().requires_grad_(True)
    Y2 = torch.nn.functional.layer_norm(X2, (16,), g2, b2, 1e-5)
    Y2.backward(gout)
    assert torch.allclose(X.grad, X2.grad, atol=1e-8)
    assert torch.allclose(g_.grad, g2.grad, atol=1e-8)
    assert torch.allclose(b_.grad, b2.grad, atol=1e-8)


def test_parser_step_score_fwd_bwd():
    torch.manual_seed(4)
    T, nF, H, P, S = 10, 13, 6, 2, 5
    pre = torch.randn(T + 1, nF, H * P, requires_grad=True)
    bias = torch.randn(H * P, requires_grad=True)
    feats = torch.randint(0, T + 1, (S, nF))
    hidden = ops.parser_step_score(pre, feats, bias)
    assert hidden.shape == (S, H)
    g = torch.randn_like(hidden)
    hidden.backward(g)
    pre2 = pre.detach().clone().requires_grad_(True)
    bias2 = bias.detach().clone().requires_grad_(True)
    slot = torch.arange(nF).unsqueeze(0)
    summed = pre2[feats.long(), slot].sum(dim=1) + bias2
    h2 = summed.view(S, P, H).max(dim=-2).values  # pieces-major
    h2.backward(g)
    assert torch.allclose(pre.grad, pre2.grad, atol=1e-6)
    assert torch.allclose(bias.grad, bias2.grad, atol=1e-6)


def test_ragged_reductions():
    X = torch.arange(12, dtype=torch.float32).reshape(6, 2)
    lengths = torch.tensor([2, 1, 3])
    s = ref.reduce_sum_ragged(X, lengths)
    assert torch.allclose(s[0], X[0] + X[1])
    assert torch.allclose(s[1], X[2])
    m = ref.reduce_mean_ragged(X, lengths)
    assert torch.allclose(m[2], X[3:].mean(dim=0))
    mx, which = ref.reduce_max_ragged(X, lengths)
    assert torch.allclose(mx[2], X[3:].max(dim=0).values)


def test_activations_match_torch_builtins():
    """mish/swish/gelu equal torch's own implementations; clipped_linear
    family obeys its closed form (Thinc kernel-surface parity)."""
    import torch
    import torch.nn.functional as F

    from spacy_ray_amd.ops import api

    x = torch.randn(257, 19)
    assert torch.allclose(api.swish(x), F.silu(x), atol=1e-6)
    assert torch.allclose(api.mish(x), F.mish(x), atol=1e-6)
    assert torch.allclose(api.gelu(x), F.gelu(x), atol=1e-5)  # erf form
    assert torch.allclose(api.relu(x), F.relu(x))
    assert torch.allclose(api.hard_tanh(x), torch.clamp(x, -1, 1))
    hs = torch.clamp(0.2 * x + 0.5, 0.0, 1.0)
    assert torch.allclose(api.hard_sigmoid(x), hs)
    # backward parity against autograd on the torch builtins
    xa = x.clone().requires_grad_(True)
    xb = x.clone().requires_grad_(True)
    api.mish(xa).sum().backward()
    F.mish(xb).sum().backward()
    assert torch.allclose(xa.grad, xb.grad, atol=1e-5)
