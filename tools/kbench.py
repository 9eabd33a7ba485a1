"""Micro-benchmark individual HIP kernels (hot-op iteration harness).

Usage (on a GPU box):  python tools/kbench.py [names...]
Names default to all.  Prints ms/call over 20 timed iterations.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

import spacy_ray_amd  # noqa: F401  (loads _srx_hip)
from spacy_ray_amd.ops.api import hip_ext

hip = hip_ext()
assert hip is not None
dev = "cuda"
torch.manual_seed(0)


def timeit(name, fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / iters * 1e3
    print(f"{name:32s} {ms:8.3f} ms/call")
    return ms


def bench_mwe():
    T, W = 1000064, 96
    X = torch.randn(T, W, device=dev, dtype=torch.bfloat16)
    Wt = torch.randn(3 * W, 3 * W, device=dev, dtype=torch.bfloat16) * 0.05
    bias = torch.randn(3 * W, device=dev, dtype=torch.bfloat16)
    g = torch.ones(W, device=dev, dtype=torch.bfloat16)
    b = torch.zeros(W, device=dev, dtype=torch.bfloat16)
    starts = torch.zeros(T, device=dev, dtype=torch.uint8)
    ends = torch.zeros(T, device=dev, dtype=torch.uint8)
    starts[::20] = 1
    ends[19::20] = 1
    timeit("mwe_layer_fwd (T=1M,W=96)",
           lambda: hip.mwe_layer_fwd(X, Wt, bias, g, b, starts, ends, None, 1e-5))
    dY = torch.randn(T, W, device=dev, dtype=torch.bfloat16)
    Y, Mout, which, mu, rstd = hip.mwe_layer_fwd(X, Wt, bias, g, b, starts,
                                                 ends, None, 1e-5)
    dm = torch.ones(T, W, device=dev, dtype=torch.bfloat16)
    timeit("mwe_bwd_stage1 (T=1M,W=96)",
           lambda: hip.mwe_bwd_stage1(dY, dm, Mout, g, mu, rstd, which))


def bench_dpre():
    import numpy as np

    nF, HP, L = 13, 128, 20
    n_docs = 50000
    T = n_docs * L
    SS = 2 * T
    rng = np.random.default_rng(0)
    # doc-major feats: row r of doc d draws from doc d's tokens or pad
    doc = np.repeat(np.arange(n_docs, dtype=np.int64), 2 * L)
    base = doc * L
    f = base[:, None] + rng.integers(0, L, (SS, nF))
    pad = rng.random((SS, nF)) < 0.25
    f[pad] = T
    feats = torch.from_numpy(f).to(dev)
    dS = torch.randn(SS, HP, device=dev, dtype=torch.bfloat16)
    # realistic sparsity: dSummed comes out of maxout_bwd — exactly one of
    # each piece pair (h, H+h) is nonzero
    pick = torch.randint(0, 2, (SS, 1, HP // 2), device=dev)
    dS.view(SS, 2, HP // 2).scatter_(1, pick, 0.0)
    dPre = torch.zeros(T + 1, nF, HP, device=dev, dtype=torch.bfloat16)
    timeit("dpre_scatter bf16 (1.9M rows)",
           lambda: hip.dpre_scatter(dS, feats, dPre, T))
    off = torch.arange(n_docs, device=dev, dtype=torch.int32) * L
    lens = torch.full((n_docs,), L, device=dev, dtype=torch.int32)
    dPre2 = torch.empty(T + 1, nF, HP, device=dev, dtype=torch.bfloat16)
    timeit("dpre_docmajor bf16 (1.9M rows)",
           lambda: hip.dpre_scatter_docmajor(dS, feats, dPre2, off, lens,
                                             T, 2, L))
    # diagnostics: atomic-op-rate hypothesis — nF=1 should be ~1/13 the
    # time if op-bound; fp32 (2x the ops of packed bf16) should be ~2x
    feats1 = feats[:, :1].contiguous()
    dPre1 = torch.zeros(T + 1, 1, HP, device=dev, dtype=torch.bfloat16)
    timeit("dpre_scatter bf16 nF=1",
           lambda: hip.dpre_scatter(dS, feats1, dPre1, T))
    dS32 = dS.float()
    dPre32 = torch.zeros(T + 1, nF, HP, device=dev, dtype=torch.float32)
    timeit("dpre_scatter fp32 (2x ops)",
           lambda: hip.dpre_scatter(dS32, feats, dPre32, T))


def bench_ce():
    SS, A = 2000000, 82
    scores = torch.randn(SS, A, device=dev, dtype=torch.bfloat16)
    valid = (torch.rand(SS, A, device=dev) < 0.5).to(torch.uint8)
    valid[:, 0] = 1
    gold = ((torch.rand(SS, A, device=dev) < 0.1).to(torch.uint8) & valid)
    timeit("transition_ce (2M rows, A=82)",
           lambda: hip.transition_ce(scores, gold, valid))


def bench_gpustate():
    n_docs, L, nL = 50000, 20, 40
    T = n_docs * L
    H, nF = 64, 13
    lens = torch.full((n_docs,), L, device=dev, dtype=torch.int32)
    off = (torch.arange(n_docs, device=dev, dtype=torch.int32) * L)
    pre = torch.randn(T + 1, nF, 2 * H, device=dev, dtype=torch.bfloat16)
    lowerB = torch.randn(2 * H, device=dev, dtype=torch.bfloat16)
    A = 2 + 2 * nL
    upperW = torch.randn(A, H, device=dev, dtype=torch.bfloat16)
    upperB = torch.randn(A, device=dev, dtype=torch.bfloat16)
    import numpy as np

    rng = np.random.default_rng(0)
    gh_np = np.zeros(T, dtype=np.int32)
    for d in range(n_docs):
        for i in range(1, L):
            gh_np[d * L + i] = rng.integers(0, i)
        gh_np[d * L] = -1
    gl_np = rng.integers(0, nL, T).astype(np.int32)
    tok_off = np.repeat(np.arange(n_docs, dtype=np.int64) * L, L)
    local = np.arange(T, dtype=np.int64) - tok_off
    ok = gh_np >= 0
    parent = tok_off[ok] + gh_np[ok]
    order = np.argsort(parent, kind="stable")
    kids_np = local[ok][order].astype(np.int32)
    kids_off_np = np.zeros(T + 1, dtype=np.int64)
    np.cumsum(np.bincount(parent, minlength=T), out=kids_off_np[1:])
    gh = torch.from_numpy(gh_np).to(dev)
    gl = torch.from_numpy(gl_np).to(dev)
    kids = torch.from_numpy(kids_np).to(dev)
    kids_off = torch.from_numpy(kids_off_np.astype(np.int32)).to(dev)
    timeit("gpu_arceager train (50k docs)",
           lambda: hip.gpu_arceager(pre, off, lens, gh, gl, kids_off, kids,
                                    lowerB, upperW, upperB, T, nL, True))
    timeit("gpu_arceager decode (50k docs)",
           lambda: hip.gpu_arceager(pre, off, lens, gh, gl, kids_off, kids,
                                    lowerB, upperW, upperB, T, nL, False))
    A2 = 1 + 4 * 18
    pre6 = torch.randn(T + 1, 6, 2 * H, device=dev, dtype=torch.bfloat16)
    upW2 = torch.randn(A2, H, device=dev, dtype=torch.bfloat16)
    upB2 = torch.randn(A2, device=dev, dtype=torch.bfloat16)
    gold = torch.zeros(T, device=dev, dtype=torch.int32)
    timeit("gpu_biluo train (50k docs)",
           lambda: hip.gpu_biluo(pre6, off, lens, gold, lowerB, upW2, upB2,
                                 T, 18, True))


def bench_attn():
    """Fused flash-style MFMA attention vs SDPA flash at trf window shapes."""
    from spacy_ray_amd.ops.api import window_attention

    B, H, L, D = 1024, 12, 96, 64  # ~one bucket's worth of windows
    lens = torch.randint(20, L + 1, (B,), device=dev, dtype=torch.int32)
    q = torch.randn(B, H, L, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    scale = D ** -0.5

    def fused_fb():
        out = window_attention(q, k, v, lens, scale, 0.1)
        out.sum().backward()
        q.grad = k.grad = v.grad = None

    timeit("attn fused fwd+bwd (1k win)", fused_fb)

    mask = (torch.arange(L, device=dev)[None, :] < lens[:, None]).view(B, 1, 1, L)

    def sdpa_fb():
        out = torch.nn.functional.scaled_dot_product_attention(
            q, k, v, attn_mask=mask, dropout_p=0.1, scale=scale)
        out.sum().backward()
        q.grad = k.grad = v.grad = None

    timeit("attn sdpa  fwd+bwd (1k win)", sdpa_fb)


ALL = {"mwe": bench_mwe, "dpre": bench_dpre, "ce": bench_ce,
       "gpustate": bench_gpustate, "attn": bench_attn}

if __name__ == "__main__":
    names = sys.argv[1:] or list(ALL)
    for n in names:
        ALL[n]()
