"""Micro-benchmarks of individual ops at bench shapes (diagnosis tool)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from spacy_ray_amd.ops import api as ops


def timeit(name, fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.3f} ms")


def main():
    dev = "cuda:0"
    T = 131072
    dt = torch.bfloat16
    X4 = torch.randn(T, 384, device=dev, dtype=dt)
    Wm = torch.randn(288, 384, device=dev, dtype=dt)
    bm = torch.randn(288, device=dev, dtype=dt)
    timeit("mixer linear [T,384]x[384,288]", lambda: torch.nn.functional.linear(X4, Wm, bm))
    Y = torch.nn.functional.linear(X4, Wm, bm)
    timeit("maxout [T,3,96]", lambda: ops.maxout(Y.view(T, 3, 96)))
    X1 = torch.randn(T, 96, device=dev, dtype=dt)
    g = torch.ones(96, device=dev, dtype=dt)
    b = torch.zeros(96, device=dev, dtype=dt)
    timeit("layernorm [T,96]", lambda: ops.layernorm(X1, g, b))
    tab = torch.randn(5000, 96, device=dev, dtype=dt)
    ids = torch.randint(0, 2**62, (T,), device=dev)
    timeit("hashembed [T]", lambda: ops.hashembed(tab, ids, 3))
    outs = [torch.randn(T, 96, device=dev, dtype=dt) for _ in range(4)]
    timeit("cat 4x[T,96]", lambda: torch.cat(outs, dim=1))
    # tagger shapes
    Wt = torch.randn(50, 96, device=dev, dtype=dt)
    timeit("tagger linear [T,96]x[96,50]", lambda: torch.nn.functional.linear(X1, Wt))
    sc = torch.nn.functional.linear(X1, Wt)
    gold = torch.randint(0, 50, (T,), device=dev)
    timeit("softmax_ce [T,50]", lambda: ops.hip_ext().softmax_ce(sc.contiguous(), gold))
    # parser precompute GEMM
    Wl = torch.randn(13 * 128, 96, device=dev, dtype=dt)
    timeit("parser precompute [T,96]x[96,1664]", lambda: torch.nn.functional.linear(X1, Wl))
    # full mixer module path
    from spacy_ray_amd.models.layers import Maxout

    mix = Maxout(384, 96, pieces=3, normalize=True).to(dev).to(dt)
    timeit("Maxout module fwd (linear+maxout+LN)", lambda: mix(X4))
    with torch.enable_grad():
        X4g = X4.requires_grad_(True)
        def fwd_bwd():
            y = mix(X4g)
            y.backward(torch.ones_like(y))
        timeit("Maxout module fwd+bwd", fwd_bwd, iters=10)


if __name__ == "__main__":
    main()
