"""A/B the hand-written MFMA GEMM vs rocBLAS (torch) on the bench shapes.

Run on a GPU box:  python tools/gemm_micro.py
"""
import time

import torch

from bnsgcn_amd.ops._ext import get_ext

ext = get_ext()
dev = "cuda:0"


def t(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def bench_shape(M, K, N):
    x = torch.randn(M, K, device=dev)
    w = torch.randn(N, K, device=dev)       # torch Linear layout [out, in]
    b = torch.randn(N, device=dev)
    g = torch.randn(M, N, device=dev)
    wt = w.t().contiguous()

    r = {}
    r["ours nt+bias"] = t(lambda: ext.gemm_nt_bias(x, w, b))
    r["torch linear"] = t(lambda: torch.nn.functional.linear(x, w, b))
    r["ours nn (dx)"] = t(lambda: ext.gemm_nn(g, w))
    r["torch mm (dx)"] = t(lambda: torch.mm(g, w))
    r["ours tn (dW)"] = t(lambda: ext.gemm_tn(g, x))
    r["torch tn (dW)"] = t(lambda: torch.mm(g.t(), x))
    print(f"[M={M} K={K} N={N}]")
    for k, v in r.items():
        fl = 2 * M * K * N / (v * 1e-3) / 1e12
        print(f"  {k:14s} {v:7.3f} ms  {fl:7.1f} TF/s")


for shape in [(232965, 1204, 256), (232965, 256, 256), (232965, 256, 41),
              (2449029, 128, 128), (716847, 512, 100)]:
    bench_shape(*shape)
