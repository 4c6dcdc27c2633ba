"""Single-kernel SpMM microbenchmark (for rocprofv3 PMC counter runs —
the full bench crashed rocprofv3's counter collection in this image)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from bnsgcn_amd.graph import CSR
from bnsgcn_amd.ops.functional import spmm_sum_raw


def main():
    assert torch.cuda.is_available()
    rng = np.random.default_rng(0)
    n, e, F = 200_000, 20_000_000, 256
    # power-law-ish columns with locality
    src = (rng.random(e) ** 4 * n).astype(np.int64)
    dst = np.sort(rng.integers(0, n, e))
    c = CSR.from_edges(src, dst, n, n)
    indptr = torch.from_numpy(c.indptr).cuda()
    indices = torch.from_numpy(c.indices).cuda()
    x = torch.randn(n, F, device="cuda")
    for _ in range(2):
        spmm_sum_raw(indptr, indices, x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 5
    for _ in range(iters):
        out = spmm_sum_raw(indptr, indices, x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    gb = e * F * 4 / 1e9
    print(f"spmm {e} edges F={F}: {dt*1e3:.3f} ms, {gb/dt:.2f} GB/s logical")


if __name__ == "__main__":
    main()
