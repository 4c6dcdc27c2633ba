"""Pre-flight coverage for the driver's multi-GPU bench path (VERDICT r1
item 1): every code path `bench.py --gpus N` exercises at N>1 must run
green here on CPU/gloo before it ever meets 8 MI355X ranks.

- all_to_all_rows edge cases: per-peer zero splits, a rank that sends
  nothing but receives, asymmetric counts (comm.py fallback + the
  size-contract plumbing the RCCL path shares).
- an actual `torch.distributed.run --nproc-per-node 4 bench.py` subprocess
  on the tiny dataset — the exact launch shape the driver uses for
  SCALE_rNN.json, minus the GPUs.
"""
import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from util_dist import run_dist, free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _a2a_rank(rank, world, send_mat, recv_mat):
    """send_mat[i][j] = rows i sends to j. Fill rows with a (src,dst,k)
    signature and verify every received row."""
    import torch
    from bnsgcn_amd.parallel import init_distributed, all_to_all_rows
    init_distributed("gloo", rank, world)
    send_counts = list(send_mat[rank])
    recv_counts = [send_mat[j][rank] for j in range(world)]
    F = 5
    blocks = []
    for j in range(world):
        n = send_counts[j]
        b = torch.zeros(n, F)
        for k in range(n):
            b[k] = torch.tensor([rank, j, k, rank * 100 + k, 1.0])
        blocks.append(b)
    send = torch.cat(blocks) if blocks else torch.zeros(0, F)
    recv = torch.empty(sum(recv_counts), F)
    all_to_all_rows(recv, send, recv_counts, send_counts)
    off = 0
    for j in range(world):
        for k in range(recv_counts[j]):
            row = recv[off + k]
            assert row[0] == j and row[1] == rank and row[2] == k, \
                (rank, j, k, row.tolist())
        off += recv_counts[j]
    return True


@pytest.mark.parametrize("send_mat", [
    # world=3, asymmetric with per-peer zeros: 0→1 only; 1 sends nothing;
    # 2 sends to 0 and 1 unequal amounts
    [[0, 3, 0], [0, 0, 0], [2, 5, 0]],
    # one hot pair, everything else silent (p→0 asymmetry)
    [[0, 0, 7], [0, 0, 0], [0, 0, 0]],
], ids=["asymmetric-zeros", "single-pair"])
def test_all_to_all_rows_zero_splits(send_mat):
    res = run_dist(3, _a2a_rank, (send_mat, None))
    assert all(res)


def test_all_to_all_rows_all_zero():
    res = run_dist(2, _a2a_rank, ([[0, 0], [0, 0]], None))
    assert all(res)


def _run_bench_subprocess(nproc: int, extra=(), device="cpu", timeout=900):
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={nproc}",
           "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()),
           os.path.join(REPO, "bench.py"),
           "--gpus", str(nproc), "--steps", "3", "--warmup", "1",
           "--dataset", "tiny", "--device", device,
           "--partition-dir", os.path.join(REPO, "bench_partition")] + list(extra)
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout,
                         env=env, cwd=REPO)
    assert out.returncode == 0, f"stdout:\n{out.stdout}\nstderr:\n{out.stderr}"
    line = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert line, out.stdout
    return json.loads(line[-1])


def test_bench_multirank_cpu_gloo():
    """The driver's exact N>1 launch (torch.distributed.run, one rank per
    'GPU') end-to-end on CPU/gloo, world=4."""
    res = _run_bench_subprocess(4)
    assert res["n_gpus"] == 4
    assert res["steps"] == 3
    assert np.isfinite(res["value"]) and res["value"] > 0
    assert res["config"]["parallelism"] == "partition-parallel p4"


@pytest.mark.gpu
def test_bench_multirank_cpu_gloo_w8_on_gpubox():
    """Same launch at the full world=8 the driver will use — run inside the
    GPU test tier so GPUTEST records it (VERDICT r1 item 1 'done' bar)."""
    res = _run_bench_subprocess(8)
    assert res["n_gpus"] == 8
    assert np.isfinite(res["value"]) and res["value"] > 0


@pytest.mark.gpu
def test_bench_8rank_reddit_gloo_cuda_oversubscribed():
    """The FULL 8-way decomposition on real GPU hardware: 8 ranks share
    cuda:0 over gloo (host-staged wires) on a quarter-scale Reddit —
    multilevel 8-way partitioning, per-rank restricted training, sampled
    halo exchange among 8 peers, bucketed all-reduce. The exact
    decomposition the driver's 8-GPU SCALE run uses, minus RCCL.
    (A full-scale run of this shape is archived in
    profiles/: bench_w8_cuda_r02, 245 ms/step host-staged.)"""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env.pop("LOCAL_RANK", None)
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node=8",
           "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()),
           os.path.join(REPO, "bench.py"),
           "--gpus", "8", "--steps", "2", "--warmup", "1",
           "--dataset", "reddit", "--data-scale", "0.25",
           "--device", "cuda:0", "--backend", "gloo",
           "--partition-dir", os.path.join(REPO, "bench_partition")]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=900,
                         env=env, cwd=REPO)
    assert out.returncode == 0, f"stdout:\n{out.stdout[-3000:]}\n" \
                                f"stderr:\n{out.stderr[-3000:]}"
    line = [l for l in out.stdout.splitlines() if l.startswith("{")]
    res = json.loads(line[-1])
    assert res["n_gpus"] == 8 and np.isfinite(res["value"])


@pytest.mark.gpu
def test_bench_multirank_gloo_cuda_oversubscribed():
    """2 ranks sharing cuda:0 over gloo (host-staged payloads): real HIP
    kernels + multi-rank halo exchange (with the bf16 wire dtype) +
    bucketed all-reduce on ONE GPU — the closest 1-GPU stand-in for the
    8-rank RCCL job (reference gloo oversubscription, main.py:45)."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    res = _run_bench_subprocess(2, extra=["--sampling-rate", "0.5",
                                          "--backend", "gloo",
                                          "--halo-dtype", "bf16"],
                                device="cuda:0")
    assert res["n_gpus"] == 2
    assert np.isfinite(res["value"]) and res["value"] > 0
