"""Counter-based Philox4x32-10 RNG — the BNS sampling core.

The reference samples boundary nodes with CPU `np.random.choice` and ships
the chosen IDs to the receiver every epoch (reference: train.py:225-236 and
the NODE-tag transfer at train.py:389). We instead use a counter-based RNG
keyed on (seed, epoch, src_rank, dst_rank): sender and receiver derive the
SAME sample independently, so the per-epoch ID exchange disappears
entirely, and sampling is reproducible (fixing reference quirk SURVEY.md
§2.5.7: numpy was never seeded).

The HIP kernel (ops/hip/kernels.hip: philox_keys_kernel) implements the
identical function; tests assert bitwise equality between this numpy
implementation and the device kernel.
"""
from __future__ import annotations

import numpy as np

_M0 = np.uint64(0xD2511F53)
_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)
_MASK32 = np.uint64(0xFFFFFFFF)


def philox4x32(c0, c1, c2, c3, k0, k1, rounds: int = 10):
    """Vectorized Philox4x32. c0 may be an array; others scalars or arrays."""
    c0 = np.asarray(c0, dtype=np.uint32)
    c1 = np.broadcast_to(np.uint32(c1), c0.shape).copy()
    c2 = np.broadcast_to(np.uint32(c2), c0.shape).copy()
    c3 = np.broadcast_to(np.uint32(c3), c0.shape).copy()
    k0 = np.uint32(k0)
    k1 = np.uint32(k1)
    for _ in range(rounds):
        p0 = c0.astype(np.uint64) * _M0
        p1 = c2.astype(np.uint64) * _M1
        hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
        lo0 = (p0 & _MASK32).astype(np.uint32)
        hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
        lo1 = (p1 & _MASK32).astype(np.uint32)
        c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
        k0 = np.uint32((np.uint64(k0) + np.uint64(_W0)) & _MASK32)
        k1 = np.uint32((np.uint64(k1) + np.uint64(_W1)) & _MASK32)
    return c0, c1, c2, c3


def bns_keys(n: int, seed: int, epoch: int, src_rank: int, dst_rank: int) -> np.ndarray:
    """63-bit sort keys for positions [0, n): int64, non-negative.

    Key layout (must match the HIP kernel exactly):
      counter = (i, epoch, src_rank<<16 | dst_rank, 0x424E5347)  # "BNSG"
      key     = (k0, k1) = (seed & 0xffffffff, seed >> 32)
      out     = ((o0 << 32) | o1) & 0x7fffffffffffffff
    """
    i = np.arange(n, dtype=np.uint32)
    c2 = np.uint32(((src_rank & 0xFFFF) << 16) | (dst_rank & 0xFFFF))
    o0, o1, _, _ = philox4x32(i, np.uint32(epoch & 0xFFFFFFFF), c2,
                              np.uint32(0x424E5347),
                              np.uint32(seed & 0xFFFFFFFF),
                              np.uint32((seed >> 32) & 0xFFFFFFFF))
    key = (o0.astype(np.uint64) << np.uint64(32)) | o1.astype(np.uint64)
    return (key & np.uint64(0x7FFFFFFFFFFFFFFF)).astype(np.int64)


def sample_boundary(n: int, s: int, seed: int, epoch: int,
                    src_rank: int, dst_rank: int) -> np.ndarray:
    """Uniform sample WITHOUT replacement of s positions from [0, n),
    returned sorted ascending (int64). Deterministic in all arguments —
    sender (src_rank) and receiver (dst_rank) call this with the same
    arguments and get the same positions.

    Equivalent in distribution to the reference's
    np.random.choice(n, s, replace=False) (train.py:233-234): the s
    smallest of n distinct random keys are a uniform s-subset.
    """
    if s >= n:
        return np.arange(n, dtype=np.int64)
    if s <= 0:
        return np.zeros(0, dtype=np.int64)
    keys = bns_keys(n, seed, epoch, src_rank, dst_rank)
    order = np.argsort(keys, kind="stable")
    return np.sort(order[:s]).astype(np.int64)
