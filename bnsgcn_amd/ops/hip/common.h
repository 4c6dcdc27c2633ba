// Common device helpers for the bnsgcn_amd gfx950 kernel library.
// CDNA4: wavefront = 64 lanes, 4 waves per 256-thread block.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------- Philox
// Philox4x32-10 — MUST match bnsgcn_amd/ops/philox.py bitwise (tests
// assert equality). Key layout documented there.
DEV_INLINE uint32_t mulhi32(uint32_t a, uint32_t b) {
  return (uint32_t)(((uint64_t)a * b) >> 32);
}

struct P4 { uint32_t c0, c1, c2, c3; };

DEV_INLINE P4 philox4x32(uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3,
                         uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t hi0 = mulhi32(c0, M0), lo0 = c0 * M0;
    uint32_t hi1 = mulhi32(c2, M1), lo1 = c2 * M1;
    uint32_t n0 = hi1 ^ c1 ^ k0;
    uint32_t n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1;
    uint32_t n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}
