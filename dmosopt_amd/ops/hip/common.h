// Common device utilities for dmosopt_amd gfx950 kernels.
// CDNA4-only: wave64, no CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE_SIZE 64

__device__ __forceinline__ float warp_reduce_sum(float v) {
  // full wave64 reduction via xor shuffles
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float warp_reduce_max(float v) {
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// ---------------------------------------------------------------- Philox4x32
// Counter-based RNG (Salmon et al. 2011), 10 rounds. Each call site derives
// independent streams from (seed, counter) — the per-kernel RNG discipline
// for reproducible batched variation ops.
struct Philox4 {
  unsigned int c0, c1, c2, c3;
};

__device__ __forceinline__ unsigned int mulhilo(unsigned int a, unsigned int b,
                                                unsigned int* hi) {
  unsigned long long p = (unsigned long long)a * b;
  *hi = (unsigned int)(p >> 32);
  return (unsigned int)p;
}

__device__ __forceinline__ Philox4 philox4x32(unsigned long long seed,
                                              unsigned long long counter) {
  unsigned int k0 = (unsigned int)seed;
  unsigned int k1 = (unsigned int)(seed >> 32);
  Philox4 s;
  s.c0 = (unsigned int)counter;
  s.c1 = (unsigned int)(counter >> 32);
  s.c2 = 0x9E3779B9u;
  s.c3 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    unsigned int hi0, hi1;
    unsigned int lo0 = mulhilo(0xD2511F53u, s.c0, &hi0);
    unsigned int lo1 = mulhilo(0xCD9E8D57u, s.c2, &hi1);
    Philox4 n;
    n.c0 = hi1 ^ s.c1 ^ k0;
    n.c1 = lo1;
    n.c2 = hi0 ^ s.c3 ^ k1;
    n.c3 = lo0;
    s = n;
    k0 += 0x9E3779B9u;
    k1 += 0xBB67AE85u;
  }
  return s;
}

// uniform in (0, 1): avoid exact 0/1 for pow() stability
__device__ __forceinline__ float u01(unsigned int x) {
  return ((float)x + 0.5f) * 2.3283064365386963e-10f;  // (x+0.5)/2^32
}
