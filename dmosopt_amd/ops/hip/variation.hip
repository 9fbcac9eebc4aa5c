// Fused variation: SBX crossover + polynomial mutation with Philox RNG.
//
// Replaces reference MOEA.py:191-239 (per-individual numpy loops): one
// launch produces an entire generation's children. Gene (row, dim) maps to
// one Philox counter, so results are reproducible from (seed) alone and
// independent of launch geometry.

#include "common.h"
#include <math.h>


// SBX / mutation child arithmetic with a PINNED fmaf order: under the
// default -ffp-contract=fast the compiler may contract the blend
// differently in different kernels (observed: sbx_batch_kernel vs
// variation_events_kernel differed by ULPs for identical Philox draws),
// which breaks the cross-path bitwise-identity guarantee the tests assert.
__device__ __forceinline__ void sbx_children(float a, float b, float beta,
                                             float lo, float hi, float* c1,
                                             float* c2) {
  const float x1 = 0.5f * fmaf(-beta, a, fmaf(beta, b, a + b));
  const float x2 = 0.5f * fmaf(beta, a, fmaf(-beta, b, a + b));
  *c1 = fminf(fmaxf(x1, lo), hi);
  *c2 = fminf(fmaxf(x2, lo), hi);
}

__device__ __forceinline__ float mutated_child(float p, float delta, float lo,
                                               float hi) {
  const float v = fmaf(hi - lo, delta, p);
  return fminf(fmaxf(v, lo), hi);
}

// children layout: C pairs first (child1 rows [0,C), child2 rows [C,2C))
// — the host reorders into event order with an index gather.
__global__ void sbx_batch_kernel(const float* __restrict__ pool,  // (K, d)
                                 const int* __restrict__ p1,      // (C,)
                                 const int* __restrict__ p2,      // (C,)
                                 const float* __restrict__ di,    // (d,)
                                 const float* __restrict__ lo,    // (d,)
                                 const float* __restrict__ hi,    // (d,)
                                 float* __restrict__ out,         // (2C, d)
                                 int C, int d,
                                 unsigned long long seed) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)C * d) return;
  const int c = (int)(idx / d);
  const int g = (int)(idx % d);
  const float a = pool[p1[c] * d + g];
  const float b = pool[p2[c] * d + g];
  const Philox4 r = philox4x32(seed, (unsigned long long)idx);
  const float u = u01(r.c0);
  const float e = 1.f / (di[g] + 1.f);
  const float beta = (u <= 0.5f) ? __powf(2.f * u, e)
                                 : __powf(1.f / (2.f * (1.f - u)), e);
  float c1, c2;
  sbx_children(a, b, beta, lo[g], hi[g], &c1, &c2);
  out[(long long)c * d + g] = c1;
  out[((long long)C + c) * d + g] = c2;
}

__global__ void mutation_batch_kernel(const float* __restrict__ pool,  // (K, d)
                                      const int* __restrict__ parents,  // (M,)
                                      const float* __restrict__ di,
                                      const float* __restrict__ lo,
                                      const float* __restrict__ hi,
                                      float* __restrict__ out,  // (M, d)
                                      int M, int d, float mutation_rate,
                                      unsigned long long seed) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)M * d) return;
  const int r = (int)(idx / d);
  const int g = (int)(idx % d);
  const float p = pool[parents[r] * d + g];
  const Philox4 ph = philox4x32(seed ^ 0x5deece66dULL, (unsigned long long)idx);
  const float u = u01(ph.c0);
  const float e = 1.f / (di[g] + 1.f);
  const float delta = (u < mutation_rate)
                          ? __powf(2.f * u, e) - 1.f
                          : 1.f - __powf(2.f * (1.f - u), e);
  out[(long long)r * d + g] = mutated_child(p, delta, lo[g], hi[g]);
}

extern "C" void launch_sbx_batch(const float* pool, const int* p1,
                                 const int* p2, const float* di,
                                 const float* lo, const float* hi, float* out,
                                 int C, int d, unsigned long long seed,
                                 hipStream_t stream) {
  long long total = (long long)C * d;
  int blocks = (int)((total + 255) / 256);
  hipLaunchKernelGGL(sbx_batch_kernel, dim3(blocks), dim3(256), 0, stream,
                     pool, p1, p2, di, lo, hi, out, C, d, seed);
}

extern "C" void launch_mutation_batch(const float* pool, const int* parents,
                                      const float* di, const float* lo,
                                      const float* hi, float* out, int M,
                                      int d, float mutation_rate,
                                      unsigned long long seed,
                                      hipStream_t stream) {
  long long total = (long long)M * d;
  int blocks = (int)((total + 255) / 256);
  hipLaunchKernelGGL(mutation_batch_kernel, dim3(blocks), dim3(256), 0, stream,
                     pool, parents, di, lo, hi, out, M, d, mutation_rate, seed);
}

// Whole-generation variation in ONE launch: slot s of the output is child
// src_rows[s] of the virtual [sbx_c1 | sbx_c2 | mutation] stack — decoded
// on the fly, so the separate sbx + mutation + cat + gather launches (and
// their ~10 us in-stream gaps each) collapse into one kernel. Philox
// counters match the split kernels exactly: identical output bits.
__global__ void variation_slots_kernel(
    const float* __restrict__ pool, const long long* __restrict__ src_rows,
    const long long* __restrict__ p1, const long long* __restrict__ p2,
    const long long* __restrict__ im, const float* __restrict__ di_c,
    const float* __restrict__ di_m, const float* __restrict__ lo,
    const float* __restrict__ hi, float* __restrict__ out, int total, int C,
    int d, float mutation_rate, unsigned long long seed_sbx,
    unsigned long long seed_mut) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)total * d) return;
  const int s = (int)(idx / d);
  const int g = (int)(idx % d);
  const long long row = src_rows[s];
  if (row < 2 * C) {
    const int c = (int)(row < C ? row : row - C);
    const float a = pool[p1[c] * d + g];
    const float b = pool[p2[c] * d + g];
    const Philox4 r = philox4x32(seed_sbx, (unsigned long long)c * d + g);
    const float u = u01(r.c0);
    const float e = 1.f / (di_c[g] + 1.f);
    const float beta = (u <= 0.5f) ? __powf(2.f * u, e)
                                   : __powf(1.f / (2.f * (1.f - u)), e);
    float c1, c2;
    sbx_children(a, b, beta, lo[g], hi[g], &c1, &c2);
    out[idx] = (row < C) ? c1 : c2;
    return;
  } else {
    const int m = (int)(row - 2 * C);
    const float p = pool[im[m] * d + g];
    const Philox4 ph =
        philox4x32(seed_mut ^ 0x5deece66dULL, (unsigned long long)m * d + g);
    const float u = u01(ph.c0);
    const float e = 1.f / (di_m[g] + 1.f);
    const float delta = (u < mutation_rate)
                            ? __powf(2.f * u, e) - 1.f
                            : 1.f - __powf(2.f * (1.f - u), e);
    out[idx] = mutated_child(p, delta, lo[g], hi[g]);
  }
}

// Event-decoded variant: thread (event, gene) writes its own output slots
// directly — crossover event k computes BOTH children (sharing one Philox
// draw, like sbx_batch_kernel) and scatters them to rows ci[2k], ci[2k+1];
// mutation event k scatters to mi[k]. The slot lists partition [0, 2C+M),
// so every output row is written exactly once and the host never builds the
// inverse src_rows map (3 numpy scatters + 2 aranges per generation in the
// slot-decoded path). Philox counters match the split kernels: identical
// output bits, only the addressing differs.
__global__ void variation_events_kernel(
    const float* __restrict__ pool, const long long* __restrict__ ci,
    const long long* __restrict__ mi, const long long* __restrict__ p1,
    const long long* __restrict__ p2, const long long* __restrict__ im,
    const float* __restrict__ di_c, const float* __restrict__ di_m,
    const float* __restrict__ lo, const float* __restrict__ hi,
    float* __restrict__ out, int C, int M, int d, float mutation_rate,
    unsigned long long seed_sbx, unsigned long long seed_mut) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)(C + M) * d) return;
  const int ev = (int)(idx / d);
  const int g = (int)(idx % d);
  if (ev < C) {
    const float a = pool[p1[ev] * d + g];
    const float b = pool[p2[ev] * d + g];
    const Philox4 r = philox4x32(seed_sbx, (unsigned long long)ev * d + g);
    const float u = u01(r.c0);
    const float e = 1.f / (di_c[g] + 1.f);
    const float beta = (u <= 0.5f) ? __powf(2.f * u, e)
                                   : __powf(1.f / (2.f * (1.f - u)), e);
    float c1, c2;
    sbx_children(a, b, beta, lo[g], hi[g], &c1, &c2);
    out[ci[2 * ev] * d + g] = c1;
    out[ci[2 * ev + 1] * d + g] = c2;
  } else {
    const int m = ev - C;
    const float p = pool[im[m] * d + g];
    const Philox4 ph =
        philox4x32(seed_mut ^ 0x5deece66dULL, (unsigned long long)m * d + g);
    const float u = u01(ph.c0);
    const float e = 1.f / (di_m[g] + 1.f);
    const float delta = (u < mutation_rate)
                            ? __powf(2.f * u, e) - 1.f
                            : 1.f - __powf(2.f * (1.f - u), e);
    out[mi[m] * d + g] = mutated_child(p, delta, lo[g], hi[g]);
  }
}

extern "C" void launch_variation_events(
    const float* pool, const long long* ci, const long long* mi,
    const long long* p1, const long long* p2, const long long* im,
    const float* di_c, const float* di_m, const float* lo, const float* hi,
    float* out, int C, int M, int d, float mutation_rate,
    unsigned long long seed_sbx, unsigned long long seed_mut,
    hipStream_t stream) {
  long long n = (long long)(C + M) * d;
  if (n <= 0) return;
  hipLaunchKernelGGL(variation_events_kernel, dim3((int)((n + 255) / 256)),
                     dim3(256), 0, stream, pool, ci, mi, p1, p2, im, di_c,
                     di_m, lo, hi, out, C, M, d, mutation_rate, seed_sbx,
                     seed_mut);
}

extern "C" void launch_variation_slots(
    const float* pool, const long long* src_rows, const long long* p1,
    const long long* p2, const long long* im, const float* di_c,
    const float* di_m, const float* lo, const float* hi, float* out,
    int total, int C, int d, float mutation_rate, unsigned long long seed_sbx,
    unsigned long long seed_mut, hipStream_t stream) {
  long long n = (long long)total * d;
  hipLaunchKernelGGL(variation_slots_kernel, dim3((int)((n + 255) / 256)),
                     dim3(256), 0, stream, pool, src_rows, p1, p2, im, di_c,
                     di_m, lo, hi, out, total, C, d, mutation_rate, seed_sbx,
                     seed_mut);
}
