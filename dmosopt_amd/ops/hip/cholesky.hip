// Batched in-place Cholesky factorization, log-determinant, and
// forward/backward substitution — the GP fit/predict linear algebra
// (replaces the LAPACK calls inside sklearn GaussianProcessRegressor,
// reference model.py:1246-1265).
//
// Design: ONE workgroup (256 threads) per matrix, right-looking blocked
// factorization (BS = 32). v2 data path (gfx950):
//   * diagonal block factored in LDS (stride-33 rows: conflict-free),
//   * panel solve with each ROW held in 32 VGPRs (float4 global loads),
//     back-substituted against the LDS diagonal block,
//   * trailing SYRK update C -= P P^T fed ENTIRELY from LDS: the panel is
//     staged in chunk pairs (CHUNK x 32 floats each) so every multiply
//     reads LDS, with 2x2 register tiles per thread.
// Sized for the MO-ASMO regime: B = objectives x SCE-UA complexes matrices
// of N <= ~4k factorized concurrently, one CU each; batch parallelism fills
// the chip. Exact fp32; non-PD pivots clamp and set info[b].

#include "common.h"
#include <math.h>

#define CHOL_BS 32
#ifndef CHOL_TPB
#define CHOL_TPB 1024
#endif
#define CHOL_CHUNK 384  // panel rows staged per LDS buffer (384*32*4 = 48 KiB)

struct CholLds {
  // all row strides padded +1: an unpadded 32-float (128 B) stride puts
  // every row on the same b32 bank group -> 32-way conflicts in the SYRK
  float S[CHOL_BS][CHOL_BS + 1];
  float Pi[CHOL_CHUNK][CHOL_BS + 1];  // i-side panel chunk
  float Pj[CHOL_CHUNK][CHOL_BS + 1];  // j-side panel chunk
  float ld_accum;
};

// ~100 KiB of LDS: above the 64 KiB static default, so allocated
// dynamically (gfx950 has 160 KiB per CU).
template <bool WAVE_DIAG>
__global__ __launch_bounds__(CHOL_TPB) void cholesky_batched_kernel(
    float* __restrict__ A,       // (B, N, N)
    float* __restrict__ logdet,  // (B,)
    int* __restrict__ info,      // (B,)
    int N) {
  extern __shared__ char smem[];
  CholLds& L = *reinterpret_cast<CholLds*>(smem);
  const int b = blockIdx.x;
  float* M = A + (long long)b * N * N;
  const int tid = threadIdx.x;
  if (tid == 0) L.ld_accum = 0.f;
  __syncthreads();

  for (int k0 = 0; k0 < N; k0 += CHOL_BS) {
    const int bs = min(CHOL_BS, N - k0);

    // --- factor the diagonal block WAVE-SYNCHRONOUSLY in wave 0: lane l
    // holds row l in registers; column values broadcast with shuffles —
    // zero barriers inside the 32-step j loop (the previous LDS version
    // paid ~100 workgroup barriers per k-step).
    if (WAVE_DIAG && tid < WAVE_SIZE) {
      const int lane = tid;
      // the j/c loops are FULLY unrolled (compile-time indices) so r[]
      // stays in registers — runtime-indexed register arrays spill
      float r[CHOL_BS];
#pragma unroll
      for (int c = 0; c < CHOL_BS; ++c)
        r[c] = (lane < bs && c < bs) ? M[(long long)(k0 + lane) * N + k0 + c]
                                     : 0.f;
      float ld_part = 0.f;
      int bad = 0;
#pragma unroll
      for (int j = 0; j < CHOL_BS; ++j) {
        if (j >= bs) break;
        // lane j holds the fully-updated pivot
        float piv = __shfl(r[j], j);
        if (piv <= 0.f) {
          bad = 1;
          piv = 1e-30f;
        }
        const float sjj = sqrtf(piv);
        if (lane == j) {
          r[j] = sjj;
          ld_part += __logf(sjj);
        }
        const float lij = (lane > j && lane < bs) ? r[j] / sjj : 0.f;
        if (lane > j && lane < bs) r[j] = lij;
        // rank-1 update: a[i][c] -= L[i][j] * L[c][j] (lower triangle)
#pragma unroll
        for (int c = j + 1; c < CHOL_BS; ++c) {
          if (c >= bs) break;
          const float lcj = __shfl(lij, c);
          if (lane >= c && lane < bs) r[c] = fmaf(-lij, lcj, r[c]);
        }
      }
      if (lane < bs) {
#pragma unroll
        for (int c = 0; c < CHOL_BS; ++c) {
          const float v = (c <= lane) ? r[c] : 0.f;
          if (c < bs) {
            L.S[lane][c] = v;
            M[(long long)(k0 + lane) * N + k0 + c] = v;
          }
        }
      }
      const float ld_sum = warp_reduce_sum(ld_part);
      if (lane == 0) {
        L.ld_accum += ld_sum;
        if (bad) info[b] = 1;  // bad is wave-uniform (piv is broadcast)
      }
    }
    if (!WAVE_DIAG) {
      // LDS/barrier variant of the diagonal-block factorization
      for (int idx = tid; idx < bs * bs; idx += CHOL_TPB)
        L.S[idx / bs][idx % bs] =
            M[(long long)(k0 + idx / bs) * N + k0 + idx % bs];
      __syncthreads();
      for (int j = 0; j < bs; ++j) {
        if (tid == 0) {
          float djj = L.S[j][j];
          if (djj <= 0.f) {
            info[b] = 1;
            djj = 1e-30f;
          }
          L.S[j][j] = sqrtf(djj);
          L.ld_accum += __logf(L.S[j][j]);
        }
        __syncthreads();
        for (int i = j + 1 + tid; i < bs; i += CHOL_TPB) L.S[i][j] /= L.S[j][j];
        __syncthreads();
        const int rem_d = bs - j - 1;
        for (int idx = tid; idx < rem_d * rem_d; idx += CHOL_TPB) {
          const int rr = j + 1 + idx / rem_d;
          const int cc = j + 1 + idx % rem_d;
          if (cc <= rr) L.S[rr][cc] -= L.S[rr][j] * L.S[cc][j];
        }
        __syncthreads();
      }
      for (int idx = tid; idx < bs * bs; idx += CHOL_TPB) {
        const int rr = idx / bs, cc = idx % bs;
        M[(long long)(k0 + rr) * N + k0 + cc] = (cc <= rr) ? L.S[rr][cc] : 0.f;
      }
    }
    __syncthreads();

    const int rem = N - k0 - bs;
    if (rem <= 0) continue;

    // --- panel solve: row_i <- row_i * L11^-T with the row in registers.
    // float4 loads need 16-B alignment: row offset i*N + k0 is 16-B aligned
    // for all i iff N % 4 == 0 (k0 is a multiple of 32).
    const bool aligned4 = (N & 3) == 0;
    for (int i = k0 + bs + tid; i < N; i += CHOL_TPB) {
      float* grow = M + (long long)i * N + k0;
      float row[CHOL_BS];
      if (aligned4) {
#pragma unroll
        for (int t = 0; t < CHOL_BS; t += 4) {
          const float4 v = *(const float4*)(grow + t);
          row[t] = v.x; row[t + 1] = v.y; row[t + 2] = v.z; row[t + 3] = v.w;
        }
      } else {
#pragma unroll
        for (int t = 0; t < CHOL_BS; ++t) row[t] = grow[t];
      }
      for (int j = 0; j < bs; ++j) {
        float v = row[j];
        for (int t = 0; t < j; ++t) v = fmaf(-row[t], L.S[j][t], v);
        row[j] = v / L.S[j][j];
      }
      if (aligned4) {
#pragma unroll
        for (int t = 0; t < CHOL_BS; t += 4) {
          float4 v;
          v.x = row[t]; v.y = row[t + 1]; v.z = row[t + 2]; v.w = row[t + 3];
          *(float4*)(grow + t) = v;
        }
      } else {
#pragma unroll
        for (int t = 0; t < CHOL_BS; ++t) grow[t] = row[t];
      }
    }
    __syncthreads();

    // --- trailing SYRK: A22 -= P P^T, LDS-fed chunk pairs, lower triangle
    for (int jc = 0; jc < rem; jc += CHOL_CHUNK) {
      const int jrows = min(CHOL_CHUNK, rem - jc);
      for (int idx = tid; idx < jrows * CHOL_BS; idx += CHOL_TPB) {
        const int r = idx / CHOL_BS, c = idx % CHOL_BS;
        L.Pj[r][c] = M[(long long)(k0 + bs + jc + r) * N + k0 + c];
      }
      __syncthreads();
      for (int ic = jc; ic < rem; ic += CHOL_CHUNK) {
        const int irows = min(CHOL_CHUNK, rem - ic);
        const bool same = (ic == jc);
        if (!same) {
          for (int idx = tid; idx < irows * CHOL_BS; idx += CHOL_TPB) {
            const int r = idx / CHOL_BS, c = idx % CHOL_BS;
            L.Pi[r][c] = M[(long long)(k0 + bs + ic + r) * N + k0 + c];
          }
        }
        __syncthreads();
        const float(*PI)[CHOL_BS + 1] = same ? L.Pj : L.Pi;
        // 2x2 register tiles over (irows x jrows)
        const int ti = (irows + 1) / 2, tj = (jrows + 1) / 2;
        for (int t = tid; t < ti * tj; t += CHOL_TPB) {
          const int tir = t / tj, tjr = t % tj;
          const int i0 = tir * 2, j0 = tjr * 2;
          const int gi0 = k0 + bs + ic + i0;
          const int gj0 = k0 + bs + jc + j0;
          if (same && gj0 > gi0 + 1) continue;  // fully-upper tile
          float acc00 = 0.f, acc01 = 0.f, acc10 = 0.f, acc11 = 0.f;
          const int i1 = min(i0 + 1, irows - 1);
          const int j1 = min(j0 + 1, jrows - 1);
#pragma unroll 8
          for (int s = 0; s < CHOL_BS; ++s) {
            const float a0 = PI[i0][s], a1 = PI[i1][s];
            const float b0 = L.Pj[j0][s], b1 = L.Pj[j1][s];
            acc00 = fmaf(a0, b0, acc00);
            acc01 = fmaf(a0, b1, acc01);
            acc10 = fmaf(a1, b0, acc10);
            acc11 = fmaf(a1, b1, acc11);
          }
          const int gi1 = k0 + bs + ic + i1;
          const int gj1 = k0 + bs + jc + j1;
          if (gj0 <= gi0) M[(long long)gi0 * N + gj0] -= acc00;
          if (j1 != j0 && gj1 <= gi0) M[(long long)gi0 * N + gj1] -= acc01;
          if (i1 != i0 && gj0 <= gi1) M[(long long)gi1 * N + gj0] -= acc10;
          if (i1 != i0 && j1 != j0 && gj1 <= gi1) M[(long long)gi1 * N + gj1] -= acc11;
        }
        __syncthreads();
      }
    }
  }
  if (tid == 0) logdet[b] = L.ld_accum;
}

// ---------------------------------------------------------------------------
// Multi-launch right-looking factorization for SMALL batch counts.
//
// The single-block kernel above runs one block per matrix: at the headline
// shape (B ~ 12 systems of N=300 from the speculative SCE-UA stages) that
// leaves 244 of 256 CUs idle. Here each 32-column step is two stream-ordered
// launches — a panel kernel (grid B) and a trailing-update SYRK kernel
// (grid B x tile-pairs, 64x64 tiles) — so the SYRK, which is ~90% of the
// FLOPs, spreads across B * O((N/64)^2) workgroups.

#define CHOLP_TPB 256
#define SYRK_TS 64  // trailing-update tile edge

// Factor the 32x32 diagonal block at (k0,k0) and panel-solve the rows below
// it. One block per matrix; also accumulates the step's logdet contribution
// deterministically (no atomics: k-steps are stream-ordered).
// Device body shared by the per-step panel kernel and the persistent
// whole-factorization kernel (chol_persist_kernel): identical arithmetic,
// only the launch wrapper differs.
template <int GROUP_COLS, int TPB>
__device__ __forceinline__ void chol_panel_body(
    float* __restrict__ Ab, float* __restrict__ logdet_b,
    int* __restrict__ info_b, int N, int k0, float* __restrict__ rhs_b,
    float (*S)[CHOL_BS + 1], float (*colbuf4)[CHOL_BS],
    float (*P)[CHOL_BS + 1]) {
  // rhs != nullptr: FUSED FORWARD SOLVE (bordered-matrix scheme). rhs (B, N)
  // starts as y and finishes as z = L^-1 y without a separate TRSV kernel:
  // this panel solves entries [k0, k0+bs) against the factored diagonal
  // block (the rhs is one extra TRSM row), and the trailing SYRK's diagonal
  // tiles apply the rank-32 update to the remaining entries. Saves the
  // ~61 us serial forward_solve_batched launch per NMLL (same serial-chain
  // structure as the factorization it now rides on).
  const int tid = threadIdx.x;
  const int bs = min(CHOL_BS, N - k0);

  // Wave-parallel diagonal factor: lane i of the first wave owns row i of
  // the 32x32 block in registers; column j's pivot and multipliers move by
  // __shfl broadcast. Zero block barriers in the serial dependency chain
  // (the barrier version spent ~96 barriers here and dominated the whole
  // multi-launch path at 54 us per panel).
  // coalesced stage of the block into LDS by the whole workgroup first —
  // a single wave doing the 32x32 global load directly issues ~1000
  // uncoalesced (1200 B stride) loads and stalls on memory latency
  for (int idx = tid; idx < bs * bs; idx += blockDim.x)
    S[idx / bs][idx % bs] = Ab[(long long)(k0 + idx / bs) * N + k0 + idx % bs];
  __syncthreads();
  // TRSM staging buffer, declared up front: waves 1+ stage the FIRST chunk
  // of trailing panel rows into LDS concurrently with wave 0's serial
  // diagonal factor (the staging reads last step's SYRK output — global
  // memory untouched by the factor — so the ~34 KB coalesced load hides
  // entirely under the factor's shuffle chain instead of serializing after
  // it; disjoint LDS regions).
  const int c0_first = k0 + CHOL_BS;
  const int rows_first = min(TPB, N - c0_first);
  if (tid >= 64) {
    for (int idx = tid - 64; idx < rows_first * CHOL_BS; idx += TPB - 64) {
      const int r = idx / CHOL_BS, c = idx % CHOL_BS;
      P[r][c] = Ab[(long long)(c0_first + r) * N + k0 + c];
    }
  } else if (GROUP_COLS == 1) {
    // pure-shuffle factor (no LDS broadcast, no fences): per column j the
    // rank-1 update's multiplier L[c][j] moves by one __shfl per target
    // column — 31 independent shuffles that pipeline freely, where the
    // LDS-broadcast rounds serialize on two __threadfence_block each.
    // fmaf order per element is ascending-j rank-1, identical to the
    // grouped variant: the factor is bitwise unchanged.
    const int lane = tid;
    float r[CHOL_BS];
#pragma unroll
    for (int t = 0; t < CHOL_BS; ++t)
      r[t] = (lane < bs && t < bs) ? S[lane][t] : 0.0f;
    float mylog = 0.0f;
    int bad = 0;
#pragma unroll
    for (int j = 0; j < CHOL_BS; ++j) {
      if (j >= bs) break;
      float d = __shfl(r[j], j);
      if (d <= 0.0f || !isfinite(d)) {
        bad = bad ? bad : (k0 + j + 1);
        d = 1e-30f;
      }
      d = sqrtf(d);
      if (lane == j) {
        r[j] = d;
        mylog += logf(d);
      } else if (lane > j) {
        r[j] /= d;
      }
      const float lij = r[j];
#pragma unroll
      for (int c = j + 1; c < CHOL_BS; ++c) {
        if (c >= bs) break;
        const float lcj = __shfl(lij, c);
        if (lane >= c) r[c] = fmaf(-lij, lcj, r[c]);
      }
    }
    if (lane < bs) {
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) {
        if (t >= bs) continue;
        S[lane][t] = r[t];
      }
    }
    for (int off = 32; off > 0; off >>= 1) {
      mylog += __shfl_down(mylog, off);
      const int ob = __shfl_down(bad, off);
      bad = bad ? bad : ob;
    }
    if (lane == 0) {
      *logdet_b += mylog;
      if (bad && *info_b == 0) *info_b = bad;
    }
  } else {
    const int lane = tid;
    float r[CHOL_BS];
#pragma unroll
    for (int t = 0; t < CHOL_BS; ++t)
      r[t] = (lane < bs && t < bs) ? S[lane][t] : 0.0f;
    float mylog = 0.0f;
    int bad = 0;
    // GROUP_COLS columns per round: within the group each column first
    // receives the updates of its in-group predecessors, is factored
    // (shfl pivot + sqrt + div), and broadcast through LDS with ONE
    // workgroup fence; then ONE fused rank-GROUP pass updates the
    // remaining columns. Quarters the update-loop rounds on the serial
    // pivot chain (the dominant latency of this kernel) while keeping the
    // fmaf sequence PER ELEMENT identical to the one-column version —
    // the factor is bitwise unchanged (SCE-UA accept decisions are
    // bit-stable). Measured A/B in profiles/README.md.
#pragma unroll
    for (int g = 0; g < CHOL_BS; g += GROUP_COLS) {
      if (g >= bs) continue;
#pragma unroll
      for (int q = 0; q < GROUP_COLS; ++q) {
        const int j = g + q;
        if (j >= bs) break;
        // in-group predecessor updates for column j (same order as the
        // sequential rank-1 passes)
#pragma unroll
        for (int p = 0; p < GROUP_COLS; ++p) {
          if (p >= q) break;
          if (lane >= j) r[j] = fmaf(-r[g + p], colbuf4[p][j], r[j]);
        }
        float d = __shfl(r[j], j);
        if (d <= 0.0f || !isfinite(d)) {
          bad = bad ? bad : (k0 + j + 1);
          d = 1e-30f;
        }
        d = sqrtf(d);
        if (lane == j) {
          r[j] = d;
          mylog += logf(d);
        } else if (lane > j) {
          r[j] /= d;
        }
        // broadcast column j through LDS with ONE workgroup fence: the
        // reads then pipeline freely, unlike a per-t __shfl chain
        // (ds_bpermute each) or a volatile pointer (per-access ordering)
        if (lane < bs) colbuf4[q][lane] = r[j];
        __threadfence_block();
      }
      // fused rank-GROUP update of the columns beyond the group
      const int gend = min(g + GROUP_COLS, bs);
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) {
        if (t < g + GROUP_COLS || t >= bs) continue;
        if (lane >= t) {
#pragma unroll
          for (int q = 0; q < GROUP_COLS; ++q) {
            if (g + q >= gend) break;
            r[t] = fmaf(-r[g + q], colbuf4[q][t], r[t]);
          }
        }
      }
    }
    // stage the factored block back to LDS (global writeback below is
    // done coalesced by the whole workgroup)
    if (lane < bs) {
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) {
        if (t >= bs) continue;
        S[lane][t] = r[t];
      }
    }
    // wave-reduce the logdet contribution and error flag
    for (int off = 32; off > 0; off >>= 1) {
      mylog += __shfl_down(mylog, off);
      const int ob = __shfl_down(bad, off);
      bad = bad ? bad : ob;
    }
    if (lane == 0) {
      *logdet_b += mylog;
      if (bad && *info_b == 0) *info_b = bad;
    }
  }
  __syncthreads();
  // fused-solve panel step: wave 0 solves the rhs segment against the
  // factored diagonal block (wave-synchronous, lane j owns entry j; same
  // shuffle scheme as forward_solve_batched_kernel's diagonal solve). The
  // segment already carries the trailing updates of all previous steps.
  if (rhs_b != nullptr && tid < 64) {
    float* yb = rhs_b + k0;
    const int j = tid;
    float v = (j < bs) ? yb[j] : 0.0f;
    for (int t = 0; t < bs; ++t) {
      const float zt = __shfl(v, t) / S[t][t];
      if (j == t) v = zt;
      else if (j > t && j < bs) v = fmaf(-S[j][t], zt, v);
    }
    if (j < bs) yb[j] = v;
  }
  // factored-block writeback by waves 1+ (concurrent with wave 0's rhs
  // solve above; both only READ S)
  for (int idx = tid - 64; idx >= 0 && idx < bs * bs; idx += TPB - 64) {
    const int i = idx / bs, t = idx % bs;
    if (t <= i) Ab[(long long)(k0 + i) * N + k0 + t] = S[i][t];
  }
  // no barrier: the first chunk's P and the factored S were published by
  // the post-factor barrier; the writes above touch disjoint addresses
  // panel solve, chunked through LDS: per-thread direct row loads are
  // uncoalesced (64 lanes touch 64 cache lines per load instruction, a
  // constant ~27 us regardless of row count), so each chunk is
  // staged with a COALESCED copy, solved entirely in LDS (one row per
  // thread; the +1-padded stride keeps lanes on distinct banks), and
  // written back coalesced. The FIRST chunk was pre-staged during the
  // factor. Trailing rows exist only under a FULL panel
  // (k0 + bs < N implies bs == CHOL_BS): constant trip counts throughout.
  for (int c0 = c0_first; c0 < N; c0 += TPB) {
    const int rows = min(TPB, N - c0);
    if (c0 != c0_first) {
      for (int idx = tid; idx < rows * CHOL_BS; idx += TPB) {
        const int r = idx / CHOL_BS, c = idx % CHOL_BS;
        P[r][c] = Ab[(long long)(c0 + r) * N + k0 + c];
      }
      __syncthreads();
    }
    if (tid < rows) {
      // own row promoted to registers: the j-loop's 512-FMA dependency
      // chain then reads only registers and broadcast S rows (the LDS
      // version serializes a ds_read into every fmaf of the chain);
      // identical fmaf order — the solve is bitwise unchanged
      float row[CHOL_BS];
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) row[t] = P[tid][t];
#pragma unroll
      for (int j = 0; j < CHOL_BS; ++j) {
        float v = row[j];
#pragma unroll
        for (int t = 0; t < j; ++t) v = fmaf(-row[t], S[j][t], v);
        row[j] = v / S[j][j];
      }
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) P[tid][t] = row[t];
    }
    __syncthreads();
    for (int idx = tid; idx < rows * CHOL_BS; idx += TPB) {
      const int r = idx / CHOL_BS, c = idx % CHOL_BS;
      Ab[(long long)(c0 + r) * N + k0 + c] = P[r][c];
    }
    __syncthreads();
  }
}


template <int GROUP_COLS, int TPB = CHOLP_TPB>
__global__ __launch_bounds__(TPB) void chol_panel_kernel(
    float* __restrict__ A, float* __restrict__ logdet, int* __restrict__ info,
    int N, int k0, float* __restrict__ rhs) {
  __shared__ float S[CHOL_BS][CHOL_BS + 1];
  __shared__ float colbuf4[GROUP_COLS][CHOL_BS];
  __shared__ float P[TPB][CHOL_BS + 1];
  const int b = blockIdx.x;
  chol_panel_body<GROUP_COLS, TPB>(
      A + (long long)b * N * N, logdet + b, info + b, N, k0,
      rhs ? rhs + (long long)b * N : nullptr, S, colbuf4, P);
}

// ------------------------------------------------------------ 64-wide panel
// PANEL64 variant: one launch factors a 64x64 diagonal block (two wave-
// parallel 32-factors + an in-LDS TRSM + SYRK between them) and solves the
// whole 64-column panel below it. Halves the panel-launch count of the
// right-looking loop (N=300: 5 launches instead of 10) — the multik path
// is launch-gap bound at ~10 us fixed cost per launch (profiles/README.md).
#define PANEL64 64

// Factor the bs_ x bs_ (<=32) diagonal sub-block of S at offset (o, o):
// wave-parallel, lane i owns row i in registers (same scheme as
// chol_panel_kernel's diagonal factor). Caller must __syncthreads() after.
__device__ __forceinline__ void factor32_at(
    float S[PANEL64][PANEL64 + 1], float* colbuf, int o, int bs_,
    float* logdet, int* info, int b, int gk0, int tid) {
  if (tid < 64) {
    const int lane = tid;
    float r[CHOL_BS];
#pragma unroll
    for (int t = 0; t < CHOL_BS; ++t)
      r[t] = (lane < bs_ && t < bs_) ? S[o + lane][o + t] : 0.0f;
    float mylog = 0.0f;
    int bad = 0;
#pragma unroll
    for (int j = 0; j < CHOL_BS; ++j) {
      if (j >= bs_) continue;
      float d = __shfl(r[j], j);
      if (d <= 0.0f || !isfinite(d)) {
        bad = bad ? bad : (gk0 + j + 1);
        d = 1e-30f;
      }
      d = sqrtf(d);
      if (lane == j) {
        r[j] = d;
        mylog = logf(d);
      } else if (lane > j) {
        r[j] /= d;
      }
      if (lane < bs_) colbuf[lane] = r[j];
      __threadfence_block();
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) {
        if (t <= j || t >= bs_) continue;
        if (lane >= t) r[t] = fmaf(-r[j], colbuf[t], r[t]);
      }
    }
    if (lane < bs_) {
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t) {
        if (t >= bs_) continue;
        S[o + lane][o + t] = r[t];
      }
    }
    for (int off = 32; off > 0; off >>= 1) {
      mylog += __shfl_down(mylog, off);
      const int ob = __shfl_down(bad, off);
      bad = bad ? bad : ob;
    }
    if (lane == 0) {
      logdet[b] += mylog;
      if (bad && info[b] == 0) info[b] = bad;
    }
  }
}

#define P64_CHUNK 128  // panel rows staged per LDS buffer

__global__ __launch_bounds__(CHOLP_TPB) void chol_panel64_kernel(
    float* __restrict__ A, float* __restrict__ logdet, int* __restrict__ info,
    int N, int k0) {
  __shared__ float S[PANEL64][PANEL64 + 1];
  __shared__ float colbuf[CHOL_BS];
  __shared__ float P[P64_CHUNK][PANEL64 + 1];
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  float* Ab = A + (long long)b * N * N;
  const int bs = min(PANEL64, N - k0);
  const int bs1 = min(CHOL_BS, bs);
  const int bs2 = bs - bs1;

  for (int idx = tid; idx < bs * bs; idx += blockDim.x)
    S[idx / bs][idx % bs] = Ab[(long long)(k0 + idx / bs) * N + k0 + idx % bs];
  __syncthreads();

  factor32_at(S, colbuf, 0, bs1, logdet, info, b, k0, tid);
  __syncthreads();

  if (bs2 > 0) {
    // TRSM: rows [32, bs) of columns [0, 32) against the factored block
    if (tid < bs2) {
      const int r = CHOL_BS + tid;
#pragma unroll 4
      for (int j = 0; j < CHOL_BS; ++j) {
        float v = S[r][j];
        for (int t = 0; t < j; ++t) v = fmaf(-S[r][t], S[j][t], v);
        S[r][j] = v / S[j][j];
      }
    }
    __syncthreads();
    // SYRK the lower-right bs2 x bs2 block (lower triangle only: the
    // factor routine never reads above the diagonal)
    for (int idx = tid; idx < bs2 * bs2; idx += blockDim.x) {
      const int i = idx / bs2, j = idx % bs2;
      if (j > i) continue;
      float acc = 0.0f;
#pragma unroll 8
      for (int k = 0; k < CHOL_BS; ++k)
        acc = fmaf(S[CHOL_BS + i][k], S[CHOL_BS + j][k], acc);
      S[CHOL_BS + i][CHOL_BS + j] -= acc;
    }
    __syncthreads();
    factor32_at(S, colbuf, CHOL_BS, bs2, logdet, info, b, k0 + CHOL_BS, tid);
    __syncthreads();
  }

  for (int idx = tid; idx < bs * bs; idx += blockDim.x) {
    const int i = idx / bs, t = idx % bs;
    if (t <= i) Ab[(long long)(k0 + i) * N + k0 + t] = S[i][t];
  }

  // 64-column panel solve below, staged through LDS in coalesced chunks
  for (int c0 = k0 + PANEL64; c0 < N; c0 += P64_CHUNK) {
    const int rows = min(P64_CHUNK, N - c0);
    for (int idx = tid; idx < rows * PANEL64; idx += CHOLP_TPB) {
      const int r = idx / PANEL64, c = idx % PANEL64;
      P[r][c] = Ab[(long long)(c0 + r) * N + k0 + c];
    }
    __syncthreads();
    if (tid < rows) {
#pragma unroll 4
      for (int j = 0; j < PANEL64; ++j) {
        float v = P[tid][j];
        for (int t = 0; t < j; ++t) v = fmaf(-P[tid][t], S[j][t], v);
        P[tid][j] = v / S[j][j];
      }
    }
    __syncthreads();
    for (int idx = tid; idx < rows * PANEL64; idx += CHOLP_TPB) {
      const int r = idx / PANEL64, c = idx % PANEL64;
      Ab[(long long)(c0 + r) * N + k0 + c] = P[r][c];
    }
    __syncthreads();
  }
}

// 64-deep trailing update: C -= Pi Pj^T with K = 64 (the PANEL64 step).
// Same tile-pair enumeration as chol_syrk_kernel; k-loop runs the full 64
// panel columns, so HALF the launches touch the same trailing bytes.
__global__ __launch_bounds__(CHOLP_TPB) void chol_syrk64_kernel(
    float* __restrict__ A, int N, int k0, int nt, int tj_fixed, int off) {
  __shared__ float Pi[SYRK_TS][PANEL64];
  __shared__ float Pj[SYRK_TS][PANEL64];
  const int b = blockIdx.x;
  float* Ab = A + (long long)b * N * N;
  const int r0 = k0 + PANEL64;
  int ti, tj;
  if (tj_fixed >= 0) {
    tj = tj_fixed;
    ti = blockIdx.y + tj_fixed;
  } else {
    int p = blockIdx.y;
    ti = 0;
    while (p > ti) { p -= ti + 1; ++ti; }
    tj = p + off;
    ti += off;
  }
  const int i0 = r0 + ti * SYRK_TS, j0 = r0 + tj * SYRK_TS;
  const int tid = threadIdx.x;

  for (int idx = tid; idx < SYRK_TS * PANEL64; idx += blockDim.x) {
    const int r = idx / PANEL64, c = idx % PANEL64;
    const int cs = c ^ ((r & 7) << 2);  // bank-spread swizzle
    Pi[r][cs] = (i0 + r < N) ? Ab[(long long)(i0 + r) * N + k0 + c] : 0.0f;
    Pj[r][cs] = (j0 + r < N) ? Ab[(long long)(j0 + r) * N + k0 + c] : 0.0f;
  }
  __syncthreads();

  typedef __attribute__((ext_vector_type(4))) float f32x4;
  const int wave = tid >> 6, lane = tid & 63;
  const int lr = lane & 15, lk = lane >> 4;
#pragma unroll
  for (int sIdx = 0; sIdx < 4; ++sIdx) {
    const int sub = wave * 4 + sIdx;
    const int r16 = (sub >> 2) * 16, c16 = (sub & 3) * 16;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int k = 0; k < PANEL64; k += 4) {
      const int ra = r16 + lr, rb = c16 + lr;
      const float a = Pi[ra][(k + lk) ^ ((ra & 7) << 2)];
      const float bv = Pj[rb][(k + lk) ^ ((rb & 7) << 2)];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int i = i0 + r16 + lk * 4 + r;
      const int j = j0 + c16 + lr;
      if (i < N && j < N && j <= i) Ab[(long long)i * N + j] -= acc[r];
    }
  }
}

// Trailing update A[ti,tj] -= P_i P_j^T over 64x64 tiles of the submatrix
// below/right of the panel; blockIdx.y enumerates lower-triangular tile
// pairs, each 256-thread block computes a 4x4 register tile per thread.
__device__ __forceinline__ void chol_syrk_tile_body(
    float* __restrict__ Ab, int N, int k0, int ti, int tj,
    float* __restrict__ rhs_b, float (*Pi)[CHOL_BS], float (*Pj)[CHOL_BS]) {
  const int r0 = k0 + CHOL_BS;  // first trailing row
  const int i0 = r0 + ti * SYRK_TS, j0 = r0 + tj * SYRK_TS;
  const int tid = threadIdx.x;

  for (int idx = tid; idx < SYRK_TS * CHOL_BS; idx += blockDim.x) {
    const int r = idx / CHOL_BS, c = idx % CHOL_BS;
    const int cs = c ^ ((r & 7) << 2);  // swizzled physical column
    Pi[r][cs] = (i0 + r < N) ? Ab[(long long)(i0 + r) * N + k0 + c] : 0.0f;
    Pj[r][cs] = (j0 + r < N) ? Ab[(long long)(j0 + r) * N + k0 + c] : 0.0f;
  }
  __syncthreads();

  // MFMA tile core: C -= Pi * Pj^T on the matrix units (operand map and
  // rationale in the original kernel comment below).
  typedef __attribute__((ext_vector_type(4))) float f32x4;
  const int wave = tid >> 6, lane = tid & 63;
  const int lr = lane & 15, lk = lane >> 4;
#pragma unroll
  for (int sIdx = 0; sIdx < 4; ++sIdx) {
    const int sub = wave * 4 + sIdx;        // 4x4 grid of 16x16 subtiles
    if (sub >= 16) break;  // blocks wider than 4 waves: extra waves idle
    const int r16 = (sub >> 2) * 16, c16 = (sub & 3) * 16;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int k = 0; k < CHOL_BS; k += 4) {
      const int ra = r16 + lr, rb = c16 + lr;
      const float a = Pi[ra][(k + lk) ^ ((ra & 7) << 2)];  // A[i][k]
      const float b = Pj[rb][(k + lk) ^ ((rb & 7) << 2)];  // B[k][j]=Pj[j][k]
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int i = i0 + r16 + lk * 4 + r;
      const int j = j0 + c16 + lr;
      if (i < N && j < N && j <= i) Ab[(long long)i * N + j] -= acc[r];
    }
  }
  // fused-solve trailing update (diagonal tiles only); see original
  // kernel comment.
  if (rhs_b != nullptr && ti == tj && tid < 64) {
    const int i = i0 + tid;
    if (i < N) {
      const float* z = rhs_b + k0;
      float acc = rhs_b[i];
#pragma unroll
      for (int t = 0; t < CHOL_BS; ++t)
        acc = fmaf(-Pi[tid][t ^ ((tid & 7) << 2)], z[t], acc);
      rhs_b[i] = acc;
    }
  }
}

__global__ __launch_bounds__(CHOLP_TPB) void chol_syrk_kernel(
    float* __restrict__ A, int N, int k0, int nt, int tj_fixed, int off,
    float* __restrict__ rhs) {
  // stride 32 with an XOR swizzle on the k-column: the MFMA operand read
  // (lane -> [row + lr][k + lk]) hits banks (row+lk) mod 32 under a +1 pad,
  // an up-to-8-way conflict (PMC: 2.2 extra cycles per LDS instruction);
  // c = k ^ ((row & 7) << 2) spreads (lr, lk) over all 32 banks.
  // v_mfma_f32_16x16x4_f32 is EXACT fp32 (identical to an fmaf chain) at
  // the f32 vector rate; operand map (cdna4_isa section 10):
  // A: lane l -> A[l&15][l>>4]; B: lane l -> B[l>>4][l&15];
  // C/D (f32x4): col = lane&15, row = (lane>>4)*4 + reg.
  __shared__ float Pi[SYRK_TS][CHOL_BS];
  __shared__ float Pj[SYRK_TS][CHOL_BS];
  const int b = blockIdx.x;
  float* Ab = A + (long long)b * N * N;
  // tile-pair selection: tj_fixed >= 0 enumerates one tile COLUMN
  // (ti = blockIdx.y + tj_fixed); otherwise blockIdx.y walks the lower
  // triangle, shifted by `off` columns/rows (used to split the first tile
  // column from the rest for cross-stream overlap)
  int ti, tj;
  if (tj_fixed >= 0) {
    tj = tj_fixed;
    ti = blockIdx.y + tj_fixed;
  } else {
    int p = blockIdx.y;
    ti = 0;
    while (p > ti) { p -= ti + 1; ++ti; }
    tj = p + off;
    ti += off;
  }
  chol_syrk_tile_body(Ab, N, k0, ti, tj,
                      rhs ? rhs + (long long)b * N : nullptr, Pi, Pj);
}

// Blocked forward substitution: solve L z = y for R right-hand sides.
// One block per (batch, rhs). Per 32-column panel: wave 0 solves the
// 32x32 diagonal block wave-synchronously (lane j owns row j, solved
// entries broadcast by shuffle), then all threads apply the rank-32
// update in parallel — 2 barriers per panel instead of 2 per column (the
// former per-column loop spent ~600 barriers on N=300).
#define TRSV_BS 32

__global__ void forward_solve_batched_kernel(const float* __restrict__ L,
                                             float* __restrict__ Y,  // (B,N,R) inout
                                             int N, int R) {
  __shared__ float S[TRSV_BS][TRSV_BS + 1];
  __shared__ float z[TRSV_BS];
  __shared__ float yp[TRSV_BS];
  const int b = blockIdx.x;
  const int r = blockIdx.y;
  if (r >= R) return;
  const float* Lb = L + (long long)b * N * N;
  float* y = Y + (long long)b * N * R;
  const int tid = threadIdx.x;
  for (int k0 = 0; k0 < N; k0 += TRSV_BS) {
    const int bs = min(TRSV_BS, N - k0);
    // stage panel of L and the panel's rhs entries into LDS in parallel
    for (int idx = tid; idx < bs * bs; idx += blockDim.x)
      S[idx / bs][idx % bs] = Lb[(long long)(k0 + idx / bs) * N + k0 + idx % bs];
    if (tid < bs) yp[tid] = y[(k0 + tid) * R + r];
    __syncthreads();
    // wave-parallel diagonal solve: lane j owns panel row j; per column t
    // the solved z_t is broadcast with __shfl — no block barrier, no
    // global traffic inside the serial dependency chain.
    if (tid < 64) {
      const int j = tid;  // lanes >= bs just follow along
      float v = (j < bs) ? yp[j] : 0.0f;
      for (int t = 0; t < bs; ++t) {
        const float zt = __shfl(v, t) / S[t][t];
        if (j == t) {
          v = zt;
          z[t] = zt;
        } else if (j > t && j < bs) {
          v = fmaf(-S[j][t], zt, v);
        }
      }
      if (j < bs) y[(k0 + j) * R + r] = v;
    }
    __syncthreads();
    // rank-32 update of the remaining rows: rows below exist only under a
    // FULL panel, so the trip count is compile-time (unrolled loads)
    for (int i = k0 + TRSV_BS + tid; i < N; i += blockDim.x) {
      float acc = y[i * R + r];
      const float* row = Lb + (long long)i * N + k0;
#pragma unroll
      for (int t = 0; t < TRSV_BS; ++t) acc = fmaf(-row[t], z[t], acc);
      y[i * R + r] = acc;
    }
    __syncthreads();
  }
}

// Blocked backward substitution: solve L^T x = z (same structure, panels
// walked from the bottom; the update reads row segments of L, coalesced).
__global__ void backward_solve_batched_kernel(const float* __restrict__ L,
                                              float* __restrict__ Y,  // (B,N,R)
                                              int N, int R) {
  __shared__ float S[TRSV_BS][TRSV_BS + 1];
  __shared__ float z[TRSV_BS];
  __shared__ float yp[TRSV_BS];
  const int b = blockIdx.x;
  const int r = blockIdx.y;
  if (r >= R) return;
  const float* Lb = L + (long long)b * N * N;
  float* y = Y + (long long)b * N * R;
  const int tid = threadIdx.x;
  const int first = ((N - 1) / TRSV_BS) * TRSV_BS;
  for (int k0 = first; k0 >= 0; k0 -= TRSV_BS) {
    const int bs = min(TRSV_BS, N - k0);
    for (int idx = tid; idx < bs * bs; idx += blockDim.x)
      S[idx / bs][idx % bs] = Lb[(long long)(k0 + idx / bs) * N + k0 + idx % bs];
    if (tid < bs) yp[tid] = y[(k0 + tid) * R + r];
    __syncthreads();
    // wave-parallel diagonal solve against S^T, columns walked bottom-up
    if (tid < 64) {
      const int j = tid;
      float v = (j < bs) ? yp[j] : 0.0f;
      for (int t = bs - 1; t >= 0; --t) {
        const float zt = __shfl(v, t) / S[t][t];
        if (j == t) {
          v = zt;
          z[t] = zt;
        } else if (j < t) {
          v = fmaf(-S[t][j], zt, v);
        }
      }
      if (j < bs) y[(k0 + j) * R + r] = v;
    }
    __syncthreads();
    // update rows above the panel: x_i -= sum_t L[k0+t][i] * z[t]
    // (specialized full-panel path so the column loads unroll)
    if (bs == TRSV_BS) {
      for (int i = tid; i < k0; i += blockDim.x) {
        float acc = y[i * R + r];
#pragma unroll
        for (int t = 0; t < TRSV_BS; ++t)
          acc = fmaf(-Lb[(long long)(k0 + t) * N + i], z[t], acc);
        y[i * R + r] = acc;
      }
    } else {
      for (int i = tid; i < k0; i += blockDim.x) {
        float acc = y[i * R + r];
        for (int t = 0; t < bs; ++t)
          acc = fmaf(-Lb[(long long)(k0 + t) * N + i], z[t], acc);
        y[i * R + r] = acc;
      }
    }
    __syncthreads();
  }
}

#include <stdlib.h>

// Final NMLL assembly: out[b] = 0.5 * sum(Z_b^2) + half_logdet[b] + c,
// inf where the factorization failed or the value is non-finite — fuses the
// (z*z).sum + scalar-combine + masking chain (a handful of torch dispatches
// per SCE-UA stage) into one launch.
__global__ void nmll_reduce_kernel(const float* __restrict__ Z,
                                   const float* __restrict__ half_logdet,
                                   const int* __restrict__ info,
                                   float* __restrict__ out, int N, float c) {
  __shared__ float part[256];
  const int b = blockIdx.x;
  const float* z = Z + (long long)b * N;
  float acc = 0.f;
  for (int i = threadIdx.x; i < N; i += blockDim.x) acc += z[i] * z[i];
  part[threadIdx.x] = acc;
  __syncthreads();
  for (int off = blockDim.x >> 1; off > 0; off >>= 1) {
    if (threadIdx.x < off) part[threadIdx.x] += part[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    float v = 0.5f * part[0] + half_logdet[b] + c;
    if (info[b] != 0 || !isfinite(v)) v = INFINITY;
    out[b] = v;
  }
}

extern "C" void launch_nmll_reduce(const float* Z, const float* half_logdet,
                                   const int* info, float* out, int B, int N,
                                   float c, hipStream_t stream) {
  hipLaunchKernelGGL(nmll_reduce_kernel, dim3(B), dim3(256), 0, stream, Z,
                     half_logdet, info, out, N, c);
}

__global__ void zero_i32_kernel(int* __restrict__ p, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0;
}

__global__ void zero_f32_kernel(float* __restrict__ p, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0.0f;
}

// panel-group-size selection: 2- vs 4-column factor rounds, same-box A/B
// via DMOSOPT_CHOL_GROUP (bitwise-identical results either way)
static inline int panel_group_cols() {
  static int g = -1;
  if (g < 0) {
    const char* e = getenv("DMOSOPT_CHOL_GROUP");
    // 2 (default) / 4 = grouped LDS-broadcast factor rounds; 1 = pure-
    // shuffle factor, measured 2.5x SLOWER (804 vs 324 us per B=12 chol,
    // same box): __shfl is ds_bpermute on CDNA4 — a full LDS round trip
    // per value — so 512 dependent shuffles lose to 16 fence-paired LDS
    // column broadcasts. Kept selectable for re-measurement.
    g = e ? atoi(e) : 2;
    if (g != 1 && g != 4) g = 2;
  }
  return g;
}
// panel workgroup width: 384 threads solve the whole N=300 trailing panel
// in ONE LDS chunk (256 needs two sequential chunk rounds: stage + solve +
// writeback + 3 barriers each); static LDS caps the width at 384
// (P[384][33] + S + colbuf = 55 KB < 64 KB). Same-box A/B via
// DMOSOPT_CHOL_TPB.
static inline int panel_tpb() {
  static int t = -1;
  if (t < 0) {
    const char* e = getenv("DMOSOPT_CHOL_TPB");
    t = (e && atoi(e) == 256) ? 256 : 384;
  }
  return t;
}
#define LAUNCH_PANEL(B_, stream_, A_, logdet_, info_, N_, k0_, rhs_)                 do {                                                                           const int tpb_ = panel_tpb();                                                if (panel_group_cols() == 1) {                                                 if (tpb_ == 384)                                                               hipLaunchKernelGGL((chol_panel_kernel<1, 384>), dim3(B_), dim3(384),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);             else                                                                           hipLaunchKernelGGL((chol_panel_kernel<1, 256>), dim3(B_), dim3(256),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);           } else if (panel_group_cols() == 2) {                                                 if (tpb_ == 384)                                                               hipLaunchKernelGGL((chol_panel_kernel<2, 384>), dim3(B_), dim3(384),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);             else                                                                           hipLaunchKernelGGL((chol_panel_kernel<2, 256>), dim3(B_), dim3(256),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);           } else {                                                                       if (tpb_ == 384)                                                               hipLaunchKernelGGL((chol_panel_kernel<4, 384>), dim3(B_), dim3(384),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);             else                                                                           hipLaunchKernelGGL((chol_panel_kernel<4, 256>), dim3(B_), dim3(256),                            0, stream_, A_, logdet_, info_, N_, k0_, rhs_);           }                                                                          } while (0)

extern "C" void launch_cholesky_multik(float* A, float* logdet, int* info,
                                       float* rhs, int B, int N,
                                       hipStream_t stream) {
  // Cross-stream software pipeline: panel(k+1) only depends on the FIRST
  // tile column of step k's trailing update (its 64 columns cover the next
  // panel), so the remaining tile pairs run on a side stream concurrently
  // with the next panel. Regions are disjoint: tj0(k) writes cols
  // [k0+32, k0+95]; rest(k) writes cols >= k0+96; panel(k+1) writes cols
  // [k0+32, k0+63] AFTER tj0(k) on the main stream.
  static int overlap = -1;
  static hipStream_t s2 = nullptr;
  static hipEvent_t evP[2], evR[2];
  if (overlap < 0) {
    // measured NEGATIVE at B=12..192 / N=300 (477 vs 386 us per call):
    // the cross-stream event waits cost more than the ~90 us of SYRK they
    // hide. Kept opt-in for larger shapes: DMOSOPT_CHOL_OVERLAP=1.
    const char* env = getenv("DMOSOPT_CHOL_OVERLAP");
    overlap = (env && env[0] == '1') ? 1 : 0;
    if (overlap) {
      hipStreamCreateWithFlags(&s2, hipStreamNonBlocking);
      for (int i = 0; i < 2; ++i) {
        hipEventCreateWithFlags(&evP[i], hipEventDisableTiming);
        hipEventCreateWithFlags(&evR[i], hipEventDisableTiming);
      }
    }
  }
  // explicit zero KERNEL instead of hipMemsetAsync: under hipGraph stream
  // capture (models/gp_core.py _NmllGraph) the captured memset node was
  // observed to race with the first panel kernel's logdet accumulation on
  // replay (bit-nondeterministic outputs with bit-identical inputs,
  // scripts_det_debug6.py); a kernel node orders correctly
  hipLaunchKernelGGL(zero_f32_kernel, dim3((B + 255) / 256), dim3(256), 0,
                     stream, logdet, B);
  static int panel64 = -1;
  if (panel64 < 0) {
    // measured WORSE than PANEL32 across the board (profiles/README.md:
    // 0.89 vs 0.41 ms at B=12 N=300; 7x at N=4096): the in-kernel
    // TRSM/SYRK serial phases and the 64-deep per-row solve chain cost far
    // more than the ~10 us/launch the halved launch count saves. Kept as
    // DMOSOPT_CHOL_PANEL=64 for re-measurement on future ROCm/driver.
    const char* env = getenv("DMOSOPT_CHOL_PANEL");
    panel64 = (env && env[0] == '6') ? 1 : 0;
  }
  if (panel64 && !overlap) {
    for (int k0 = 0; k0 < N; k0 += PANEL64) {
      hipLaunchKernelGGL(chol_panel64_kernel, dim3(B), dim3(CHOLP_TPB), 0,
                         stream, A, logdet, info, N, k0);
      const int trailing = N - k0 - PANEL64;
      if (trailing > 0) {
        const int nt = (trailing + SYRK_TS - 1) / SYRK_TS;
        hipLaunchKernelGGL(chol_syrk64_kernel, dim3(B, nt * (nt + 1) / 2),
                           dim3(CHOLP_TPB), 0, stream, A, N, k0, nt, -1, 0);
      }
    }
    return;
  }
  if (!overlap) {
    for (int k0 = 0; k0 < N; k0 += CHOL_BS) {
      LAUNCH_PANEL(B, stream, A, logdet, info, N, k0, rhs);
      const int trailing = N - k0 - CHOL_BS;
      if (trailing > 0) {
        const int nt = (trailing + SYRK_TS - 1) / SYRK_TS;
        hipLaunchKernelGGL(chol_syrk_kernel, dim3(B, nt * (nt + 1) / 2),
                           dim3(CHOLP_TPB), 0, stream, A, N, k0, nt, -1, 0,
                           rhs);
      }
    }
    return;
  }
  int kstep = 0;
  bool have_rest_prev = false;
  // the overlap path does not carry a fused rhs (launch_cholesky_fused_solve
  // refuses it): the tj0/rest split would double-apply the diagonal tiles
  for (int k0 = 0; k0 < N; k0 += CHOL_BS, ++kstep) {
    LAUNCH_PANEL(B, stream, A, logdet, info, N, k0, nullptr);
    const int trailing = N - k0 - CHOL_BS;
    if (trailing <= 0) continue;
    const int nt = (trailing + SYRK_TS - 1) / SYRK_TS;
    const int slot = kstep & 1;
    if (nt > 1) hipEventRecord(evP[slot], stream);
    // tj0(k) waits for rest(k-1): their column regions overlap
    if (have_rest_prev) hipStreamWaitEvent(stream, evR[slot ^ 1], 0);
    hipLaunchKernelGGL(chol_syrk_kernel, dim3(B, nt), dim3(CHOLP_TPB), 0,
                       stream, A, N, k0, nt, 0, 0, nullptr);
    if (nt > 1) {
      const int rest_pairs = (nt - 1) * nt / 2;
      hipStreamWaitEvent(s2, evP[slot], 0);
      hipLaunchKernelGGL(chol_syrk_kernel, dim3(B, rest_pairs),
                         dim3(CHOLP_TPB), 0, s2, A, N, k0, nt, -1, 1,
                         nullptr);
      hipEventRecord(evR[slot], s2);
      have_rest_prev = true;
    } else {
      have_rest_prev = false;
    }
  }
  // rejoin the side stream before anything downstream reads the factor
  if (have_rest_prev) hipStreamWaitEvent(stream, evR[(kstep - 1) & 1], 0);
}



// ------------------------------------------------ persistent whole-chol
// ONE launch factorizes all B matrices: grid (B, 1 + T). Block (b, 0) is
// matrix b's PANEL block — it loops the right-looking steps, running the
// shared chol_panel_body per step; blocks (b, 1..T) are its SYRK TILE
// blocks. Steps hand off through two per-matrix device-scope flags
// (panel_flag counts completed panels, tiles_done counts completed tiles
// cumulatively — monotone counters, so no reset races):
//   tiles of step s spin until panel_flag >= s+1, then update their
//   64x64 trailing tile; the panel of step s+1 spins until tiles_done ==
//   cumulative pairs through step s. Replaces 2*ceil(N/32)-1 dispatches
//   (each ~5-8 us of entry/exit on this stack even inside a hipGraph)
//   with intra-kernel handoffs. DEADLOCK SAFETY: the launcher verifies
//   full-grid residency via the occupancy API (all blocks co-resident by
//   construction), and both spin loops are BOUNDED — on timeout they
//   stamp info[b] and exit instead of hanging the device.
#define PERSIST_TPB 384

static __device__ __forceinline__ int persist_pairs(int N, int k0) {
  const int trailing = N - k0 - CHOL_BS;
  if (trailing <= 0) return 0;
  const int nt = (trailing + SYRK_TS - 1) / SYRK_TS;
  return nt * (nt + 1) / 2;
}

__global__ __launch_bounds__(PERSIST_TPB) void chol_persist_kernel(
    float* __restrict__ A, float* __restrict__ logdet, int* __restrict__ info,
    float* __restrict__ rhs, int* __restrict__ ws, int N) {
  extern __shared__ char smem[];
  const int b = blockIdx.x;
  const int role = blockIdx.y;  // 0 = panel block, >=1 = tile block role-1
  const int tid = threadIdx.x;
  float* Ab = A + (long long)b * N * N;
  float* rhs_b = rhs ? rhs + (long long)b * N : nullptr;
  int* panel_flag = ws + 2 * b;
  int* tiles_done = ws + 2 * b + 1;

  if (role == 0) {
    float(*S)[CHOL_BS + 1] = (float(*)[CHOL_BS + 1])smem;
    float(*colbuf)[CHOL_BS] =
        (float(*)[CHOL_BS])(smem + sizeof(float) * CHOL_BS * (CHOL_BS + 1));
    float(*P)[CHOL_BS + 1] =
        (float(*)[CHOL_BS + 1])(smem + sizeof(float) * (CHOL_BS * (CHOL_BS + 1) +
                                                        2 * CHOL_BS));
    int cum = 0;
    for (int k0 = 0, step = 0; k0 < N; k0 += CHOL_BS, ++step) {
      if (step > 0) {
        if (tid == 0) {
          long long it = 0;
          while (atomicAdd(tiles_done, 0) < cum) {
            __builtin_amdgcn_s_sleep(8);
            if (++it > 5000000LL) {  // ~1 s at ~512 cycles per sleep(8)  // bounded: fail loudly, not a hang
              if (atomicCAS(info + b, 0, -777) == 0) {}
              break;
            }
          }
        }
        __syncthreads();
        __threadfence();  // acquire: invalidate L1 before reading tiles' output
        if (*(volatile int*)(info + b) < 0) return;  // timeout sentinel
      }
      chol_panel_body<2, PERSIST_TPB>(Ab, logdet + b, info + b, N, k0, rhs_b,
                                      S, colbuf, P);
      __threadfence();  // release: panel writes visible device-wide
      __syncthreads();
      if (tid == 0) atomicExch(panel_flag, step + 1);
      cum += persist_pairs(N, k0);
    }
  } else {
    float(*Pi)[CHOL_BS] = (float(*)[CHOL_BS])smem;
    float(*Pj)[CHOL_BS] =
        (float(*)[CHOL_BS])(smem + sizeof(float) * SYRK_TS * CHOL_BS);
    const int t = role - 1;
    for (int k0 = 0, step = 0; k0 < N; k0 += CHOL_BS, ++step) {
      const int pairs = persist_pairs(N, k0);
      if (t >= pairs) continue;  // no tile for this block at this step
      if (tid == 0) {
        long long it = 0;
        while (atomicAdd(panel_flag, 0) < step + 1) {
          __builtin_amdgcn_s_sleep(8);
          if (++it > 5000000LL) {  // ~1 s at ~512 cycles per sleep(8)
            if (atomicCAS(info + b, 0, -778) == 0) {}
            break;
          }
        }
      }
      __syncthreads();
      __threadfence();  // acquire before reading the panel output
      if (*(volatile int*)(info + b) < 0) return;  // timeout sentinel
      int p = t, ti = 0;
      while (p > ti) { p -= ti + 1; ++ti; }
      const int tj = p;
      chol_syrk_tile_body(Ab, N, k0, ti, tj, rhs_b, Pi, Pj);
      __threadfence();  // release tile writes
      __syncthreads();
      if (tid == 0) atomicAdd(tiles_done, 1);
    }
  }
}

extern "C" int launch_cholesky_persist(float* A, float* logdet, int* info,
                                       float* rhs, int* ws, int B, int N,
                                       hipStream_t stream) {
  // only the default panel configuration is instantiated
  if (panel_group_cols() != 2) return -1;
  const int trailing0 = N - CHOL_BS;
  if (trailing0 <= 0) return -1;
  const int nt0 = (trailing0 + SYRK_TS - 1) / SYRK_TS;
  const int T = nt0 * (nt0 + 1) / 2;
  const size_t lds = sizeof(float) * (CHOL_BS * (CHOL_BS + 1) + 2 * CHOL_BS +
                                      PERSIST_TPB * (CHOL_BS + 1));
  static int max_resident = -1;
  if (max_resident < 0) {
    hipFuncSetAttribute((const void*)chol_persist_kernel,
                        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    int per_cu = 0;
    hipOccupancyMaxActiveBlocksPerMultiprocessor(
        &per_cu, (const void*)chol_persist_kernel, PERSIST_TPB, lds);
    hipDeviceProp_t prop;
    int dev = 0;
    hipGetDevice(&dev);
    hipGetDeviceProperties(&prop, dev);
    max_resident = per_cu * prop.multiProcessorCount;
  }
  // DEADLOCK GUARD: every block must be co-resident for the spin handoffs
  if ((long long)B * (1 + T) > max_resident) return -1;
  hipLaunchKernelGGL(zero_i32_kernel, dim3((2 * B + 255) / 256), dim3(256), 0,
                     stream, ws, 2 * B);
  hipLaunchKernelGGL(zero_f32_kernel, dim3((B + 255) / 256), dim3(256), 0,
                     stream, logdet, B);
  hipLaunchKernelGGL(chol_persist_kernel, dim3(B, 1 + T), dim3(PERSIST_TPB),
                     lds, stream, A, logdet, info, rhs, ws, N);
  return 0;
}

// bf16-SYRK variant of the right-looking multik path (config-#2 precision
// route): panels factor in exact fp32 (chol_panel_kernel), only the
// trailing C -= P P^T runs on the bf16 matrix units (matern_bf16.hip).
extern "C" void launch_chol_syrk_bf16(float*, int, int, int, int, int, int,
                                      int, hipStream_t);

extern "C" void launch_cholesky_multik_bf16(float* A, float* logdet,
                                            int* info, int B, int N,
                                            hipStream_t stream) {
  hipLaunchKernelGGL(zero_f32_kernel, dim3((B + 255) / 256), dim3(256), 0,
                     stream, logdet, B);
  for (int k0 = 0; k0 < N; k0 += CHOL_BS) {
    LAUNCH_PANEL(B, stream, A, logdet, info, N, k0, nullptr);
    const int trailing = N - k0 - CHOL_BS;
    if (trailing > 0) {
      const int nt = (trailing + SYRK_TS - 1) / SYRK_TS;
      launch_chol_syrk_bf16(A, N, k0, nt, -1, 0, nt * (nt + 1) / 2, B,
                            stream);
    }
  }
}

extern "C" int launch_cholesky_persist(float*, float*, int*, float*, int*,
                                       int, int, hipStream_t);

extern "C" int launch_cholesky_fused_solve(float* A, float* logdet,
                                           int* info, float* rhs, int* ws,
                                           int B, int N, hipStream_t stream) {
  static int ok = -2;
  static int persist_on = -1;
  if (ok == -2) {
    const char* m = getenv("DMOSOPT_CHOL_MODE");
    const char* o = getenv("DMOSOPT_CHOL_OVERLAP");
    const char* p = getenv("DMOSOPT_CHOL_PANEL");
    const char* f = getenv("DMOSOPT_CHOL_FUSED_SOLVE");
    ok = ((m && m[0] == 's') || (o && o[0] == '1') || (p && p[0] == '6') ||
          (f && f[0] == '0'))
             ? 0
             : 1;
    // measured 205 vs 136 ms/epoch (same box, bit-identical HV): the
    // intra-kernel handoffs + pointer-based LDS codegen cost far more
    // than the ~16 saved dispatches. Kept as an opt-in experiment.
    const char* pe = getenv("DMOSOPT_CHOL_PERSIST");
    persist_on = (pe && pe[0] == '1') ? 1 : 0;
  }
  if (!ok || N <= CHOL_BS || B > 48) return -1;
  if (persist_on && ws != nullptr &&
      launch_cholesky_persist(A, logdet, info, rhs, ws, B, N, stream) == 0)
    return 0;
  launch_cholesky_multik(A, logdet, info, rhs, B, N, stream);
  return 0;
}

extern "C" void launch_cholesky_batched(float* A, float* logdet, int* info,
                                        int B, int N, hipStream_t stream) {
  static int multik_max_b = -2;
  if (multik_max_b == -2) {
    // below this batch count the one-block-per-matrix kernel cannot fill
    // the 256 CUs, so the multi-launch right-looking path wins; override
    // with DMOSOPT_CHOL_MODE=single|multik for A/B measurement
    const char* env = getenv("DMOSOPT_CHOL_MODE");
    if (env && env[0] == 's') multik_max_b = 0;
    else if (env && env[0] == 'm') multik_max_b = 1 << 30;
    else multik_max_b = 48;
  }
  if (B <= multik_max_b && N > CHOL_BS) {
    launch_cholesky_multik(A, logdet, info, nullptr, B, N, stream);
    return;
  }
  static int mode = -1;
  if (mode < 0) {
    // same-box A/B (profiles/README.md): LDS/barrier diag factor measured
    // 0.80 ms vs 1.27 ms wave-sync at B=18, N=300 — LDS is the default;
    // DMOSOPT_CHOL_DIAG=wave selects the wave-synchronous variant.
    const char* env = getenv("DMOSOPT_CHOL_DIAG");
    mode = (env && env[0] == 'w') ? 1 : 0;
    hipFuncSetAttribute((const void*)cholesky_batched_kernel<true>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        (int)sizeof(CholLds));
    hipFuncSetAttribute((const void*)cholesky_batched_kernel<false>,
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        (int)sizeof(CholLds));
  }
  if (mode)
    hipLaunchKernelGGL(cholesky_batched_kernel<true>, dim3(B), dim3(CHOL_TPB),
                       sizeof(CholLds), stream, A, logdet, info, N);
  else
    hipLaunchKernelGGL(cholesky_batched_kernel<false>, dim3(B), dim3(CHOL_TPB),
                       sizeof(CholLds), stream, A, logdet, info, N);
}

extern "C" void launch_forward_solve_batched(const float* L, float* Y, int B,
                                             int N, int R, hipStream_t stream) {
  dim3 block(256, 1);
  dim3 grid(B, R);
  hipLaunchKernelGGL(forward_solve_batched_kernel, grid, block, 0, stream, L,
                     Y, N, R);
}

extern "C" void launch_backward_solve_batched(const float* L, float* Y, int B,
                                              int N, int R, hipStream_t stream) {
  dim3 block(256, 1);
  dim3 grid(B, R);
  hipLaunchKernelGGL(backward_solve_batched_kernel, grid, block, 0, stream, L,
                     Y, N, R);
}
