// bf16 Matern cross-kernel assembly + bf16 SYRK trailing update (gfx950).
//
// The BASELINE config-#2 precision path: inputs are rounded to bf16 and the
// pairwise dot products run on the bf16 matrix units
// (v_mfma_f32_16x16x32_bf16, fp32 accumulate — 16x the f32-MFMA rate), with
// everything after the dot (norms, Matern transform, output K) in fp32.
// d2 is formed as |q~|^2 + |x~|^2 - 2 q~.x~ where BOTH the norms and the
// cross term use the SAME bf16-rounded values, so d2 >= 0 up to fp32
// accumulation error (clamped) — mixing rounded cross terms with unrounded
// norms would inject O(|q| * bf16_eps) signed error instead.
//
// The TRAIN-side kernel assembly (SCE-UA NMLL search) stays fp32
// (matern.hip): bf16 d2 perturbations flip SCE-UA accept decisions and
// change the fit's iteration count (profiles/README.md round-1 note); the
// prediction path tolerates tolerance-level error, the fit path does not.
//
// Fragment layout for mfma_f32_16x16x32_bf16 (cdna4 ISA):
//   A: lane l -> A[l & 15][(l >> 4) * 8 + j], j = 0..7 (8 bf16 = 4 VGPRs)
//   B: lane l -> B[(l >> 4) * 8 + j][l & 15]
//   C/D (f32x4): col = lane & 15, row = (lane >> 4) * 4 + reg
// mfma_bf16_probe() below exists so a unit test can verify this mapping on
// hardware against a torch matmul oracle.

#include "common.h"
#include <math.h>

#define TILE 32
#define TPB 256

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

template <int NU>
__device__ __forceinline__ float matern_transform_f(float d2) {
  if (NU == 0) return __expf(-0.5f * d2);
  const float r = sqrtf(fmaxf(d2, 0.f));
  if (NU == 1) return __expf(-r);
  if (NU == 3) {
    const float s = 1.7320508075688772f * r;
    return (1.f + s) * __expf(-s);
  }
  const float s = 2.23606797749979f * r;
  return (1.f + s + (5.f / 3.f) * d2) * __expf(-s);
}

// ------------------------------------------------------------- layout probe
// D(16,16) = A(16,32) x B(32,16), one wave, using the assumed fragment maps.
__global__ void mfma_bf16_probe_kernel(const float* __restrict__ A,
                                       const float* __restrict__ B,
                                       float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  const int lr = lane & 15, lk = lane >> 4;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (__bf16)A[lr * 32 + lk * 8 + j];
    b[j] = (__bf16)B[(lk * 8 + j) * 16 + lr];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[(lk * 4 + r) * 16 + lr] = acc[r];
}

extern "C" void launch_mfma_bf16_probe(const float* A, const float* B,
                                       float* D, hipStream_t s) {
  hipLaunchKernelGGL(mfma_bf16_probe_kernel, dim3(1), dim3(64), 0, s, A, B, D);
}

// -------------------------------------------------------- bf16 cross kernel
// K[b] (P,N) fp32 = sf2 * matern_nu(|q~_i - x~_j|) from bf16-rounded scaled
// rows. 32x32 tile per 256-thread block; K dimension padded to 32.
template <int NU, bool ANISO>
__global__ void matern_cross_bf16_kernel(
    const float* __restrict__ Xq,     // (P, D)
    const float* __restrict__ X,      // (N, D)
    const float* __restrict__ theta,  // (B, p)
    float* __restrict__ K,            // (B, P, N)
    int P, int N, int D, int theta_stride,
    const float* __restrict__ q_lb, const float* __restrict__ q_invrg) {
  extern __shared__ char lds_raw[];
  const int Kpad = (D + 31) & ~31;
  // +8 shorts of row padding: consecutive rows' 16-byte fragment reads land
  // on different bank groups
  const int DS = Kpad + 8;
  __bf16* q_tile = (__bf16*)lds_raw;                 // [TILE][DS]
  __bf16* x_tile = q_tile + TILE * DS;               // [TILE][DS]
  float* qn = (float*)(x_tile + TILE * DS);          // [TILE]
  float* xn = qn + TILE;                             // [TILE]

  const int b = blockIdx.z;
  const int tile_p = blockIdx.y * TILE;
  const int tile_n = blockIdx.x * TILE;
  const float sf2 = __expf(theta[b * theta_stride + 0]);

  for (int idx = threadIdx.x; idx < TILE * DS; idx += TPB) {
    const int row = idx / DS;
    const int col = idx % DS;
    float qv = 0.f, xv = 0.f;
    if (col < D) {
      const float inv_ell =
          ANISO ? __expf(-theta[b * theta_stride + 1 + col])
                : __expf(-theta[b * theta_stride + 1]);
      const int gq = tile_p + row;
      if (gq < P) {
        qv = Xq[gq * D + col];
        if (q_lb != nullptr) qv = (qv - q_lb[col]) * q_invrg[col];
        qv *= inv_ell;
      }
      const int gx = tile_n + row;
      if (gx < N) xv = X[gx * D + col] * inv_ell;
    }
    q_tile[row * DS + col] = (__bf16)qv;
    x_tile[row * DS + col] = (__bf16)xv;
  }
  __syncthreads();

  // norms of the ROUNDED rows (fp32 accumulate)
  if (threadIdx.x < 2 * TILE) {
    const bool is_q = threadIdx.x < TILE;
    const int row = is_q ? threadIdx.x : threadIdx.x - TILE;
    const __bf16* src = (is_q ? q_tile : x_tile) + row * DS;
    float acc = 0.f;
    for (int k = 0; k < D; ++k) {
      const float v = (float)src[k];
      acc = fmaf(v, v, acc);
    }
    (is_q ? qn : xn)[row] = acc;
  }
  __syncthreads();

  // wave w owns 16x16 subtile (w>>1, w&1); K-loop in steps of 32
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int lr = lane & 15, lk = lane >> 4;
  const int r16 = (wave >> 1) * 16, c16 = (wave & 1) * 16;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < Kpad; k0 += 32) {
    const bf16x8 a =
        *(const bf16x8*)(q_tile + (r16 + lr) * DS + k0 + lk * 8);
    const bf16x8 bv =
        *(const bf16x8*)(x_tile + (c16 + lr) * DS + k0 + lk * 8);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int lp = r16 + lk * 4 + r;
    const int ln = c16 + lr;
    const int gp = tile_p + lp;
    const int gn = tile_n + ln;
    if (gp >= P || gn >= N) continue;
    const float d2 = fmaxf(qn[lp] + xn[ln] - 2.f * acc[r], 0.f);
    K[((long long)b * P + gp) * N + gn] = sf2 * matern_transform_f<NU>(d2);
  }
}

extern "C" void launch_matern_cross_bf16(const float* Xq, const float* X,
                                         const float* theta, float* K, int B,
                                         int P, int N, int D,
                                         int theta_stride, int nu_code,
                                         int aniso, const float* q_lb,
                                         const float* q_invrg,
                                         hipStream_t stream) {
  dim3 grid((N + TILE - 1) / TILE, (P + TILE - 1) / TILE, B);
  const int Kpad = (D + 31) & ~31;
  const size_t lds = (size_t)2 * TILE * (Kpad + 8) * sizeof(__bf16) +
                     2 * TILE * sizeof(float);
#define DISPATCH(NU, AN)                                                     \
  hipLaunchKernelGGL((matern_cross_bf16_kernel<NU, AN>), grid, dim3(TPB),    \
                     lds, stream, Xq, X, theta, K, P, N, D, theta_stride,    \
                     q_lb, q_invrg)
  switch (nu_code) {
    case 0: if (aniso) DISPATCH(0, true); else DISPATCH(0, false); break;
    case 1: if (aniso) DISPATCH(1, true); else DISPATCH(1, false); break;
    case 3: if (aniso) DISPATCH(3, true); else DISPATCH(3, false); break;
    default: if (aniso) DISPATCH(5, true); else DISPATCH(5, false); break;
  }
#undef DISPATCH
}

// ------------------------------------------------- bf16 SYRK trailing update
// C -= Pi * Pj^T with the 32-wide panels rounded to bf16 and the products
// on the bf16 matrix units (fp32 accumulate). Same tiling/indexing contract
// as chol_syrk_kernel (cholesky.hip): CHOL_BS=32 panel, SYRK_TS=64 tiles.
#define BS16_CHOL_BS 32
#define BS16_SYRK_TS 64
#define BS16_TPB 256

__global__ __launch_bounds__(BS16_TPB) void chol_syrk_bf16_kernel(
    float* __restrict__ A, int N, int k0, int nt, int tj_fixed, int off) {
  // bf16 panels: row stride 40 shorts (80 B) staggers the 16-byte fragment
  // reads of consecutive rows across bank groups
  __shared__ __bf16 Pi[BS16_SYRK_TS][BS16_CHOL_BS + 8];
  __shared__ __bf16 Pj[BS16_SYRK_TS][BS16_CHOL_BS + 8];
  const int b = blockIdx.x;
  float* Ab = A + (long long)b * N * N;
  const int r0 = k0 + BS16_CHOL_BS;
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
  const int i0 = r0 + ti * BS16_SYRK_TS, j0 = r0 + tj * BS16_SYRK_TS;
  const int tid = threadIdx.x;

  for (int idx = tid; idx < BS16_SYRK_TS * BS16_CHOL_BS; idx += blockDim.x) {
    const int r = idx / BS16_CHOL_BS, c = idx % BS16_CHOL_BS;
    Pi[r][c] = (__bf16)((i0 + r < N) ? Ab[(long long)(i0 + r) * N + k0 + c] : 0.0f);
    Pj[r][c] = (__bf16)((j0 + r < N) ? Ab[(long long)(j0 + r) * N + k0 + c] : 0.0f);
  }
  __syncthreads();

  const int wave = tid >> 6, lane = tid & 63;
  const int lr = lane & 15, lk = lane >> 4;
#pragma unroll
  for (int sIdx = 0; sIdx < 4; ++sIdx) {
    const int sub = wave * 4 + sIdx;
    const int r16 = (sub >> 2) * 16, c16 = (sub & 3) * 16;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    // one 16x16x32 MFMA covers the whole 32-wide panel
    const bf16x8 a = *(const bf16x8*)(&Pi[r16 + lr][lk * 8]);
    const bf16x8 bv = *(const bf16x8*)(&Pj[c16 + lr][lk * 8]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bv, acc, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int i = i0 + r16 + lk * 4 + r;
      const int j = j0 + c16 + lr;
      if (i < N && j < N && j <= i) Ab[(long long)i * N + j] -= acc[r];
    }
  }
}

extern "C" void launch_chol_syrk_bf16(float* A, int N, int k0, int nt,
                                      int tj_fixed, int off, int n_pairs,
                                      int B, hipStream_t stream) {
  hipLaunchKernelGGL(chol_syrk_bf16_kernel, dim3(B, n_pairs),
                     dim3(BS16_TPB), 0, stream, A, N, k0, nt, tj_fixed, off);
}
