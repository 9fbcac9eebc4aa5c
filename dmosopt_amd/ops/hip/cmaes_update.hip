// Batched MO-CMA-ES rank-1 Cholesky update (reference CMAES.py:489-537).
//
// For K chosen offspring at once (the 'large-pop covariance update' of
// BASELINE config #5):
//   pc'   = (1-cc) pc [+ sqrt(cc(2-cc)) z  if psucc < pthresh]
//   w     = Ainv pc'
//   if max(w) > 1e-20:
//     a = sqrt(alpha), b = a/|w|^2 (sqrt(1 + beta/alpha |w|^2) - 1)
//     A'    = a A + b pc' w^T
//     Ainv' = Ainv/a - c w (w^T Ainv)
// One workgroup per individual; pc', w and w^T Ainv staged in LDS; the two
// d x d outer-product updates are grid-stride elementwise work.

#include "common.h"
#include <math.h>

#define CMAES_TPB 256
#define CMAES_DMAX 256  // max parameter dimension staged in LDS

__global__ __launch_bounds__(CMAES_TPB) void cmaes_update_kernel(
    float* __restrict__ A,      // (K, d, d)
    float* __restrict__ Ainv,   // (K, d, d)
    float* __restrict__ pc,     // (K, d) inout
    const float* __restrict__ z,      // (K, d) normalized steps
    const float* __restrict__ psucc,  // (K,)
    int K, int d, float cc, float ccov, float pthresh) {
  __shared__ float s_pc[CMAES_DMAX];
  __shared__ float s_w[CMAES_DMAX];
  __shared__ float s_wAinv[CMAES_DMAX];
  __shared__ float s_norm;

  const int k = blockIdx.x;
  const int tid = threadIdx.x;
  float* Ak = A + (long long)k * d * d;
  float* Aik = Ainv + (long long)k * d * d;
  float* pck = pc + (long long)k * d;
  const bool below = psucc[k] < pthresh;
  const float csq = sqrtf(cc * (2.f - cc));
  const float alpha = below ? (1.f - ccov) : (1.f - ccov) + ccov * cc * (2.f - cc);
  const float beta = ccov;

  // pc update
  for (int i = tid; i < d; i += CMAES_TPB) {
    float v = (1.f - cc) * pck[i];
    if (below) v = fmaf(csq, z[(long long)k * d + i], v);
    s_pc[i] = v;
    pck[i] = v;
  }
  __syncthreads();

  // w = Ainv pc' ; wAinv = pc'^T Ainv^T ... (w^T Ainv)_j = sum_i w_i Ainv[i][j]
  for (int i = tid; i < d; i += CMAES_TPB) {
    float acc = 0.f;
    const float* row = Aik + (long long)i * d;
    for (int j = 0; j < d; ++j) acc = fmaf(row[j], s_pc[j], acc);
    s_w[i] = acc;
  }
  __syncthreads();
  if (tid == 0) {
    float n2 = 0.f, mx = -INFINITY;
    for (int i = 0; i < d; ++i) {
      n2 = fmaf(s_w[i], s_w[i], n2);
      mx = fmaxf(mx, s_w[i]);
    }
    s_norm = (mx > 1e-20f) ? n2 : -1.f;  // negative => skip update
  }
  __syncthreads();
  const float norm_w2 = s_norm;
  if (norm_w2 < 0.f) return;  // noise-level update skipped (ref :527)

  for (int j = tid; j < d; j += CMAES_TPB) {
    float acc = 0.f;
    for (int i = 0; i < d; ++i) acc = fmaf(s_w[i], Aik[(long long)i * d + j], acc);
    s_wAinv[j] = acc;
  }
  __syncthreads();

  const float a = sqrtf(alpha);
  const float root = sqrtf(1.f + beta / alpha * norm_w2);
  const float bcoef = a / norm_w2 * (root - 1.f);
  const float ccoef = 1.f / (a * norm_w2) * (1.f - 1.f / root);

  for (int idx = tid; idx < d * d; idx += CMAES_TPB) {
    const int i = idx / d, j = idx % d;
    Ak[idx] = fmaf(bcoef, s_pc[i] * s_w[j], a * Ak[idx]);
    Aik[idx] = fmaf(-ccoef, s_w[i] * s_wAinv[j], (1.f / a) * Aik[idx]);
  }
}

extern "C" void launch_cmaes_update(float* A, float* Ainv, float* pc,
                                    const float* z, const float* psucc, int K,
                                    int d, float cc, float ccov, float pthresh,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(cmaes_update_kernel, dim3(K), dim3(CMAES_TPB), 0, stream,
                     A, Ainv, pc, z, psucc, K, d, cc, ccov, pthresh);
}
