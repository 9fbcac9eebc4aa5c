// AGE-MOEA survival selection on device.
//
// Replaces the host loop of survival_score (reference AGEMOEA.py:389-442;
// our incremental O(m^2) variant in moea/agemoea.py): the selection is
// inherently serial (each round picks the remaining point with the largest
// sum of its two smallest distances TO THE SELECTED SET, then that point
// joins the set), but every round's argmax and distance-merge are parallel
// over m. One workgroup runs the whole loop with d1/d2 in LDS.
//
// Round structure (2 barriers per round): per-thread strided argmax ->
// wave shfl_xor reduce (no barrier) -> wave leaders to LDS -> barrier ->
// every thread folds the 4 leader entries (uniform result, no reduce
// tree) -> merge column sel into d1/d2 (the owner thread of sel consumes
// it instead of merging, so no select/merge barrier) -> barrier.
//
// d1 == -inf marks consumed AND preselected points (both are excluded
// from the argmax and the merge). D is row-normalized by each point's own
// norm (D[i][j] = d(i,j)/nn[i]) and therefore NOT symmetric: the merge
// reads COLUMN sel of row i, exactly like the host path's
// distances[remaining, best]. Ties resolve toward the lowest index
// (np.argmax parity).

#include "common.h"
#include <math.h>

#define AGES_TPB 256
#define AGES_WAVES (AGES_TPB / WAVE_SIZE)

extern "C" __global__ __launch_bounds__(AGES_TPB) void agemoea_survival_kernel(
    const float* __restrict__ D,      // (m, m) normalized distances
    const unsigned char* __restrict__ preselected,  // (m,) extreme points
    float* __restrict__ crowd,        // (m,) out: 2-NN score per point
    int m) {
  extern __shared__ float lds[];
  float* d1 = lds;           // (m,) smallest distance to selected set
  float* d2 = lds + m;       // (m,) second smallest
  float* red_v = d2 + m;     // (AGES_WAVES,)
  int* red_i = (int*)(red_v + AGES_WAVES);

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE_SIZE - 1);
  const int wave = tid >> 6;

  __shared__ int n_pre;
  if (tid == 0) {
    int c = 0;
    for (int j = 0; j < m; ++j) c += preselected[j] ? 1 : 0;
    n_pre = c;
  }
  __syncthreads();
  for (int i = tid; i < m; i += AGES_TPB) {
    float a = HUGE_VALF, b = HUGE_VALF;
    if (!preselected[i]) {
      for (int j = 0; j < m; ++j) {
        if (!preselected[j]) continue;
        const float v = D[(long long)i * m + j];
        if (v < a) { b = a; a = v; }
        else if (v < b) { b = v; }
      }
      crowd[i] = 0.f;
    } else {
      a = -HUGE_VALF;  // excluded from selection and merging
      crowd[i] = HUGE_VALF;
    }
    d1[i] = a;
    d2[i] = b;
  }
  __syncthreads();

  const int n_rounds = m - n_pre;
  for (int round = 0; round < n_rounds; ++round) {
    float bv = -HUGE_VALF;
    int bi = m;
    for (int i = tid; i < m; i += AGES_TPB) {
      const float a = d1[i];
      if (a == -HUGE_VALF) continue;  // consumed or preselected
      const float b = d2[i];
      const float s = isinf(b) ? a : a + b;
      if (s > bv || (s == bv && i < bi)) { bv = s; bi = i; }
    }
    // wave-level reduce via xor shuffles (no barriers)
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(bv, off, WAVE_SIZE);
      const int oi = __shfl_xor(bi, off, WAVE_SIZE);
      if (ov > bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
    }
    if (lane == 0) {
      red_v[wave] = bv;
      red_i[wave] = bi;
    }
    __syncthreads();
    // every thread folds the wave leaders -> uniform (sel, score)
    float sv = red_v[0];
    int sel = red_i[0];
#pragma unroll
    for (int w = 1; w < AGES_WAVES; ++w) {
      const float ov = red_v[w];
      const int oi = red_i[w];
      if (ov > sv || (ov == sv && oi < sel)) { sv = ov; sel = oi; }
    }
    if (sel >= m) break;  // nothing alive (defensive)
    if (tid == 0) crowd[sel] = sv;
    // merge column sel; the owner thread consumes sel in the same pass
    for (int i = tid; i < m; i += AGES_TPB) {
      if (i == sel) {
        d1[i] = -HUGE_VALF;
        continue;
      }
      if (d1[i] == -HUGE_VALF) continue;
      const float dn = D[(long long)i * m + sel];
      if (dn < d2[i]) d2[i] = dn;
      if (d2[i] < d1[i]) { const float t = d1[i]; d1[i] = d2[i]; d2[i] = t; }
    }
    __syncthreads();
  }
}

extern "C" void launch_agemoea_survival(const float* D,
                                        const unsigned char* preselected,
                                        float* crowd, int m, hipStream_t s) {
  const size_t lds = (size_t)(2 * m + AGES_WAVES) * sizeof(float) +
                     AGES_WAVES * sizeof(int);
  hipLaunchKernelGGL(agemoea_survival_kernel, dim3(1), dim3(AGES_TPB), lds, s,
                     D, preselected, crowd, m);
}

// ----------------------------------------------------- fused Minkowski D
// D[i][j] = (sum_k |Y[i,k]-Y[j,k]|^p)^(1/p) / nn[i] with
// nn[i] = (sum_k |Y[i,k]|^p)^(1/p): replaces torch.cdist(p) + vector_norm
// + divide (three passes over the m^2 matrix, 543 us/gen at m=2048 in the
// config-#3 profile) with one write-once pass. d <= 16.
extern "C" __global__ void minkowski_norm_matrix_kernel(
    const float* __restrict__ Y,  // (m, d)
    float* __restrict__ D,        // (m, m)
    int m, int d, float p) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)m * m) return;
  const int i = (int)(idx / m), j = (int)(idx % m);
  const float inv_p = 1.0f / p;
  float acc = 0.f, nrm = 0.f;
  if (p == 2.0f) {
    for (int k = 0; k < d; ++k) {
      const float yi = Y[i * d + k];
      const float t = yi - Y[j * d + k];
      acc = fmaf(t, t, acc);
      nrm = fmaf(yi, yi, nrm);
    }
    D[idx] = sqrtf(acc) / sqrtf(nrm);
    return;
  }
  for (int k = 0; k < d; ++k) {
    const float yi = Y[i * d + k];
    acc += powf(fabsf(yi - Y[j * d + k]), p);
    nrm += powf(fabsf(yi), p);
  }
  D[idx] = powf(acc, inv_p) / powf(nrm, inv_p);
}

extern "C" void launch_minkowski_norm_matrix(const float* Y, float* D, int m,
                                             int d, float p, hipStream_t s) {
  const long long total = (long long)m * m;
  const int threads = 256;
  const long long blocks = (total + threads - 1) / threads;
  hipLaunchKernelGGL(minkowski_norm_matrix_kernel, dim3((unsigned)blocks),
                     dim3(threads), 0, s, Y, D, m, d, p);
}
