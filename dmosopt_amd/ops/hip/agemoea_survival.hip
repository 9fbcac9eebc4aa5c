// AGE-MOEA greedy 2-NN survival selection on device.
//
// Replaces the host loop of survival_score (reference AGEMOEA.py:389-442;
// our incremental O(m^2) variant in moea/agemoea.py): the selection is
// inherently serial (each round picks the remaining point with the largest
// sum of its two smallest distances TO THE SELECTED SET, then that point
// joins the set), but every round's argmax and distance-merge are parallel
// over m. One workgroup runs the whole loop with d1/d2 in LDS; the merge
// reads COLUMN sel of the (m x m) matrix (D is row-normalized by each
// point's own norm, so NOT symmetric) — strided reads served from L2
// after the first pass.
//
// Semantics match the numpy path bit-for-bit in selection ORDER except for
// argmax ties (numpy picks the lowest index; the LDS tree reduction below
// also resolves ties toward the lowest index to keep parity).

#include "common.h"
#include <math.h>

#define AGES_TPB 256

extern "C" __global__ __launch_bounds__(AGES_TPB) void agemoea_survival_kernel(
    const float* __restrict__ D,      // (m, m) normalized distances
    const unsigned char* __restrict__ preselected,  // (m,) extreme points
    float* __restrict__ crowd,        // (m,) out: 2-NN score per point
    int m) {
  extern __shared__ float lds[];
  float* d1 = lds;           // (m,) smallest distance to selected set
  float* d2 = lds + m;       // (m,) second smallest
  // reduction scratch: AGES_TPB floats + AGES_TPB ints
  float* red_v = d2 + m;
  int* red_i = (int*)(red_v + AGES_TPB);

  const int tid = threadIdx.x;

  // count preselected and init d1/d2 from the preselected columns
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
    }
    d1[i] = a;
    d2[i] = b;
    crowd[i] = preselected[i] ? HUGE_VALF : 0.f;
  }
  __syncthreads();

  const int n_rounds = m - n_pre;
  for (int round = 0; round < n_rounds; ++round) {
    // parallel argmax over alive points of (isinf(d2) ? d1 : d1+d2);
    // alive encoding: d1 == -inf marks a consumed point
    float best_v = -HUGE_VALF;
    int best_i = m;
    for (int i = tid; i < m; i += AGES_TPB) {
      const float a = d1[i];
      if (a == -HUGE_VALF || crowd[i] == HUGE_VALF) continue;  // consumed/pre
      const float b = d2[i];
      const float s = isinf(b) ? a : a + b;
      // lowest-index tie resolution
      if (s > best_v || (s == best_v && i < best_i)) { best_v = s; best_i = i; }
    }
    red_v[tid] = best_v;
    red_i[tid] = best_i;
    __syncthreads();
    for (int off = AGES_TPB / 2; off > 0; off >>= 1) {
      if (tid < off) {
        const float ov = red_v[tid + off];
        const int oi = red_i[tid + off];
        if (ov > red_v[tid] || (ov == red_v[tid] && oi < red_i[tid])) {
          red_v[tid] = ov;
          red_i[tid] = oi;
        }
      }
      __syncthreads();
    }
    const int sel = red_i[0];
    const float sel_score = red_v[0];
    __syncthreads();
    if (sel >= m) break;  // nothing alive (defensive)
    if (tid == 0) {
      crowd[sel] = sel_score;
      d1[sel] = -HUGE_VALF;  // consume
    }
    __syncthreads();
    // merge the new member's distances into everyone's two smallest.
    // D is row-normalized (D[i][j] = d(i,j)/nn[i]) and therefore NOT
    // symmetric: point i's distance to the new member sel is column sel
    // of ROW i (matching the host path's distances[remaining, best]).
    for (int i = tid; i < m; i += AGES_TPB) {
      if (d1[i] == -HUGE_VALF || crowd[i] == HUGE_VALF) continue;
      const float dn = D[(long long)i * m + sel];
      if (dn < d2[i]) d2[i] = dn;
      if (d2[i] < d1[i]) { const float t = d1[i]; d1[i] = d2[i]; d2[i] = t; }
    }
    __syncthreads();
  }
  // preselected points keep +inf scores (the caller overwrites with np.inf)
}

extern "C" void launch_agemoea_survival(const float* D,
                                        const unsigned char* preselected,
                                        float* crowd, int m, hipStream_t s) {
  const size_t lds =
      (size_t)(2 * m + AGES_TPB) * sizeof(float) + AGES_TPB * sizeof(int);
  hipLaunchKernelGGL(agemoea_survival_kernel, dim3(1), dim3(AGES_TPB), lds, s,
                     D, preselected, crowd, m);
}
