// Exact hypervolume kernels: 2D/3D sweeps, batched EHVI, and the
// stream-compaction step of the Lacour box decomposition.
//
// Replaces reference hv_box_decomposition.py:44-461 on device:
//  * hv2d: one workgroup, LDS bitonic sort by f0 + inclusive prefix-min of
//    f1 + clamped staircase area (the union-of-boxes integral needs no
//    dominance pruning: the running min makes dominated points contribute
//    zero width at the right height).
//  * hv3d: z-sorted plane sweep with one workgroup PER SLICE; each slice
//    re-scans the x-sorted point list with a wave64 shfl prefix-min over
//    the z<=z_i mask. O(n^2/64) parallel work across n concurrent
//    workgroups instead of the reference's sequential slice loop.
//  * ehvi: per-candidate thread computes the sum over boxes of the product
//    over dims of std*(phi(l)-phi(u)) + mu*(Phi(u)-Phi(l)), all in fp64
//    (matches the fp64 host oracle to ~1e-12; no log-space tricks needed
//    at CDNA4's 1:2 fp64 rate for this tiny op).
//  * Lacour insert: two deterministic passes (flag kernel -> host cumsum ->
//    scatter kernel) that reproduce the numpy emission order exactly, so
//    the coordinate dedupe picks identical representatives.
//
// CDNA4-only (gfx950), wave64. fp64 throughout: HV values feed termination
// decisions and the oracle tests demand <=1e-6 relative error.

#include "common.h"
#include <math.h>

#define HV2D_MAX_N 8192  // LDS: 8192 * 16 B = 128 KB (<160 KB per CU)

__device__ __forceinline__ double wave_scan_min_incl(double v) {
  // inclusive prefix-min over the 64 lanes of a wave
  for (int off = 1; off < WAVE_SIZE; off <<= 1) {
    const double up = __shfl_up(v, off, WAVE_SIZE);
    if ((int)(threadIdx.x & (WAVE_SIZE - 1)) >= off) v = fmin(v, up);
  }
  return v;
}

// --------------------------------------------------------------------- hv2d
extern "C" __global__ void hv2d_kernel(const double* __restrict__ P,  // (n,2)
                                       const double* __restrict__ ref,  // (2,)
                                       double* __restrict__ out, int n) {
  extern __shared__ double lds[];          // xs[M] | ys[M]
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  int M = 1;
  while (M < n) M <<= 1;
  double* xs = lds;
  double* ys = lds + M;
  const double refx = ref[0], refy = ref[1];
  for (int i = tid; i < M; i += nthreads) {
    if (i < n) {
      xs[i] = P[2 * i];
      ys[i] = P[2 * i + 1];
    } else {
      xs[i] = HUGE_VAL;  // pad sorts to the end, contributes zero width
      ys[i] = HUGE_VAL;
    }
  }
  __syncthreads();
  // bitonic sort ascending by x (ties need no secondary key: equal x means
  // zero width for the earlier one regardless of order)
  for (int k = 2; k <= M; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = tid; i < M; i += nthreads) {
        const int ixj = i ^ j;
        if (ixj > i) {
          const bool up = ((i & k) == 0);
          if ((xs[i] > xs[ixj]) == up) {
            double t = xs[i]; xs[i] = xs[ixj]; xs[ixj] = t;
            t = ys[i]; ys[i] = ys[ixj]; ys[ixj] = t;
          }
        }
      }
      __syncthreads();
    }
  }
  // inclusive prefix-min of y in x order (Hillis-Steele in LDS)
  for (int off = 1; off < M; off <<= 1) {
    double mine = HUGE_VAL;
    for (int i = tid; i < M; i += nthreads) {
      // read phase must complete before writes: stage into registers
      mine = (i >= off) ? fmin(ys[i], ys[i - off]) : ys[i];
      // stash in xs-free scratch? use a second barrier scheme instead:
      // write to a temp slot interleaved — simpler: two-pass with barrier
      lds[2 * M + i] = mine;  // scratch region
    }
    __syncthreads();
    for (int i = tid; i < M; i += nthreads) ys[i] = lds[2 * M + i];
    __syncthreads();
  }
  // staircase area: sum over i of clamped width * clamped height
  double acc = 0.0;
  for (int i = tid; i < n; i += nthreads) {
    const double x0 = fmin(xs[i], refx);
    const double x1 = (i + 1 < n) ? fmin(xs[i + 1], refx) : refx;
    const double h = refy - ys[i];
    if (x1 > x0 && h > 0.0) acc += (x1 - x0) * h;
  }
  // block reduction
  __shared__ double partial[16];
  double w = acc;
  for (int off = 32; off > 0; off >>= 1) w += __shfl_xor(w, off, WAVE_SIZE);
  const int wave = tid / WAVE_SIZE;
  if ((tid & (WAVE_SIZE - 1)) == 0) partial[wave] = w;
  __syncthreads();
  if (tid == 0) {
    double total = 0.0;
    for (int i = 0; i < (nthreads + WAVE_SIZE - 1) / WAVE_SIZE; ++i)
      total += partial[i];
    *out = total;
  }
}

// --------------------------------------------------------------------- hv3d
// One workgroup (one wave) per z-slice i: area_i = HV2D of the points with
// z <= z_thr[i] (x-sorted), out[i] = dz[i] * area_i. Host sums out[].
extern "C" __global__ void hv3d_slices_kernel(
    const double* __restrict__ Px,   // (n,3) sorted by x ascending
    const double* __restrict__ z_thr,  // (n,) slice thresholds (z-sorted)
    const double* __restrict__ dz,     // (n,) slice heights
    const double* __restrict__ ref,    // (3,)
    double* __restrict__ out, int n) {
  const int slice = blockIdx.x;
  if (slice >= n) return;
  const double zt = z_thr[slice];
  const double refx = ref[0], refy = ref[1];
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  double carry = HUGE_VAL;  // running y-min from previous chunks
  double acc = 0.0;
  for (int base = 0; base < n; base += WAVE_SIZE) {
    const int i = base + lane;
    double x = HUGE_VAL, y = HUGE_VAL, xn = refx;
    if (i < n) {
      x = Px[3 * i];
      y = (Px[3 * i + 2] <= zt) ? Px[3 * i + 1] : HUGE_VAL;
      xn = (i + 1 < n) ? Px[3 * (i + 1)] : refx;
    }
    double ymin = fmin(wave_scan_min_incl(y), carry);
    if (i < n) {
      const double x0 = fmin(x, refx);
      const double x1 = fmin(xn, refx);
      const double h = refy - ymin;
      if (x1 > x0 && h > 0.0) acc += (x1 - x0) * h;
    }
    carry = __shfl(ymin, WAVE_SIZE - 1, WAVE_SIZE);
  }
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_xor(acc, off, WAVE_SIZE);
  if (lane == 0) out[slice] = dz[slice] * acc;
}

// --------------------------------------------------------------------- EHVI
// One thread per candidate; boxes stream from L2 (all threads read the same
// rows). Infinite bounds encode as +-HUGE_VAL exactly like the host oracle.
extern "C" __global__ void ehvi_kernel(const double* __restrict__ L,  // (nb,d)
                                       const double* __restrict__ U,  // (nb,d)
                                       const double* __restrict__ mu,   // (B,d)
                                       const double* __restrict__ var,  // (B,d)
                                       double* __restrict__ out,        // (B,)
                                       int B, int nb, int d) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const double inv_sqrt2 = 0.70710678118654752440;
  const double inv_sqrt2pi = 0.39894228040143267794;
  double m[16], s[16];
  for (int k = 0; k < d; ++k) {
    m[k] = mu[b * d + k];
    s[k] = sqrt(var[b * d + k]);
  }
  double total = 0.0;
  for (int box = 0; box < nb; ++box) {
    double prod = 1.0;
    for (int k = 0; k < d; ++k) {
      const double l = L[box * d + k];
      const double u = U[box * d + k];
      const double zl = (l - m[k]) / s[k];
      const double zu = (u - m[k]) / s[k];
      const bool linf = isinf(l), uinf = isinf(u);
      const double Phi_l = linf ? 0.0 : 0.5 * (1.0 + erf(zl * inv_sqrt2));
      const double Phi_u = uinf ? 1.0 : 0.5 * (1.0 + erf(zu * inv_sqrt2));
      const double phi_l = linf ? 0.0 : inv_sqrt2pi * exp(-0.5 * zl * zl);
      const double phi_u = uinf ? 0.0 : inv_sqrt2pi * exp(-0.5 * zu * zu);
      prod *= s[k] * (phi_l - phi_u) + m[k] * (Phi_u - Phi_l);
    }
    total += prod;
  }
  out[b] = total;
}

// ------------------------------------------------------------ Lacour insert
// Pass 1: per-UB flags. dominated[u] = all(z < coords[u]); okj[u*d+j] = 1
// when the step-3 candidate (z_j, u_-j) is admissible for j < d-1.
extern "C" __global__ void lacour_flags_kernel(
    const double* __restrict__ coords,   // (U,d)
    const long long* __restrict__ defs,  // (U,d)
    const double* __restrict__ pts_aug,  // (n+d,d)
    const double* __restrict__ z,        // (d,)
    unsigned char* __restrict__ dominated,  // (U,)
    unsigned char* __restrict__ okj,        // (U,d)  (col d-1 unused)
    int U, int d) {
  const int u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= U) return;
  bool dom = true;
  for (int k = 0; k < d; ++k)
    if (!(z[k] < coords[u * d + k])) { dom = false; break; }
  dominated[u] = dom ? 1 : 0;
  for (int j = 0; j < d; ++j) okj[u * d + j] = 0;
  if (!dom) return;
  for (int j = 0; j < d - 1; ++j) {
    double mx = -HUGE_VAL;
    for (int k = 0; k < d; ++k) {
      if (k == j) continue;
      const double c = pts_aug[defs[u * d + k] * d + j];
      mx = fmax(mx, c);
    }
    if (mx < z[j]) okj[u * d + j] = 1;
  }
}

// Pass 2: deterministic scatter in the numpy emission order:
//   block A: replaced-last-coordinate rows for every dominated u (u order)
//   block B_j (j = 0..d-2): admissible step-3 rows per j (u order)
//   block K: the non-dominated survivors (u order)
// slotA/slotBj/slotK are exclusive-prefix-sum slot indices computed host
// side (torch cumsum), baseBj/baseK are the block base offsets.
extern "C" __global__ void lacour_scatter_kernel(
    const double* __restrict__ coords, const long long* __restrict__ defs,
    const double* __restrict__ z, const unsigned char* __restrict__ dominated,
    const unsigned char* __restrict__ okj,
    const long long* __restrict__ slotA,   // (U,) exclusive cumsum of dominated
    const long long* __restrict__ slotBj,  // (U,d) exclusive cumsum per column j
    const long long* __restrict__ baseBj,  // (d,) block base for each j
    long long baseK, long long point_idx,
    double* __restrict__ out_coords, long long* __restrict__ out_defs,
    int U, int d) {
  const int u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= U) return;
  if (dominated[u]) {
    // block A row: coords with last coord replaced by z[d-1]
    long long row = slotA[u];
    for (int k = 0; k < d; ++k) {
      out_coords[row * d + k] = (k == d - 1) ? z[k] : coords[u * d + k];
      out_defs[row * d + k] = (k == d - 1) ? point_idx : defs[u * d + k];
    }
    for (int j = 0; j < d - 1; ++j) {
      if (!okj[u * d + j]) continue;
      row = baseBj[j] + slotBj[u * d + j];
      for (int k = 0; k < d; ++k) {
        out_coords[row * d + k] = (k == j) ? z[k] : coords[u * d + k];
        out_defs[row * d + k] = (k == j) ? point_idx : defs[u * d + k];
      }
    }
  } else {
    // survivor rank = u - (#dominated before u); slotA is the exclusive
    // cumsum of the dominated flags
    const long long srow = baseK + (u - slotA[u]);
    for (int k = 0; k < d; ++k) {
      out_coords[srow * d + k] = coords[u * d + k];
      out_defs[srow * d + k] = defs[u * d + k];
    }
  }
}

// Box volumes per UB (eq. 2 of Lacour et al.): vol_u = (ref_0 - C[u,0,0]) *
// prod_{j>=1} (coords[u,j] - max_{k<j} C[u,k,j]), zero if any factor <= 0.
extern "C" __global__ void lacour_volumes_kernel(
    const double* __restrict__ coords, const long long* __restrict__ defs,
    const double* __restrict__ pts_aug, const double* __restrict__ ref,
    double* __restrict__ vol, int U, int d) {
  const int u = blockIdx.x * blockDim.x + threadIdx.x;
  if (u >= U) return;
  double v = ref[0] - pts_aug[defs[u * d + 0] * d + 0];
  bool ok = v > 0.0;
  for (int j = 1; j < d && ok; ++j) {
    double mx = -HUGE_VAL;
    for (int k = 0; k < j; ++k)
      mx = fmax(mx, pts_aug[defs[u * d + k] * d + j]);
    const double lj = coords[u * d + j] - mx;
    ok = lj > 0.0;
    v *= lj;
  }
  vol[u] = ok ? v : 0.0;
}

// ------------------------------------------------------------ launch shims
extern "C" void launch_hv2d(const double* P, const double* ref, double* out,
                            int n, hipStream_t s) {
  int M = 1;
  while (M < n) M <<= 1;
  const int threads = 1024;
  // LDS: xs[M] + ys[M] + scratch[M]
  const size_t lds = (size_t)(3 * M) * sizeof(double);
  hipLaunchKernelGGL(hv2d_kernel, dim3(1), dim3(threads), lds, s, P, ref, out,
                     n);
}

extern "C" void launch_hv3d_slices(const double* Px, const double* z_thr,
                                   const double* dz, const double* ref,
                                   double* out, int n, hipStream_t s) {
  hipLaunchKernelGGL(hv3d_slices_kernel, dim3(n), dim3(WAVE_SIZE), 0, s, Px,
                     z_thr, dz, ref, out, n);
}

extern "C" void launch_ehvi(const double* L, const double* U, const double* mu,
                            const double* var, double* out, int B, int nb,
                            int d, hipStream_t s) {
  const int threads = 256;
  const int blocks = (B + threads - 1) / threads;
  hipLaunchKernelGGL(ehvi_kernel, dim3(blocks), dim3(threads), 0, s, L, U, mu,
                     var, out, B, nb, d);
}

extern "C" void launch_lacour_flags(const double* coords,
                                    const long long* defs,
                                    const double* pts_aug, const double* z,
                                    unsigned char* dominated,
                                    unsigned char* okj, int U, int d,
                                    hipStream_t s) {
  const int threads = 256;
  const int blocks = (U + threads - 1) / threads;
  hipLaunchKernelGGL(lacour_flags_kernel, dim3(blocks), dim3(threads), 0, s,
                     coords, defs, pts_aug, z, dominated, okj, U, d);
}

extern "C" void launch_lacour_scatter(
    const double* coords, const long long* defs, const double* z,
    const unsigned char* dominated, const unsigned char* okj,
    const long long* slotA, const long long* slotBj, const long long* baseBj,
    long long baseK, long long point_idx, double* out_coords,
    long long* out_defs, int U, int d, hipStream_t s) {
  const int threads = 256;
  const int blocks = (U + threads - 1) / threads;
  hipLaunchKernelGGL(lacour_scatter_kernel, dim3(blocks), dim3(threads), 0, s,
                     coords, defs, z, dominated, okj, slotA, slotBj, baseBj,
                     baseK, point_idx, out_coords, out_defs, U, d);
}

extern "C" void launch_lacour_volumes(const double* coords,
                                      const long long* defs,
                                      const double* pts_aug, const double* ref,
                                      double* vol, int U, int d,
                                      hipStream_t s) {
  const int threads = 256;
  const int blocks = (U + threads - 1) / threads;
  hipLaunchKernelGGL(lacour_volumes_kernel, dim3(blocks), dim3(threads), 0, s,
                     coords, defs, pts_aug, ref, vol, U, d);
}
