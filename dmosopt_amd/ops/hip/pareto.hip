// Pareto ranking (Dominance Degree Matrix) and crowding distance.
//
// Replaces reference dda.py:25-120 and indicators.py:12-51.
//
// pareto_rank design (three size regimes, dispatched in bindings.cpp):
//   N <= ~2048: bit-matrix path — dom_bits_kernel packs "i dominates j"
//     grid-wide into 32 bits/word, then peel_bits_kernel peels all fronts
//     in one small block with popcount(dom_mask & front_mask);
//   fallbacks: single-launch from-Y peel (LDS-staged objectives), and for
//     large N the DDA matrix (tile-wise, 32-row objective slabs in LDS)
//     with chased peel_front/commit_front launches or a matvec
//     dominator-count peel driven from the host.
//
// crowding design: one workgroup per objective dimension; (value, index)
// pairs bitonic-sorted in LDS (column min/max read off the sorted ends);
// per-dim gap rows are written WITHOUT atomics and summed in fixed order
// so results are bit-identical across replicated ranks.

#include "common.h"
#include <math.h>

#define PTILE 32
#define PTPB 256

__global__ void dominance_matrix_kernel(const float* __restrict__ Y,  // (N, m)
                                        int* __restrict__ D,          // (N, N)
                                        int N, int m) {
  extern __shared__ float lds[];  // [2][PTILE][m]
  float* yi = lds;                // rows i (tile_y)
  float* yj = lds + PTILE * m;    // rows j (tile_x)
  const int ti = blockIdx.y * PTILE;
  const int tj = blockIdx.x * PTILE;
  for (int idx = threadIdx.x; idx < PTILE * m; idx += PTPB) {
    const int r = idx / m, c = idx % m;
    yi[idx] = (ti + r < N) ? Y[(ti + r) * m + c] : 0.f;
    yj[idx] = (tj + r < N) ? Y[(tj + r) * m + c] : 0.f;
  }
  __syncthreads();
  const int ty = threadIdx.x / 16, tx = threadIdx.x % 16;
#pragma unroll
  for (int sy = 0; sy < 2; ++sy)
#pragma unroll
    for (int sx = 0; sx < 2; ++sx) {
      const int li = ty * 2 + sy, lj = tx * 2 + sx;
      const int gi = ti + li, gj = tj + lj;
      if (gi >= N || gj >= N) continue;
      int cnt = 0;
      for (int k = 0; k < m; ++k)
        cnt += (yi[li * m + k] <= yj[lj * m + k]) ? 1 : 0;
      D[(long long)gi * N + gj] = cnt;
    }
}

// Zero mutual-domination entries of identical rows: D[i][j] = 0 where
// D[i][j] == m and D[j][i] == m (reference dda.py:47-49). Includes the
// diagonal (a row is always 'identical' to itself) — without this every
// column max is >= m and no front would ever peel.
__global__ void zero_identical_kernel(int* __restrict__ D, int N, int m) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)N * N) return;
  const int i = (int)(idx / N), j = (int)(idx % N);
  if (i <= j && D[(long long)i * N + j] == m && D[(long long)j * N + i] == m) {
    D[(long long)i * N + j] = 0;
    D[(long long)j * N + i] = 0;
  }
}

// One peel pass: for each alive column j compute max_i(D[i][j]) over alive
// rows; columns with max < m belong to the current front.
__global__ void peel_front_kernel(const int* __restrict__ D,
                                  const unsigned char* __restrict__ alive,
                                  unsigned char* __restrict__ front,  // out
                                  int* __restrict__ n_front,          // out
                                  int N, int m) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= N) return;
  unsigned char f = 0;
  if (alive[j]) {
    int mx = -1;
    const int* col = D + j;
    for (int i = 0; i < N; ++i)
      if (alive[i]) mx = max(mx, col[(long long)i * N]);
    f = (mx < m) ? 1 : 0;
  }
  front[j] = f;
  if (f) atomicAdd(n_front, 1);
}

__global__ void commit_front_kernel(const unsigned char* __restrict__ front,
                                    unsigned char* __restrict__ alive,
                                    int* __restrict__ rank, int k, int N) {
  const int j = blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= N) return;
  if (front[j]) {
    rank[j] = k;
    alive[j] = 0;
  }
}

// Whole peel in ONE launch for N <= PEEL1_NMAX: a single 1024-thread block
// computes dominator counts once (O(N^2) strided) and then peels fronts
// with O(front * N) updates per front — replacing ~2 launches per front
// (~45 us each at low occupancy) with one ~tens-of-us kernel.
#define PEEL1_NMAX 2048
#define PEEL1_TPB 1024

__global__ __launch_bounds__(PEEL1_TPB) void peel_single_block_kernel(
    const int* __restrict__ D,  // (N, N) dominance degrees, identicals zeroed
    int* __restrict__ rank,     // (N,) out
    int N, int m) {
  __shared__ int n_dom[PEEL1_NMAX];
  __shared__ int front[PEEL1_NMAX];
  __shared__ int front_sz;
  __shared__ int remaining;
  const int tid = threadIdx.x;

  for (int j = tid; j < N; j += PEEL1_TPB) {
    int cnt = 0;
    for (int i = 0; i < N; ++i) cnt += (D[(long long)i * N + j] == m) ? 1 : 0;
    n_dom[j] = cnt;
  }
  if (tid == 0) remaining = N;
  __syncthreads();

  for (int k = 0; remaining > 0 && k <= N; ++k) {
    if (tid == 0) front_sz = 0;
    __syncthreads();
    for (int j = tid; j < N; j += PEEL1_TPB) {
      if (n_dom[j] == 0) {
        rank[j] = k;
        n_dom[j] = -1;  // peeled
        front[atomicAdd(&front_sz, 1)] = j;
      }
    }
    __syncthreads();
    const int fs = front_sz;
    if (fs == 0) break;  // safety
    // subtract peeled rows' domination counts: threads cover (f, j) pairs
    for (long long t = tid; t < (long long)fs * N; t += PEEL1_TPB) {
      const int f = front[t / N];
      const int j = (int)(t % N);
      if (n_dom[j] > 0 && D[(long long)f * N + j] == m) atomicSub(&n_dom[j], 1);
    }
    __syncthreads();
    if (tid == 0) remaining -= fs;
    __syncthreads();
  }
}

extern "C" void launch_peel_single_block(const int* D, int* rank, int N, int m,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(peel_single_block_kernel, dim3(1), dim3(PEEL1_TPB), 0,
                     stream, D, rank, N, m);
}

// Whole ranking from Y in ONE launch: Y (N x m) is staged in LDS and
// dominance is recomputed on the fly — the (N x N) int32 dominance matrix
// (640 KB at N=400) never exists, so the single CU running this block does
// LDS compares instead of pulling the matrix from HBM twice. Semantics
// identical to dominance_degree_matrix + zero_identical + peel: i dominates
// j iff all objectives <= and not all equal.
template <int TPB>
__global__ __launch_bounds__(TPB) void peel_from_y_kernel(
    const float* __restrict__ Y, int* __restrict__ rank, int N, int m) {
  extern __shared__ char sh_raw[];
  float* Ys = (float*)sh_raw;                 // N * m
  int* n_dom = (int*)(Ys + (size_t)N * m);    // N
  int* front = n_dom + N;                     // N
  int* ctrl = front + N;                      // [front_sz, remaining]
  const int tid = threadIdx.x;

  for (int i = tid; i < N * m; i += TPB) Ys[i] = Y[i];
  for (int j = tid; j < N; j += TPB) n_dom[j] = 0;
  if (tid == 0) ctrl[1] = N;
  __syncthreads();

  // dominator counts: thread OWNS columns j (register accumulate — no
  // atomics, no 64-bit div/mod in the pair loop)
  for (int j = tid; j < N; j += TPB) {
    int cnt = 0;
    for (int i = 0; i < N; ++i) {
      bool le = true, lt = false;
      for (int t = 0; t < m; ++t) {
        const float a = Ys[i * m + t], b = Ys[j * m + t];
        le &= (a <= b);
        lt |= (a < b);
      }
      cnt += (le && lt) ? 1 : 0;
    }
    n_dom[j] = cnt;
  }
  __syncthreads();

  for (int k = 0; ctrl[1] > 0 && k <= N; ++k) {
    if (tid == 0) ctrl[0] = 0;
    __syncthreads();
    for (int j = tid; j < N; j += TPB) {
      if (n_dom[j] == 0) {
        rank[j] = k;
        n_dom[j] = -1;
        front[atomicAdd(&ctrl[0], 1)] = j;
      }
    }
    __syncthreads();
    const int fs = ctrl[0];
    if (fs == 0) break;
    // column-owned update: subtract each peeled row's domination of j
    for (int j = tid; j < N; j += TPB) {
      if (n_dom[j] <= 0) continue;
      int dec = 0;
      for (int q = 0; q < fs; ++q) {
        const int f = front[q];
        bool le = true, lt = false;
        for (int c = 0; c < m; ++c) {
          const float a = Ys[f * m + c], b = Ys[j * m + c];
          le &= (a <= b);
          lt |= (a < b);
        }
        dec += (le && lt) ? 1 : 0;
      }
      n_dom[j] -= dec;
    }
    __syncthreads();
    if (tid == 0) ctrl[1] -= fs;
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Bit-matrix ranking: dominators of column j packed 32/word, built GRID-WIDE
// (the count pass was the single-CU bottleneck of the one-launch peel), then
// a single small block peels fronts with popcount(dom_mask & front_mask) —
// each round costs ~N/TPB * N/32 word-ops instead of N^2/TPB pair compares.
// M > 0: compile-time objective count (fully unrolled compare, row j
// cached in registers instead of re-read from L2 for every word); M == 0:
// generic runtime-m fallback.
template <int M>
__global__ void dom_bits_kernel(const float* __restrict__ Y,
                                unsigned int* __restrict__ Dbits,  // (N, W)
                                int N, int m, int W) {
  const long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long long)N * W) return;
  const int j = (int)(idx / W), w = (int)(idx % W);
  unsigned int bits = 0;
  const int i0 = w * 32;
  const int iend = min(32, N - i0);
  if (M > 0) {
    float cj[M > 0 ? M : 1];
#pragma unroll
    for (int t = 0; t < M; ++t) cj[t] = Y[j * M + t];
    for (int b = 0; b < iend; ++b) {
      const int i = i0 + b;
      bool le = true, lt = false;
#pragma unroll
      for (int t = 0; t < M; ++t) {
        const float a = Y[i * M + t];
        le &= (a <= cj[t]);
        lt |= (a < cj[t]);
      }
      if (le && lt && i != j) bits |= (1u << b);
    }
  } else {
    for (int b = 0; b < iend; ++b) {
      const int i = i0 + b;
      if (i == j) continue;
      bool le = true, lt = false;
      for (int t = 0; t < m; ++t) {
        const float a = Y[i * m + t], c = Y[j * m + t];
        le &= (a <= c);
        lt |= (a < c);
      }
      if (le && lt) bits |= (1u << b);
    }
  }
  Dbits[idx] = bits;
}

#define PEELB_TPB 256

__global__ __launch_bounds__(PEELB_TPB) void peel_bits_kernel(
    const unsigned int* __restrict__ Dbits, int* __restrict__ rank, int N,
    int W) {
  extern __shared__ char sh_raw[];
  unsigned int* Db = (unsigned int*)sh_raw;      // N * W
  unsigned int* fmask = Db + (size_t)N * W;      // W
  int* n_dom = (int*)(fmask + W);                // N
  int* ctrl = n_dom + N;                         // [front_sz, remaining]
  const int tid = threadIdx.x;

  for (int i = tid; i < N * W; i += PEELB_TPB) Db[i] = Dbits[i];
  if (tid == 0) ctrl[1] = N;
  __syncthreads();
  for (int j = tid; j < N; j += PEELB_TPB) {
    int c = 0;
    for (int w = 0; w < W; ++w) c += __popc(Db[j * W + w]);
    n_dom[j] = c;
  }
  __syncthreads();

  for (int k = 0; ctrl[1] > 0 && k <= N; ++k) {
    if (tid == 0) ctrl[0] = 0;
    for (int w = tid; w < W; w += PEELB_TPB) fmask[w] = 0u;
    __syncthreads();
    for (int j = tid; j < N; j += PEELB_TPB) {
      if (n_dom[j] == 0) {
        rank[j] = k;
        n_dom[j] = -1;
        atomicOr(&fmask[j >> 5], 1u << (j & 31));
        atomicAdd(&ctrl[0], 1);
      }
    }
    __syncthreads();
    const int fs = ctrl[0];
    if (fs == 0) break;
    for (int j = tid; j < N; j += PEELB_TPB) {
      if (n_dom[j] <= 0) continue;
      int dec = 0;
      for (int w = 0; w < W; ++w) dec += __popc(Db[j * W + w] & fmask[w]);
      n_dom[j] -= dec;
    }
    __syncthreads();
    if (tid == 0) ctrl[1] -= fs;
    __syncthreads();
  }
}

#define LAUNCH_DOM_BITS(blocks_, stream_, Y_, D_, N_, m_, W_)                  do {                                                                           if (m_ == 2)                                                                   hipLaunchKernelGGL(dom_bits_kernel<2>, dim3(blocks_), dim3(256), 0,                               stream_, Y_, D_, N_, m_, W_);                          else if (m_ == 3)                                                              hipLaunchKernelGGL(dom_bits_kernel<3>, dim3(blocks_), dim3(256), 0,                               stream_, Y_, D_, N_, m_, W_);                          else if (m_ == 5)                                                              hipLaunchKernelGGL(dom_bits_kernel<5>, dim3(blocks_), dim3(256), 0,                               stream_, Y_, D_, N_, m_, W_);                          else                                                                           hipLaunchKernelGGL(dom_bits_kernel<0>, dim3(blocks_), dim3(256), 0,                               stream_, Y_, D_, N_, m_, W_);                        } while (0)

extern "C" int launch_peel_bits(const float* Y, unsigned int* Dbits_scratch,
                                int* rank, int N, int m, hipStream_t stream) {
  const int W = (N + 31) / 32;
  const size_t lds = ((size_t)N * W + W) * sizeof(unsigned int) +
                     (N + 2) * sizeof(int);
  if (lds > 144 * 1024) return -1;
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)peel_bits_kernel,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    attr_set = true;
  }
  const long long total = (long long)N * W;
  LAUNCH_DOM_BITS(((int)((total + 255) / 256)), stream, Y, Dbits_scratch, N,
                  m, W);
  hipLaunchKernelGGL(peel_bits_kernel, dim3(1), dim3(PEELB_TPB), lds, stream,
                     Dbits_scratch, rank, N, W);
  return 0;
}

// ---------------------------------------------------- cooperative peel
// Grid-wide SYNC-FREE-FROM-HOST ranking for N beyond the one-workgroup
// kernels' capacity (and their one-CU serialization): the packed dominator
// bit-matrix lives in GLOBAL memory (L2-resident: N=8192 -> 8 MB), every
// CU participates, and the front-peel loop synchronizes with grid.sync()
// instead of host readbacks (the chased matvec path's .item() every 16
// fronts stalls pipelined generation loops — NOTES.md).
#include <hip/hip_cooperative_groups.h>

__global__ __launch_bounds__(256) void coop_peel_bits_kernel(
    const unsigned int* __restrict__ Dbits,  // (N, W) i-dominates-j bits
    unsigned int* __restrict__ fmask,        // (3*W,) triple-buffered scratch
    int* __restrict__ n_dom,                 // (N,) scratch
    int* __restrict__ ctrl,                  // (2,) scratch (only [0] used)
    int* __restrict__ rank,                  // (N,) out
    int N, int W, int stop) {
  // ONE grid.sync per front round (the peel is grid-barrier-latency bound):
  // the detect step is fused into the update step — when a point's
  // dominator count hits zero it immediately stamps its rank and flags
  // itself in the NEXT round's front mask. The mask is TRIPLE-buffered:
  // round k reads buf[k%3], writes buf[(k+1)%3], and clears buf[(k+2)%3]
  // (== the buffer round k-1 read, dead since the last barrier), so no
  // round ever waits on a separate clear phase.
  //
  // stop: peel until >= stop points are ranked, then stamp the remaining
  // alive points with a SENTINEL rank (last completed front + 1) and exit.
  // Exact for truncation selection (nsga2_select): the straddling front is
  // always completed before stopping, so every point that can enter the
  // kept top-`stop` carries its true rank; points beyond are discarded by
  // the caller and only need a rank larger than any kept one. stop >= N
  // peels everything (the public pareto_rank default).
  cooperative_groups::grid_group grid = cooperative_groups::this_grid();
  const int gtid = blockIdx.x * blockDim.x + threadIdx.x;
  const int gsize = gridDim.x * blockDim.x;

  for (int j = gtid; j < N; j += gsize) {
    int c = 0;
    const unsigned int* row = Dbits + (size_t)j * W;
    for (int w = 0; w < W; ++w) c += __popc(row[w]);
    n_dom[j] = c;
    rank[j] = 0;
  }
  for (int w = gtid; w < 3 * W; w += gsize) fmask[w] = 0u;
  if (gtid == 0) ctrl[0] = 0;
  grid.sync();
  // front 0 into buffer 0
  for (int j = gtid; j < N; j += gsize) {
    if (n_dom[j] == 0) {
      n_dom[j] = -1;
      atomicOr(&fmask[j >> 5], 1u << (j & 31));
      atomicAdd(&ctrl[0], 1);
    }
  }
  grid.sync();

  int prev = -1;
  for (int k = 0; k < N; ++k) {
    const int total = ctrl[0];  // uniform: all writes pre-barrier
    if (total >= N || total == prev) break;
    if (total >= stop) {  // fronts 0..k are complete
      for (int j = gtid; j < N; j += gsize)
        if (n_dom[j] >= 0) rank[j] = k + 1;  // sentinel > every true rank
      break;
    }
    prev = total;
    const unsigned int* rd = fmask + (size_t)(k % 3) * W;
    unsigned int* wr = fmask + (size_t)((k + 1) % 3) * W;
    unsigned int* cl = fmask + (size_t)((k + 2) % 3) * W;
    for (int w = gtid; w < W; w += gsize) cl[w] = 0u;
    for (int j = gtid; j < N; j += gsize) {
      if (n_dom[j] <= 0) continue;
      int dec = 0;
      const unsigned int* row = Dbits + (size_t)j * W;
      for (int w = 0; w < W; ++w) dec += __popc(row[w] & rd[w]);
      if (dec) {
        const int nd = n_dom[j] - dec;
        if (nd == 0) {
          rank[j] = k + 1;
          n_dom[j] = -1;
          atomicOr(&wr[j >> 5], 1u << (j & 31));
          atomicAdd(&ctrl[0], 1);
        } else {
          n_dom[j] = nd;
        }
      }
    }
    grid.sync();
  }
}

extern "C" int launch_coop_peel(const float* Y, unsigned int* Dbits,
                                unsigned int* fmask, int* n_dom, int* ctrl,
                                int* rank, int N, int m, int stop,
                                hipStream_t stream) {
  const int W = (N + 31) / 32;
  if (stop <= 0 || stop > N) stop = N;
  static int coop_ok = -1;
  if (coop_ok < 0) {
    int dev = 0, attr = 0;
    hipGetDevice(&dev);
    hipDeviceGetAttribute(&attr, hipDeviceAttributeCooperativeLaunch, dev);
    coop_ok = attr ? 1 : 0;
  }
  if (!coop_ok) return -1;
  static int max_blocks = -1;
  if (max_blocks < 0) {
    int per_cu = 0;
    hipOccupancyMaxActiveBlocksPerMultiprocessor(
        &per_cu, (const void*)coop_peel_bits_kernel, 256, 0);
    hipDeviceProp_t prop;
    int dev = 0;
    hipGetDevice(&dev);
    hipGetDeviceProperties(&prop, dev);
    max_blocks = per_cu * prop.multiProcessorCount;
    if (max_blocks < 1) max_blocks = 1;
  }
  const long long total = (long long)N * W;
  LAUNCH_DOM_BITS(((int)((total + 255) / 256)), stream, Y, Dbits, N, m, W);
  int blocks = (N + 255) / 256;
  if (blocks > max_blocks) blocks = max_blocks;
  void* args[] = {(void*)&Dbits, (void*)&fmask, (void*)&n_dom,
                  (void*)&ctrl, (void*)&rank, (void*)&N, (void*)&W,
                  (void*)&stop};
  const hipError_t err = hipLaunchCooperativeKernel(
      (const void*)coop_peel_bits_kernel, dim3(blocks), dim3(256),
      args, 0, stream);
  return err == hipSuccess ? 0 : -1;
}

extern "C" int launch_peel_from_y(const float* Y, int* rank, int N, int m,
                                  hipStream_t stream) {
  const size_t lds = (size_t)N * m * sizeof(float) + (2 * N + 2) * sizeof(int);
  if (lds > 144 * 1024) return -1;  // caller falls back to the D-matrix path
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute((const void*)peel_from_y_kernel<64>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    hipFuncSetAttribute((const void*)peel_from_y_kernel<256>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    hipFuncSetAttribute((const void*)peel_from_y_kernel<PEEL1_TPB>,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 144 * 1024);
    attr_set = true;
  }
  // a converged population peels into MANY single-digit fronts: the round
  // loop is then bound by block-barrier latency, which for a 16-wave block
  // is ~1 us x ~3 barriers x #fronts. A single-wave block makes barriers
  // ~free; its lower count-pass parallelism only matters for large N.
  const char* tpb_env = getenv("DMOSOPT_PEEL_TPB");
  const int tpb = tpb_env ? atoi(tpb_env) : (N <= 1024 ? 256 : PEEL1_TPB);
  if (tpb <= 64)
    hipLaunchKernelGGL(peel_from_y_kernel<64>, dim3(1), dim3(64), lds, stream,
                       Y, rank, N, m);
  else if (tpb <= 256)
    hipLaunchKernelGGL(peel_from_y_kernel<256>, dim3(1), dim3(256), lds,
                       stream, Y, rank, N, m);
  else
    hipLaunchKernelGGL(peel_from_y_kernel<PEEL1_TPB>, dim3(1), dim3(PEEL1_TPB),
                       lds, stream, Y, rank, N, m);
  return 0;
}

extern "C" void launch_dominance_matrix(const float* Y, int* D, int N, int m,
                                        hipStream_t stream) {
  dim3 grid((N + PTILE - 1) / PTILE, (N + PTILE - 1) / PTILE);
  size_t lds_bytes = 2 * PTILE * m * sizeof(float);
  hipLaunchKernelGGL(dominance_matrix_kernel, grid, dim3(PTPB), lds_bytes,
                     stream, Y, D, N, m);
  long long total = (long long)N * N;
  int blocks = (int)((total + 255) / 256);
  hipLaunchKernelGGL(zero_identical_kernel, dim3(blocks), dim3(256), 0, stream,
                     D, N, m);
}

extern "C" void launch_peel_front(const int* D, const unsigned char* alive,
                                  unsigned char* front, int* n_front, int N,
                                  int m, hipStream_t stream) {
  int blocks = (N + 255) / 256;
  hipLaunchKernelGGL(peel_front_kernel, dim3(blocks), dim3(256), 0, stream, D,
                     alive, front, n_front, N, m);
}

extern "C" void launch_commit_front(const unsigned char* front,
                                    unsigned char* alive, int* rank, int k,
                                    int N, hipStream_t stream) {
  int blocks = (N + 255) / 256;
  hipLaunchKernelGGL(commit_front_kernel, dim3(blocks), dim3(256), 0, stream,
                     front, alive, rank, k, N);
}

// ------------------------------------------------------------- crowding
// One workgroup per objective dimension. Bitonic sort of (normalized value,
// index) in LDS; N padded to the next power of two with +inf sentinels.
#define CROWD_TPB 256

template <int TPB>
__global__ __launch_bounds__(TPB) void crowding_kernel(
    const float* __restrict__ Y,  // (N, m)
    float* __restrict__ out,      // (m, N)
    int N, int m, int npow2) {
  extern __shared__ float sh[];
  float* vals = sh;                       // npow2
  int* idxs = (int*)(sh + npow2);         // npow2
  const int j = blockIdx.x;               // objective dim

  // RAW column values: sorting is affine-invariant, and after the sort the
  // column min/max are vals[0]/vals[N-1] — the former host-side min/max/
  // span prep (~6 torch launches per call) is free here.
  for (int i = threadIdx.x; i < npow2; i += TPB) {
    vals[i] = (i < N) ? Y[i * m + j] : INFINITY;
    idxs[i] = i;
  }
  __syncthreads();

  // bitonic sort ascending (stable order not required: equal normalized
  // values produce zero gaps either way)
  for (int ksz = 2; ksz <= npow2; ksz <<= 1) {
    for (int jsz = ksz >> 1; jsz > 0; jsz >>= 1) {
      for (int i = threadIdx.x; i < npow2; i += TPB) {
        const int ixj = i ^ jsz;
        if (ixj > i) {
          const bool up = ((i & ksz) == 0);
          const bool swap = up ? (vals[i] > vals[ixj]) : (vals[i] < vals[ixj]);
          if (swap) {
            float tv = vals[i]; vals[i] = vals[ixj]; vals[ixj] = tv;
            int tixx = idxs[i]; idxs[i] = idxs[ixj]; idxs[ixj] = tixx;
          }
        }
      }
      __syncthreads();
    }
  }

  // gaps: boundary = 1, interior = (vals[i+1] - vals[i-1]) / span.
  // Each dim writes its own row of out (m, N) — NO atomics, so the sum
  // order is fixed and results are bit-identical across replicated ranks
  // (float atomicAdd order would not be).
  const float l = vals[0];
  const float h = vals[N - 1];
  float s = h - l;
  if (!(s > 0.f) || !isfinite(s)) s = 1.f;
  for (int i = threadIdx.x; i < N; i += TPB) {
    float d = (i == 0 || i == N - 1) ? 1.f : (vals[i + 1] - vals[i - 1]) / s;
    if (isnan(d)) d = 0.f;
    out[(long long)j * N + idxs[i]] = d;
  }
}

// out must be an (m, N) buffer; the host sums over dim 0 deterministically.
extern "C" void launch_crowding(const float* Y, float* out, int N, int m,
                                hipStream_t stream) {
  int npow2 = 1;
  while (npow2 < N) npow2 <<= 1;
  size_t lds_bytes = npow2 * (sizeof(float) + sizeof(int));
  // wide blocks for big sorts: at npow2=4096 a 256-thread block walks 16
  // elements per bitonic pass; 1024 threads walk 4 (measured 208 us ->
  // see profiles/README). Small sorts keep 256 (idle waves only add
  // barrier latency).
  if (npow2 > 512)
    hipLaunchKernelGGL(crowding_kernel<1024>, dim3(m), dim3(1024), lds_bytes,
                       stream, Y, out, N, m, npow2);
  else
    hipLaunchKernelGGL(crowding_kernel<CROWD_TPB>, dim3(m), dim3(CROWD_TPB),
                       lds_bytes, stream, Y, out, N, m, npow2);
}
