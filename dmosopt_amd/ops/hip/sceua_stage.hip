// Fused SCE-UA CCE stage kernels.
//
// One CCE stage of the batched hyperparameter search (models/sceua.py,
// semantics of reference model.py:1419-1753) previously queued ~30 small
// torch elementwise/gather kernels around the batched NMLL evaluation.
// These two kernels replace that chain:
//   sceua_propose:  centroid + reflect(/oob random)/contract/random
//                   candidate assembly, Philox RNG in-kernel
//   sceua_accept:   sequential acceptance rule, worst-point replacement,
//                   per-complex re-sort, device-side icall accounting
// Every (stream, complex) pair is one small workgroup — the stage has
// S*G of them, so these kernels are latency-bound glue; the point is the
// ~28 fewer stream entries per stage, not their own throughput.

#include "common.h"
#include <math.h>

#define SCEUA_TPB 64

// cx: (S, G, npg, nopt) f32, cf: (S, G, npg) f32, lcs: (nps,) int32
// cand out: (3, S*G, nopt) f32 in reflect/contract/random order.
__global__ void sceua_propose_kernel(const float* __restrict__ cx,
                                     const int* __restrict__ lcs,
                                     const float* __restrict__ bl,
                                     const float* __restrict__ bu,
                                     float* __restrict__ cand, int S, int G,
                                     int npg, int nopt, int nps,
                                     unsigned long long seed) {
  const int sg = blockIdx.x;
  const int s = sg / G, g = sg % G;
  (void)s;
  const float* base = cx + ((long long)sg) * npg * nopt;
  const int wi = lcs[nps - 1];  // worst simplex position
  __shared__ int oob_flag;
  if (threadIdx.x == 0) oob_flag = 0;
  __syncthreads();

  float* c_ref = cand + (long long)sg * nopt;
  float* c_con = cand + ((long long)(S * G) + sg) * nopt;
  float* c_rnd = cand + ((long long)(2 * S * G) + sg) * nopt;

  // pass 1: centroid/reflect/contract + oob detection
  for (int d = threadIdx.x; d < nopt; d += SCEUA_TPB) {
    float ce = 0.f;
    for (int i = 0; i < nps - 1; ++i)
      ce += base[(long long)lcs[i] * nopt + d];
    ce /= (float)(nps - 1);
    const float sw = base[(long long)wi * nopt + d];
    const float ref = ce + (ce - sw);
    if (ref < bl[d] || ref > bu[d]) atomicOr(&oob_flag, 1);
    c_ref[d] = ref;
    c_con[d] = sw + 0.5f * (ce - sw);
  }
  __syncthreads();
  // pass 2: randoms (Philox keyed by (seed, sg, d)) + oob replacement
  for (int d = threadIdx.x; d < nopt; d += SCEUA_TPB) {
    const float span = bu[d] - bl[d];
    Philox4 r = philox4x32(seed, ((unsigned long long)sg << 32) | (unsigned)d);
    c_rnd[d] = bl[d] + span * u01(r.c0);
    if (oob_flag) c_ref[d] = bl[d] + span * u01(r.c1);
  }
}

// fall: (3, S*G) f32 NMLL values for reflect/contract/random candidates.
// act: (S,) int32. icall: (S,) int32 accumulated on device.
__global__ void sceua_accept_kernel(float* __restrict__ cx,
                                    float* __restrict__ cf,
                                    const float* __restrict__ cand,
                                    const float* __restrict__ fall,
                                    const int* __restrict__ lcs,
                                    const int* __restrict__ act,
                                    int* __restrict__ icall, int S, int G,
                                    int npg, int nopt, int nps) {
  extern __shared__ char sh_raw[];
  float* keys = (float*)sh_raw;      // npg
  int* ord = (int*)(keys + npg);     // npg
  float* rowbuf = (float*)(ord + npg);  // npg * nopt (sorted copy)

  const int sg = blockIdx.x;
  const int s = sg / G;
  const int wi = lcs[nps - 1];
  float* X = cx + (long long)sg * npg * nopt;
  float* F = cf + (long long)sg * npg;

  const float f_ref = fall[sg];
  const float f_con = fall[S * G + sg];
  const float f_rnd = fall[2 * S * G + sg];
  const float fw = F[wi];
  const bool use_con = f_ref > fw;
  const bool use_rnd = use_con && (f_con > fw);
  const float fnew = use_rnd ? f_rnd : (use_con ? f_con : f_ref);
  const float* snew = use_rnd ? (cand + ((long long)(2 * S * G) + sg) * nopt)
                      : use_con ? (cand + ((long long)(S * G) + sg) * nopt)
                                : (cand + (long long)sg * nopt);
  if (threadIdx.x == 0)
    atomicAdd(&icall[s], 1 + (use_con ? 1 : 0) + (use_rnd ? 1 : 0));
  if (!act[s]) return;

  // replace worst simplex point
  for (int d = threadIdx.x; d < nopt; d += blockDim.x) X[(long long)wi * nopt + d] = snew[d];
  if (threadIdx.x == 0) F[wi] = fnew;
  __syncthreads();

  // re-sort the complex ascending by objective (stable insertion order:
  // equal keys keep their index order, matching torch.argsort(stable))
  if (threadIdx.x == 0) {
    for (int i = 0; i < npg; ++i) {
      keys[i] = F[i];
      ord[i] = i;
    }
    for (int i = 1; i < npg; ++i) {
      const float kv = keys[i];
      const int ov = ord[i];
      int j = i - 1;
      while (j >= 0 && keys[j] > kv) {
        keys[j + 1] = keys[j];
        ord[j + 1] = ord[j];
        --j;
      }
      keys[j + 1] = kv;
      ord[j + 1] = ov;
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < npg * nopt; idx += blockDim.x)
    rowbuf[idx] = X[(long long)ord[idx / nopt] * nopt + idx % nopt];
  __syncthreads();
  for (int idx = threadIdx.x; idx < npg * nopt; idx += blockDim.x)
    X[idx] = rowbuf[idx];
  for (int i = threadIdx.x; i < npg; i += blockDim.x) F[i] = keys[i];
}

extern "C" void launch_sceua_propose(const float* cx, const int* lcs,
                                     const float* bl, const float* bu,
                                     float* cand, int S, int G, int npg,
                                     int nopt, int nps,
                                     unsigned long long seed,
                                     hipStream_t stream) {
  hipLaunchKernelGGL(sceua_propose_kernel, dim3(S * G), dim3(SCEUA_TPB), 0,
                     stream, cx, lcs, bl, bu, cand, S, G, npg, nopt, nps,
                     seed);
}

extern "C" int launch_sceua_accept(float* cx, float* cf, const float* cand,
                                   const float* fall, const int* lcs,
                                   const int* act, int* icall, int S, int G,
                                   int npg, int nopt, int nps,
                                   hipStream_t stream) {
  const size_t lds = (size_t)npg * sizeof(float) + npg * sizeof(int) +
                     (size_t)npg * nopt * sizeof(float);
  if (lds > 60 * 1024) return -1;  // caller uses the torch stage path
  hipLaunchKernelGGL(sceua_accept_kernel, dim3(S * G), dim3(SCEUA_TPB), lds,
                     stream, cx, cf, cand, fall, lcs, act, icall, S, G, npg,
                     nopt, nps);
  return 0;
}
