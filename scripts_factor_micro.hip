// Micro-benchmark: what costs 2000 cycles per column in the panel factor?
// Variants of the 32-column diagonal factor, timed with wall clock around
// 2000 back-to-back launches (B=12 blocks, one wave active per block).
//
// build+run (GPU box):
//   hipcc --offload-arch=gfx950 -O3 scripts_factor_micro.hip -o /tmp/fm && /tmp/fm
#include <hip/hip_runtime.h>
#include <math.h>
#include <stdio.h>

#define BS 32

__device__ __forceinline__ float shfl_f(float v, int src) {
  return __shfl(v, src, 64);
}

// V0: full 2-col grouped factor (mirror of chol_panel_kernel's loop)
// V1: no __threadfence_block
// V2: no LDS broadcast at all (results wrong; rank-2 uses stale garbage)
// V3: no sqrt/div/log (linear ops only)
// V4: rounds without rank-2 update tail
template <int VARIANT>
__global__ __launch_bounds__(384) void factor_kernel(const float* __restrict__ A,
                                                     float* __restrict__ out,
                                                     int iters) {
  __shared__ float S[BS][BS + 1];
  __shared__ float colbuf[2][BS];
  const int tid = threadIdx.x;
  const int b = blockIdx.x;
  for (int idx = tid; idx < BS * BS; idx += blockDim.x)
    S[idx / BS][idx % BS] = A[b * BS * BS + idx];
  __syncthreads();
  float acc = 0.f;
  for (int it = 0; it < iters; ++it) {
    if (tid < 64) {
      const int lane = tid;
      float r[BS];
#pragma unroll
      for (int t = 0; t < BS; ++t) r[t] = S[lane][t] + it * 1e-9f;
      float mylog = 0.0f;
#pragma unroll
      for (int g = 0; g < BS; g += 2) {
#pragma unroll
        for (int q = 0; q < 2; ++q) {
          const int j = g + q;
          if (q == 1 && VARIANT != 2) {
            if (lane >= j) r[j] = fmaf(-r[g], colbuf[0][j], r[j]);
          }
          float d = shfl_f(r[j], j);
          if (VARIANT == 3) {
            d = d + 1.f;
            if (lane == j) r[j] = d;
            else if (lane > j) r[j] *= d;
          } else {
            if (d <= 0.0f || !isfinite(d)) d = 1e-30f;
            d = sqrtf(d);
            if (lane == j) { r[j] = d; mylog += logf(d); }
            else if (lane > j) r[j] /= d;
          }
          if (VARIANT != 2) {
            if (lane < BS) colbuf[q][lane] = r[j];
            if (VARIANT != 1) __threadfence_block();
          }
        }
        if (VARIANT != 4) {
#pragma unroll
          for (int t = 0; t < BS; ++t) {
            if (t < g + 2) continue;
            if (lane >= t) {
#pragma unroll
              for (int q = 0; q < 2; ++q) {
                const float c = (VARIANT == 2) ? shfl_f(r[g + q], t) : colbuf[q][t];
                r[t] = fmaf(-r[g + q], c, r[t]);
              }
            }
          }
        }
      }
      acc += r[lane & (BS - 1)] + mylog;
    }
    __syncthreads();
  }
  if (tid == 0) out[b] = acc;
}

int main() {
  const int B = 12, iters = 2000;
  float *A, *out;
  hipMalloc(&A, B * BS * BS * sizeof(float));
  hipMalloc(&out, B * sizeof(float));
  float h[B * BS * BS];
  for (int i = 0; i < B * BS * BS; ++i) h[i] = 10.0f + (i % 7) * 0.01f;
  hipMemcpy(A, h, sizeof(h), hipMemcpyHostToDevice);
  const char* names[] = {"full(2-col LDS+fence)", "no fence", "shfl only (no LDS)",
                         "no sqrt/div/log", "no rank-2 tail"};
  for (int v = 0; v <= 4; ++v) {
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    // warmup
    switch (v) {
      case 0: hipLaunchKernelGGL(factor_kernel<0>, dim3(B), dim3(384), 0, 0, A, out, 10); break;
      case 1: hipLaunchKernelGGL(factor_kernel<1>, dim3(B), dim3(384), 0, 0, A, out, 10); break;
      case 2: hipLaunchKernelGGL(factor_kernel<2>, dim3(B), dim3(384), 0, 0, A, out, 10); break;
      case 3: hipLaunchKernelGGL(factor_kernel<3>, dim3(B), dim3(384), 0, 0, A, out, 10); break;
      case 4: hipLaunchKernelGGL(factor_kernel<4>, dim3(B), dim3(384), 0, 0, A, out, 10); break;
    }
    hipDeviceSynchronize();
    hipEventRecord(e0);
    switch (v) {
      case 0: hipLaunchKernelGGL(factor_kernel<0>, dim3(B), dim3(384), 0, 0, A, out, iters); break;
      case 1: hipLaunchKernelGGL(factor_kernel<1>, dim3(B), dim3(384), 0, 0, A, out, iters); break;
      case 2: hipLaunchKernelGGL(factor_kernel<2>, dim3(B), dim3(384), 0, 0, A, out, iters); break;
      case 3: hipLaunchKernelGGL(factor_kernel<3>, dim3(B), dim3(384), 0, 0, A, out, iters); break;
      case 4: hipLaunchKernelGGL(factor_kernel<4>, dim3(B), dim3(384), 0, 0, A, out, iters); break;
    }
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms = 0;
    hipEventElapsedTime(&ms, e0, e1);
    printf("V%d %-22s: %7.3f us per 32-col factor\n", v, names[v], ms * 1000.0f / iters);
  }
  return 0;
}
