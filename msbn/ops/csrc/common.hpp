// Common device helpers for the msbn gfx950 (CDNA4) BatchNorm kernels.
//
// Design notes (MI355X-first, per /opt/skills/guides/cdna_hip_programming.md):
//  * wave64 is the scheduling quantum: all shuffle reductions are width-64.
//  * memory-bound kernels vectorize loads to 16 B/lane (Pack<T,V>).
//  * statistics accumulate in fp64 per-thread and across the two-stage
//    (partial -> finalize) reduction: deterministic (no atomics) and immune
//    to the sum-of-squares cancellation that plagues fp32 E[x^2] - E[x]^2.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define MSBN_WAVE 64
#define MSBN_BLOCK 256

namespace msbn {

// ---------------------------------------------------------------- vector pack
template <typename T, int V>
struct alignas(sizeof(T) * V) Pack {
  T v[V];
};

// ------------------------------------------------------------- dtype convert
__device__ __forceinline__ float to_f(float x) { return x; }
__device__ __forceinline__ float to_f(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ float to_f(__half x) { return __half2float(x); }

template <typename T>
__device__ __forceinline__ T from_f(float x);
template <>
__device__ __forceinline__ float from_f<float>(float x) {
  return x;
}
template <>
__device__ __forceinline__ __hip_bfloat16 from_f<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <>
__device__ __forceinline__ __half from_f<__half>(float x) {
  return __float2half(x);
}

// --------------------------------------------------------- wave64 reductions
__device__ __forceinline__ void wave_reduce_pair(double& a, double& b) {
#pragma unroll
  for (int off = MSBN_WAVE / 2; off > 0; off >>= 1) {
    a += __shfl_down(a, off, MSBN_WAVE);
    b += __shfl_down(b, off, MSBN_WAVE);
  }
}

// Block-level {sum, sumsq} reduction: wave64 shuffle tree, then LDS partials
// (one double2 per wave), wave 0 combines.  Result valid on thread 0 only.
// lds must hold 2 * (blockDim.x/64) doubles.
__device__ __forceinline__ void block_reduce_pair(double& a, double& b,
                                                  double* lds) {
  wave_reduce_pair(a, b);
  const int lane = threadIdx.x & (MSBN_WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int nw = blockDim.x >> 6;
  if (lane == 0) {
    lds[2 * wid] = a;
    lds[2 * wid + 1] = b;
  }
  __syncthreads();
  if (wid == 0) {
    a = (lane < nw) ? lds[2 * lane] : 0.0;
    b = (lane < nw) ? lds[2 * lane + 1] : 0.0;
    wave_reduce_pair(a, b);
  }
}

__device__ __forceinline__ int64_t i64min(int64_t a, int64_t b) {
  return a < b ? a : b;
}

}  // namespace msbn
