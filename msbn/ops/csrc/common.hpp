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

// ------------------------------------------------ nontemporal pack load/store
// Streaming tensors (activations / gradients: read or written exactly once
// per launch and far larger than L2) use nontemporal accesses — the nt cache
// bit keeps them from evicting the hot small data (per-channel coefs, fp64
// workspace) and measured slightly ahead of cached loads at BN shapes.
// Per-channel coefficient loads stay CACHED (reused by every block).
template <int BYTES>
struct NTVec;
template <>
struct NTVec<32> {  // 2x dwordx4 (e.g. fp32 V=8)
  using type = int __attribute__((vector_size(32)));
};
template <>
struct NTVec<16> {
  using type = int __attribute__((vector_size(16)));
};
template <>
struct NTVec<8> {
  using type = int __attribute__((vector_size(8)));
};
template <>
struct NTVec<4> {
  using type = int;
};
template <>
struct NTVec<2> {
  using type = short;
};

template <typename T, int V>
__device__ __forceinline__ Pack<T, V> nt_load(const T* p) {
#ifdef MSBN_DISABLE_NT  // cached-access A/B build (tools/kernel_bench.py)
  return *reinterpret_cast<const Pack<T, V>*>(p);
#else
  using VT = typename NTVec<(int)sizeof(T) * V>::type;
  union {
    VT v;
    Pack<T, V> pk;
  } u;
  u.v = __builtin_nontemporal_load(reinterpret_cast<const VT*>(p));
  return u.pk;
#endif
}

template <typename T, int V>
__device__ __forceinline__ void nt_store(T* p, const Pack<T, V>& val) {
#ifdef MSBN_DISABLE_NT
  *reinterpret_cast<Pack<T, V>*>(p) = val;
#else
  using VT = typename NTVec<(int)sizeof(T) * V>::type;
  union {
    VT v;
    Pack<T, V> pk;
  } u;
  u.pk = val;
  __builtin_nontemporal_store(u.v, reinterpret_cast<VT*>(p));
#endif
}

template <typename T>
__device__ __forceinline__ T nt_load1(const T* p) {
  return nt_load<T, 1>(p).v[0];
}

template <typename T>
__device__ __forceinline__ void nt_store1(T* p, T val) {
  Pack<T, 1> pk;
  pk.v[0] = val;
  nt_store<T, 1>(p, pk);
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

// -------------------------------------------- chunked fp32->fp64 accumulator
// The streaming reduction loops are VALU-bound on MI355X when they
// accumulate every element in fp64 (f64 vector ops run at half rate: 16
// half-rate f64 ops per 16 B load caps the stats kernels at ~4.8 TB/s of
// the 6.3 TB/s ceiling).  Accumulate a short fp32 chunk (<= kAccFlush loop
// iterations) and FLUSH into the fp64 totals periodically: full-rate inner
// loop, and the fp64 totals still kill E[x^2]-E[x]^2 cancellation.  fp32
// chunk error is ~sqrt(n)*2^-24 over <=512 elements — far below the fp32
// reference tolerance the tests compare against.
#define MSBN_ACC_FLUSH 64

struct AccPair {
  double a = 0.0, b = 0.0;  // fp64 totals
  float fa = 0.f, fb = 0.f;  // fp32 chunk

  __device__ __forceinline__ void add(float v) {
    fa += v;
    fb += v * v;
  }
  __device__ __forceinline__ void add2(float s, float sq) {
    fa += s;
    fb += sq;
  }
  __device__ __forceinline__ void flush() {
    a += (double)fa;
    b += (double)fb;
    fa = fb = 0.f;
  }
};

}  // namespace msbn
