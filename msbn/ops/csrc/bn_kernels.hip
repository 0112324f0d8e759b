// msbn BatchNorm kernels for MI355X (gfx950, CDNA4).
//
// Replaces the stock SyncBatchNorm kernel family K1-K10 with
// an MI355X-first design:
//
//  * Reductions are TWO-STAGE: a `partial` kernel with a grid sized for the
//    256-CU chip (the stock one-block-per-channel shape leaves 3/4 of the chip
//    idle at C=64) writes per-chunk fp64 {sum, sumsq} pairs to a workspace; a
//    tiny `finalize` kernel combines them.  Deterministic (no atomics), and
//    fp64 accumulation kills the E[x^2]-E[x]^2 cancellation risk outright.
//  * wave64 shuffle reductions + LDS partials inside each block.
//  * 16 B/lane vectorized loads (Pack<T,V>) for bf16/fp16/fp32, NCHW and
//    channels-last (NHWC) layouts both first-class.
//  * The sync path's cross-rank message is built IN the finalize kernel
//    ([mean | invstd | count] packed fp32), and the combine kernel masks
//    zero-count ranks in-device: no cat(), no GPU->CPU sync, hipGraph-safe.
//  * Elementwise kernels apply per-channel affine coefficients precomputed by
//    tiny kernels: y = x*scale+shift, dx = a*dy + b*x + d  (grid-stride,
//    capped grid, fused multiply-add form).
#include "bn_ops.hpp"
#include "common.hpp"

#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include <algorithm>

namespace msbn {

namespace {

// =====================================================================
// stats partial: NCHW   grid(C, nchunkN, nchunkS) block 256
// ws layout: [nchunk][C][2] fp64, chunk = by*gridDim.z + bz
// =====================================================================
template <typename T, int V>
__global__ void bn_stats_partial_nchw(const T* __restrict__ x,
                                      double* __restrict__ ws, int64_t N,
                                      int64_t C, int64_t S, int64_t chunkN,
                                      int64_t chunkS) {
  const int64_t c = blockIdx.x;
  const int64_t n0 = blockIdx.y * chunkN;
  const int64_t n1 = i64min(n0 + chunkN, N);
  const int64_t s0 = blockIdx.z * chunkS;
  const int64_t s1 = i64min(s0 + chunkS, S);
  AccPair acc;
  int since_flush = 0;
  for (int64_t n = n0; n < n1; ++n) {
    const T* row = x + (n * C + c) * S;
    if (V == 1) {
      for (int64_t s = s0 + threadIdx.x; s < s1; s += blockDim.x) {
        acc.add(to_f(nt_load1(&row[s])));
        if (++since_flush == MSBN_ACC_FLUSH) {
          acc.flush();
          since_flush = 0;
        }
      }
    } else {
      for (int64_t s = s0 + (int64_t)threadIdx.x * V; s < s1;
           s += (int64_t)blockDim.x * V) {
        Pack<T, V> pk = nt_load<T, V>(&row[s]);
#pragma unroll
        for (int k = 0; k < V; ++k) acc.add(to_f(pk.v[k]));
        if (++since_flush == MSBN_ACC_FLUSH / V) {
          acc.flush();
          since_flush = 0;
        }
      }
    }
  }
  acc.flush();
  double a = acc.a, b = acc.b;
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, b, lds);
  if (threadIdx.x == 0) {
    const int64_t chunk = (int64_t)blockIdx.y * gridDim.z + blockIdx.z;
    double* out = ws + (c * gridDim.y * gridDim.z + chunk) * 2;
    out[0] = a;
    out[1] = b;
  }
}

// V==1 fallback for layouts where the spatial extent is small or odd
// (e.g. S=49 at C=2048): flat index over the whole (n, s) chunk keeps every
// lane busy (the row-loop form idles blockDim-S threads per row).
template <typename T>
__global__ void bn_stats_partial_nchw_flat(const T* __restrict__ x,
                                           double* __restrict__ ws, int64_t N,
                                           int64_t C, int64_t S,
                                           int64_t chunk_len) {
  const int64_t c = blockIdx.x;
  const int64_t NS = N * S;
  const int64_t p0 = (int64_t)blockIdx.y * chunk_len;
  const int64_t p1 = i64min(p0 + chunk_len, NS);
  AccPair acc;
  int since_flush = 0;
  for (int64_t p = p0 + threadIdx.x; p < p1; p += blockDim.x) {
    const int64_t n = p / S, s = p - n * S;
    acc.add(to_f(nt_load1(&x[(n * C + c) * S + s])));
    if (++since_flush == MSBN_ACC_FLUSH) {
      acc.flush();
      since_flush = 0;
    }
  }
  acc.flush();
  double a = acc.a, b = acc.b;
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, b, lds);
  if (threadIdx.x == 0) {
    double* out = ws + (c * gridDim.y + blockIdx.y) * 2;
    out[0] = a;
    out[1] = b;
  }
}

// =====================================================================
// stats partial: NHWC   grid(ctiles, nchunk) block 256
// Each block covers channels [tile*lpr*V, ...) and a chunk of rows; thread
// (lane, rowoff) owns V consecutive channels.  LDS tree-reduce over rowoff.
// =====================================================================
template <typename T, int V>
__global__ void bn_stats_partial_nhwc(const T* __restrict__ x,
                                      double* __restrict__ ws, int64_t rows,
                                      int64_t C, int64_t chunk_rows, int lpr) {
  const int rpi = blockDim.x / lpr;  // rows in flight per iteration
  const int lane = threadIdx.x % lpr;
  const int rowoff = threadIdx.x / lpr;
  const bool active = rowoff < rpi;
  const int64_t c = (int64_t)blockIdx.x * lpr * V + (int64_t)lane * V;
  const bool inb = active && (c + V <= C);

  AccPair acc[V];

  if (inb) {
    const int64_t r0 = (int64_t)blockIdx.y * chunk_rows;
    const int64_t r1 = i64min(r0 + chunk_rows, rows);
    int since_flush = 0;
    for (int64_t r = r0 + rowoff; r < r1; r += rpi) {
      Pack<T, V> pk = nt_load<T, V>(&x[r * C + c]);
#pragma unroll
      for (int k = 0; k < V; ++k) acc[k].add(to_f(pk.v[k]));
      if (++since_flush == MSBN_ACC_FLUSH) {
#pragma unroll
        for (int k = 0; k < V; ++k) acc[k].flush();
        since_flush = 0;
      }
    }
  }
  double a[V], b[V];
#pragma unroll
  for (int k = 0; k < V; ++k) {
    acc[k].flush();
    a[k] = acc[k].a;
    b[k] = acc[k].b;
  }

  // LDS tree-reduce across the rowoff dimension (generic, non-pow2 safe).
  // LDS layout [k][tid] with separate sum/sumsq planes: lane l of a wave
  // lands on bank (2*l) mod 64 for ds_*_b64 -> conflict-free (the [tid][k]
  // layout measured ~20 extra LDS cycles/instr via SQ_LDS_BANK_CONFLICT).
  __shared__ double sdata[MSBN_BLOCK * V * 2];
#pragma unroll
  for (int k = 0; k < V; ++k) {
    sdata[k * MSBN_BLOCK + threadIdx.x] = a[k];
    sdata[(V + k) * MSBN_BLOCK + threadIdx.x] = b[k];
  }
  for (int st = 1; st < rpi; st <<= 1) {
    __syncthreads();
    if (active && (rowoff & ((st << 1) - 1)) == 0 && rowoff + st < rpi) {
      const int other = threadIdx.x + st * lpr;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        sdata[k * MSBN_BLOCK + threadIdx.x] += sdata[k * MSBN_BLOCK + other];
        sdata[(V + k) * MSBN_BLOCK + threadIdx.x] +=
            sdata[(V + k) * MSBN_BLOCK + other];
      }
    }
  }
  __syncthreads();
  if (rowoff == 0 && c < C) {
    const int64_t chunk = blockIdx.y;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      if (c + k < C) {
        double* out = ws + ((c + k) * gridDim.y + chunk) * 2;
        out[0] = sdata[k * MSBN_BLOCK + threadIdx.x];
        out[1] = sdata[(V + k) * MSBN_BLOCK + threadIdx.x];
      }
    }
  }
}

// =====================================================================
// stats finalize: combine chunk partials -> mean/invstd (+count, +running).
// ONE WAVE per channel (ws is channel-major [C][nchunks][2] -> coalesced
// lane-strided loads + wave64 shuffle reduce); nchunks may be in the
// thousands for small-C layers without serializing.
// =====================================================================
template <typename RT, typename WT>
__global__ void bn_stats_finalize(const double* __restrict__ ws, int nchunks,
                                  int64_t C, double count, float eps,
                                  float* __restrict__ mean,
                                  float* __restrict__ invstd,
                                  float* __restrict__ count_out,
                                  RT* __restrict__ rmean, RT* __restrict__ rvar,
                                  float momentum, const WT* __restrict__ w,
                                  const WT* __restrict__ bws,
                                  float* __restrict__ scale_out,
                                  float* __restrict__ shift_out) {
  const int lane = threadIdx.x & (MSBN_WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int64_t c = (int64_t)blockIdx.x * (blockDim.x >> 6) + wid;
  if (c >= C) return;
  double a = 0.0, b = 0.0;
  for (int ch = lane; ch < nchunks; ch += MSBN_WAVE) {
    a += ws[(c * nchunks + ch) * 2];
    b += ws[(c * nchunks + ch) * 2 + 1];
  }
  wave_reduce_pair(a, b);
  if (lane != 0) return;
  const double m = a / count;
  double var = b / count - m * m;
  var = var > 0.0 ? var : 0.0;
  const float istd = (float)rsqrt(var + (double)eps);
  mean[c] = (float)m;
  invstd[c] = istd;
  if (c == 0 && count_out != nullptr) count_out[0] = (float)count;
  if (rmean != nullptr) {
    const double unbiased = count > 1.0 ? var * (count / (count - 1.0)) : var;
    rmean[c] = from_f<RT>((1.f - momentum) * to_f(rmean[c]) + momentum * (float)m);
    rvar[c] =
        from_f<RT>((1.f - momentum) * to_f(rvar[c]) + momentum * (float)unbiased);
  }
  if (scale_out != nullptr) {
    const float sc = istd * (w != nullptr ? to_f(w[c]) : 1.f);
    scale_out[c] = sc;
    shift_out[c] = -(float)m * sc + (bws != nullptr ? to_f(bws[c]) : 0.f);
  }
}

// =====================================================================
// gather: combine W ranks' packed [mean | invstd | count] rows.
// Zero-count ranks masked HERE (device-side; no host sync).
// =====================================================================
template <typename RT, typename WT>
__global__ void bn_gather_stats(const float* __restrict__ packed_all, int W,
                                int64_t C, float eps, float momentum,
                                float* __restrict__ mean,
                                float* __restrict__ invstd,
                                float* __restrict__ count_out,
                                RT* __restrict__ rmean, RT* __restrict__ rvar,
                                const WT* __restrict__ wpar,
                                const WT* __restrict__ bpar,
                                float* __restrict__ scale_out,
                                float* __restrict__ shift_out) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const int64_t row = 2 * C + 1;
  double n_tot = 0.0, m_acc = 0.0, ex2 = 0.0;
  for (int w = 0; w < W; ++w) {
    const float cnt = packed_all[w * row + 2 * C];
    if (cnt > 0.f) {
      const double m = packed_all[w * row + c];
      const double istd = packed_all[w * row + C + c];
      const double var = 1.0 / (istd * istd) - (double)eps;
      n_tot += cnt;
      m_acc += cnt * m;
      ex2 += cnt * (var + m * m);
    }
  }
  float m_out = 0.f, istd_out = 0.f;
  if (n_tot > 0.0) {
    const double m_g = m_acc / n_tot;
    double var_g = ex2 / n_tot - m_g * m_g;
    var_g = var_g > 0.0 ? var_g : 0.0;
    m_out = (float)m_g;
    istd_out = (float)rsqrt(var_g + (double)eps);
    if (rmean != nullptr) {
      const double unbiased =
          n_tot > 1.0 ? var_g * (n_tot / (n_tot - 1.0)) : var_g;
      rmean[c] =
          from_f<RT>((1.f - momentum) * to_f(rmean[c]) + momentum * (float)m_g);
      rvar[c] = from_f<RT>((1.f - momentum) * to_f(rvar[c]) +
                           momentum * (float)unbiased);
    }
  }
  mean[c] = m_out;
  invstd[c] = istd_out;
  if (c == 0 && count_out != nullptr) count_out[0] = (float)n_tot;
  if (scale_out != nullptr) {
    const float sc = istd_out * (wpar != nullptr ? to_f(wpar[c]) : 1.f);
    scale_out[c] = sc;
    shift_out[c] = -m_out * sc + (bpar != nullptr ? to_f(bpar[c]) : 0.f);
  }
}

// =====================================================================
// per-channel affine coefficient kernels (tiny)
// =====================================================================
template <typename WT>
__global__ void bn_affine_fwd(const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const WT* __restrict__ w,
                              const WT* __restrict__ b, int64_t C,
                              float* __restrict__ scale,
                              float* __restrict__ shift) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float sc = invstd[c] * (w != nullptr ? to_f(w[c]) : 1.f);
  scale[c] = sc;
  shift[c] = -mean[c] * sc + (b != nullptr ? to_f(b[c]) : 0.f);
}

// dx = a*dy + b*x + d with
//   f1 = invstd*gamma, f2 = sum_dy/n, f3 = invstd^2*sum_dy_xmu/n
//   a = f1, b = -f1*f3, d = f1*(f3*mean - f2)
template <typename WT>
__global__ void bn_affine_bwd(const float* __restrict__ mean,
                              const float* __restrict__ invstd,
                              const WT* __restrict__ w,
                              const float* __restrict__ sum_dy,
                              const float* __restrict__ sum_dy_xmu,
                              const float* __restrict__ count, int64_t C,
                              float* __restrict__ coef_a,
                              float* __restrict__ coef_b,
                              float* __restrict__ coef_d) {
  const int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float n = count[0];
  if (n <= 0.f) {
    coef_a[c] = coef_b[c] = coef_d[c] = 0.f;
    return;
  }
  const float istd = invstd[c];
  const float f1 = istd * (w != nullptr ? to_f(w[c]) : 1.f);
  const float f2 = sum_dy[c] / n;
  const float f3 = istd * istd * sum_dy_xmu[c] / n;
  coef_a[c] = f1;
  coef_b[c] = -f1 * f3;
  coef_d[c] = f1 * (f3 * mean[c] - f2);
}

// =====================================================================
// elementwise: y = act(x*scale[c] + shift[c] [+ res])
// RELU / RES are compile-time: the fused BN(+add)+ReLU epilogue replaces the
// eager clamp / add kernels and their full-tensor round trips (guide G13:
// fuse into the producing kernel).
// =====================================================================
template <typename T, int V, bool RELU, bool RES>
__global__ void bn_elemt_nchw(const T* __restrict__ x,
                              const T* __restrict__ res, T* __restrict__ y,
                              const float* __restrict__ scale,
                              const float* __restrict__ shift, int64_t total,
                              int64_t C, int64_t S) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i * V < total; i += stride) {
    const int64_t e = i * V;
    const int64_t c = (e / S) % C;
    const float sc = scale[c], sh = shift[c];
    if (V == 1) {
      float z = to_f(nt_load1(&x[e])) * sc + sh;
      if (RES) z += to_f(nt_load1(&res[e]));
      if (RELU) z = fmaxf(z, 0.f);
      nt_store1(&y[e], from_f<T>(z));
    } else {
      Pack<T, V> px = nt_load<T, V>(&x[e]);
      Pack<T, V> pr;
      if (RES) pr = nt_load<T, V>(&res[e]);
      Pack<T, V> py;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float z = to_f(px.v[k]) * sc + sh;
        if (RES) z += to_f(pr.v[k]);
        if (RELU) z = fmaxf(z, 0.f);
        py.v[k] = from_f<T>(z);
      }
      nt_store<T, V>(&y[e], py);
    }
  }
}

template <typename T, int V, bool RELU, bool RES>
__global__ void bn_elemt_nhwc(const T* __restrict__ x,
                              const T* __restrict__ res, T* __restrict__ y,
                              const float* __restrict__ scale,
                              const float* __restrict__ shift, int64_t total,
                              int64_t C) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i * V < total; i += stride) {
    const int64_t e = i * V;
    const int64_t c = e % C;
    if (V == 1) {
      float z = to_f(nt_load1(&x[e])) * scale[c] + shift[c];
      if (RES) z += to_f(nt_load1(&res[e]));
      if (RELU) z = fmaxf(z, 0.f);
      nt_store1(&y[e], from_f<T>(z));
    } else {
      Pack<T, V> px = nt_load<T, V>(&x[e]);
      Pack<float, V> ps = *reinterpret_cast<const Pack<float, V>*>(&scale[c]);
      Pack<float, V> pb = *reinterpret_cast<const Pack<float, V>*>(&shift[c]);
      Pack<T, V> pr;
      if (RES) pr = nt_load<T, V>(&res[e]);
      Pack<T, V> py;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float z = to_f(px.v[k]) * ps.v[k] + pb.v[k];
        if (RES) z += to_f(pr.v[k]);
        if (RELU) z = fmaxf(z, 0.f);
        py.v[k] = from_f<T>(z);
      }
      nt_store<T, V>(&y[e], py);
    }
  }
}

// =====================================================================
// backward reduce partial: {sum_g, sum_g*(x-mean)} per channel, where
// g = dy masked by the fused-ReLU gate when MASK: the pre-activation
// z = scale[c]*x + shift[c] (+ res) is recomputed in-kernel, so the fused
// forward never has to store a mask or the pre-activation tensor.
// =====================================================================
template <typename T, int V, bool MASK, bool RES, bool GMOUT>
__global__ void bn_bwd_reduce_partial_nchw(const T* __restrict__ dy,
                                           const T* __restrict__ x,
                                           const T* __restrict__ res,
                                           T* __restrict__ gm_out,
                                           const float* __restrict__ mean,
                                           const float* __restrict__ scale,
                                           const float* __restrict__ shift,
                                           double* __restrict__ ws, int64_t N,
                                           int64_t C, int64_t S, int64_t chunkN,
                                           int64_t chunkS) {
  const int64_t c = blockIdx.x;
  const float m = mean[c];
  const float sc = MASK ? scale[c] : 0.f;
  const float sh = MASK ? shift[c] : 0.f;
  const int64_t n0 = blockIdx.y * chunkN;
  const int64_t n1 = i64min(n0 + chunkN, N);
  const int64_t s0 = blockIdx.z * chunkS;
  const int64_t s1 = i64min(s0 + chunkS, S);
  AccPair acc;
  int since_flush = 0;
  for (int64_t n = n0; n < n1; ++n) {
    const int64_t base = (n * C + c) * S;
    if (V == 1) {
      for (int64_t s = s0 + threadIdx.x; s < s1; s += blockDim.x) {
        float g = to_f(nt_load1(&dy[base + s]));
        const float xv = to_f(nt_load1(&x[base + s]));
        if (MASK) {
          float z = sc * xv + sh;
          if (RES) z += to_f(nt_load1(&res[base + s]));
          if (z <= 0.f) g = 0.f;
        }
        if (GMOUT) nt_store1(&gm_out[base + s], from_f<T>(g));
        acc.add2(g, g * (xv - m));
        if (++since_flush == MSBN_ACC_FLUSH) {
          acc.flush();
          since_flush = 0;
        }
      }
    } else {
      for (int64_t s = s0 + (int64_t)threadIdx.x * V; s < s1;
           s += (int64_t)blockDim.x * V) {
        Pack<T, V> pg = nt_load<T, V>(&dy[base + s]);
        Pack<T, V> px = nt_load<T, V>(&x[base + s]);
        Pack<T, V> pr, pm;
        if (RES) pr = nt_load<T, V>(&res[base + s]);
#pragma unroll
        for (int k = 0; k < V; ++k) {
          float g = to_f(pg.v[k]);
          const float xv = to_f(px.v[k]);
          if (MASK) {
            float z = sc * xv + sh;
            if (RES) z += to_f(pr.v[k]);
            if (z <= 0.f) g = 0.f;
          }
          if (GMOUT) pm.v[k] = from_f<T>(g);
          acc.add2(g, g * (xv - m));
        }
        if (GMOUT) nt_store<T, V>(&gm_out[base + s], pm);
        if (++since_flush == MSBN_ACC_FLUSH / V) {
          acc.flush();
          since_flush = 0;
        }
      }
    }
  }
  acc.flush();
  double a = acc.a, b = acc.b;
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, b, lds);
  if (threadIdx.x == 0) {
    const int64_t chunk = (int64_t)blockIdx.y * gridDim.z + blockIdx.z;
    double* out = ws + (c * gridDim.y * gridDim.z + chunk) * 2;
    out[0] = a;
    out[1] = b;
  }
}

template <typename T, bool MASK, bool RES, bool GMOUT>
__global__ void bn_bwd_reduce_partial_nchw_flat(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ res, T* __restrict__ gm_out,
    const float* __restrict__ mean,
    const float* __restrict__ scale, const float* __restrict__ shift,
    double* __restrict__ ws, int64_t N, int64_t C, int64_t S,
    int64_t chunk_len) {
  const int64_t c = blockIdx.x;
  const float m = mean[c];
  const float sc = MASK ? scale[c] : 0.f;
  const float sh = MASK ? shift[c] : 0.f;
  const int64_t NS = N * S;
  const int64_t p0 = (int64_t)blockIdx.y * chunk_len;
  const int64_t p1 = i64min(p0 + chunk_len, NS);
  AccPair acc;
  int since_flush = 0;
  for (int64_t p = p0 + threadIdx.x; p < p1; p += blockDim.x) {
    const int64_t n = p / S, s = p - n * S;
    const int64_t e = (n * C + c) * S + s;
    float g = to_f(nt_load1(&dy[e]));
    const float xv = to_f(nt_load1(&x[e]));
    if (MASK) {
      float z = sc * xv + sh;
      if (RES) z += to_f(nt_load1(&res[e]));
      if (z <= 0.f) g = 0.f;
    }
    if (GMOUT) nt_store1(&gm_out[e], from_f<T>(g));
    acc.add2(g, g * (xv - m));
    if (++since_flush == MSBN_ACC_FLUSH) {
      acc.flush();
      since_flush = 0;
    }
  }
  acc.flush();
  double a = acc.a, b = acc.b;
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, b, lds);
  if (threadIdx.x == 0) {
    double* out = ws + (c * gridDim.y + blockIdx.y) * 2;
    out[0] = a;
    out[1] = b;
  }
}

template <typename T, int V, bool MASK, bool RES, bool GMOUT>
__global__ void bn_bwd_reduce_partial_nhwc(const T* __restrict__ dy,
                                           const T* __restrict__ x,
                                           const T* __restrict__ res,
                                           T* __restrict__ gm_out,
                                           const float* __restrict__ mean,
                                           const float* __restrict__ scale,
                                           const float* __restrict__ shift,
                                           double* __restrict__ ws,
                                           int64_t rows, int64_t C,
                                           int64_t chunk_rows, int lpr) {
  const int rpi = blockDim.x / lpr;
  const int lane = threadIdx.x % lpr;
  const int rowoff = threadIdx.x / lpr;
  const bool active = rowoff < rpi;
  const int64_t c = (int64_t)blockIdx.x * lpr * V + (int64_t)lane * V;
  const bool inb = active && (c + V <= C);

  AccPair acc[V];
  float m[V], scv[V], shv[V];
#pragma unroll
  for (int k = 0; k < V; ++k) m[k] = scv[k] = shv[k] = 0.f;
  if (inb) {
#pragma unroll
    for (int k = 0; k < V; ++k) {
      m[k] = mean[c + k];
      if (MASK) {
        scv[k] = scale[c + k];
        shv[k] = shift[c + k];
      }
    }
    const int64_t r0 = (int64_t)blockIdx.y * chunk_rows;
    const int64_t r1 = i64min(r0 + chunk_rows, rows);
    int since_flush = 0;
    for (int64_t r = r0 + rowoff; r < r1; r += rpi) {
      Pack<T, V> pg = nt_load<T, V>(&dy[r * C + c]);
      Pack<T, V> px = nt_load<T, V>(&x[r * C + c]);
      Pack<T, V> pr, pm;
      if (RES) pr = nt_load<T, V>(&res[r * C + c]);
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float g = to_f(pg.v[k]);
        const float xv = to_f(px.v[k]);
        if (MASK) {
          float z = scv[k] * xv + shv[k];
          if (RES) z += to_f(pr.v[k]);
          if (z <= 0.f) g = 0.f;
        }
        if (GMOUT) pm.v[k] = from_f<T>(g);
        acc[k].add2(g, g * (xv - m[k]));
      }
      if (GMOUT) nt_store<T, V>(&gm_out[r * C + c], pm);
      if (++since_flush == MSBN_ACC_FLUSH) {
#pragma unroll
        for (int k = 0; k < V; ++k) acc[k].flush();
        since_flush = 0;
      }
    }
  }
  double a[V], b[V];
#pragma unroll
  for (int k = 0; k < V; ++k) {
    acc[k].flush();
    a[k] = acc[k].a;
    b[k] = acc[k].b;
  }
  // LDS layout [k][tid] with separate sum/sumsq planes: lane l of a wave
  // lands on bank (2*l) mod 64 for ds_*_b64 -> conflict-free (the [tid][k]
  // layout measured ~20 extra LDS cycles/instr via SQ_LDS_BANK_CONFLICT).
  __shared__ double sdata[MSBN_BLOCK * V * 2];
#pragma unroll
  for (int k = 0; k < V; ++k) {
    sdata[k * MSBN_BLOCK + threadIdx.x] = a[k];
    sdata[(V + k) * MSBN_BLOCK + threadIdx.x] = b[k];
  }
  for (int st = 1; st < rpi; st <<= 1) {
    __syncthreads();
    if (active && (rowoff & ((st << 1) - 1)) == 0 && rowoff + st < rpi) {
      const int other = threadIdx.x + st * lpr;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        sdata[k * MSBN_BLOCK + threadIdx.x] += sdata[k * MSBN_BLOCK + other];
        sdata[(V + k) * MSBN_BLOCK + threadIdx.x] +=
            sdata[(V + k) * MSBN_BLOCK + other];
      }
    }
  }
  __syncthreads();
  if (rowoff == 0 && c < C) {
    const int64_t chunk = blockIdx.y;
#pragma unroll
    for (int k = 0; k < V; ++k) {
      if (c + k < C) {
        double* out = ws + ((c + k) * gridDim.y + chunk) * 2;
        out[0] = sdata[k * MSBN_BLOCK + threadIdx.x];
        out[1] = sdata[(V + k) * MSBN_BLOCK + threadIdx.x];
      }
    }
  }
}

// one wave per channel (see bn_stats_finalize)
template <typename WT>
__global__ void bn_bwd_reduce_finalize(const double* __restrict__ ws,
                                       int nchunks, int64_t C,
                                       const float* __restrict__ invstd,
                                       float* __restrict__ sum_dy,
                                       float* __restrict__ sum_dy_xmu,
                                       WT* __restrict__ grad_weight,
                                       WT* __restrict__ grad_bias) {
  const int lane = threadIdx.x & (MSBN_WAVE - 1);
  const int wid = threadIdx.x >> 6;
  const int64_t c = (int64_t)blockIdx.x * (blockDim.x >> 6) + wid;
  if (c >= C) return;
  double a = 0.0, b = 0.0;
  for (int ch = lane; ch < nchunks; ch += MSBN_WAVE) {
    a += ws[(c * nchunks + ch) * 2];
    b += ws[(c * nchunks + ch) * 2 + 1];
  }
  wave_reduce_pair(a, b);
  if (lane != 0) return;
  if (sum_dy != nullptr) sum_dy[c] = (float)a;
  if (sum_dy_xmu != nullptr) sum_dy_xmu[c] = (float)b;
  if (grad_weight != nullptr) grad_weight[c] = from_f<WT>((float)(b * invstd[c]));
  if (grad_bias != nullptr) grad_bias[c] = from_f<WT>((float)a);
}

// =====================================================================
// backward elementwise: dx = a[c]*g + b[c]*x + d[c], with the fused-ReLU
// gate recomputed in-kernel when MASK (g = z>0 ? dy : 0); RESG additionally
// writes the residual-branch gradient dres = g (the add node's pass-through).
// =====================================================================
template <typename T, int V, bool MASK, bool RES, bool RESG>
__global__ void bn_bwd_elemt_nchw(const T* __restrict__ dy,
                                  const T* __restrict__ x,
                                  const T* __restrict__ res,
                                  T* __restrict__ dx, T* __restrict__ dres,
                                  const float* __restrict__ ca,
                                  const float* __restrict__ cb,
                                  const float* __restrict__ cd,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  int64_t total, int64_t C, int64_t S) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i * V < total; i += stride) {
    const int64_t e = i * V;
    const int64_t c = (e / S) % C;
    const float A = ca[c], B = cb[c], D = cd[c];
    const float sc = MASK ? scale[c] : 0.f;
    const float sh = MASK ? shift[c] : 0.f;
    if (V == 1) {
      float g = to_f(nt_load1(&dy[e]));
      const float xv = to_f(nt_load1(&x[e]));
      if (MASK) {
        float z = sc * xv + sh;
        if (RES) z += to_f(nt_load1(&res[e]));
        if (z <= 0.f) g = 0.f;
      }
      nt_store1(&dx[e], from_f<T>(A * g + B * xv + D));
      if (RESG) nt_store1(&dres[e], from_f<T>(g));
    } else {
      Pack<T, V> pg = nt_load<T, V>(&dy[e]);
      Pack<T, V> px = nt_load<T, V>(&x[e]);
      Pack<T, V> pr;
      if (RES) pr = nt_load<T, V>(&res[e]);
      Pack<T, V> po, pq;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float g = to_f(pg.v[k]);
        const float xv = to_f(px.v[k]);
        if (MASK) {
          float z = sc * xv + sh;
          if (RES) z += to_f(pr.v[k]);
          if (z <= 0.f) g = 0.f;
        }
        po.v[k] = from_f<T>(A * g + B * xv + D);
        if (RESG) pq.v[k] = from_f<T>(g);
      }
      nt_store<T, V>(&dx[e], po);
      if (RESG) nt_store<T, V>(&dres[e], pq);
    }
  }
}

template <typename T, int V, bool MASK, bool RES, bool RESG>
__global__ void bn_bwd_elemt_nhwc(const T* __restrict__ dy,
                                  const T* __restrict__ x,
                                  const T* __restrict__ res,
                                  T* __restrict__ dx, T* __restrict__ dres,
                                  const float* __restrict__ ca,
                                  const float* __restrict__ cb,
                                  const float* __restrict__ cd,
                                  const float* __restrict__ scale,
                                  const float* __restrict__ shift,
                                  int64_t total, int64_t C) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i * V < total; i += stride) {
    const int64_t e = i * V;
    const int64_t c = e % C;
    if (V == 1) {
      float g = to_f(nt_load1(&dy[e]));
      const float xv = to_f(nt_load1(&x[e]));
      if (MASK) {
        float z = scale[c] * xv + shift[c];
        if (RES) z += to_f(nt_load1(&res[e]));
        if (z <= 0.f) g = 0.f;
      }
      nt_store1(&dx[e], from_f<T>(ca[c] * g + cb[c] * xv + cd[c]));
      if (RESG) nt_store1(&dres[e], from_f<T>(g));
    } else {
      Pack<T, V> pg = nt_load<T, V>(&dy[e]);
      Pack<T, V> px = nt_load<T, V>(&x[e]);
      Pack<float, V> pa = *reinterpret_cast<const Pack<float, V>*>(&ca[c]);
      Pack<float, V> pb = *reinterpret_cast<const Pack<float, V>*>(&cb[c]);
      Pack<float, V> pd = *reinterpret_cast<const Pack<float, V>*>(&cd[c]);
      Pack<float, V> psc, psh;
      if (MASK) {
        psc = *reinterpret_cast<const Pack<float, V>*>(&scale[c]);
        psh = *reinterpret_cast<const Pack<float, V>*>(&shift[c]);
      }
      Pack<T, V> pr;
      if (RES) pr = nt_load<T, V>(&res[e]);
      Pack<T, V> po, pq;
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float g = to_f(pg.v[k]);
        const float xv = to_f(px.v[k]);
        if (MASK) {
          float z = psc.v[k] * xv + psh.v[k];
          if (RES) z += to_f(pr.v[k]);
          if (z <= 0.f) g = 0.f;
        }
        po.v[k] = from_f<T>(pa.v[k] * g + pb.v[k] * xv + pd.v[k]);
        if (RESG) pq.v[k] = from_f<T>(g);
      }
      nt_store<T, V>(&dx[e], po);
      if (RESG) nt_store<T, V>(&dres[e], pq);
    }
  }
}

// =====================================================================
// small-plane world-1 fused kernels (the stock K10-style single-launch
// family, MI355X-gated): block-per-channel, TWO passes over an L2-resident
// plane.  Used when plane = N*S is small (GAN / small-per-GPU-batch
// regime): there the per-launch ramp of the 3-kernel
// two-stage pipeline dominates and the plane fits cache, so the second
// pass re-reads from L2 (plain cached accesses on purpose — no nt).
// NCHW only; fp32 running stats / fp32 affine (host gates eligibility).
// =====================================================================
// V-wide flat indexing: pack pp covers elements [pp*V, pp*V+V) of the
// channel plane; S % V == 0 (host-gated) keeps every pack inside one row.
// int32 arithmetic throughout (plane <= 32K elems by gate).
template <typename T, int V, bool RELU, bool RES>
__global__ void bn_fwd_fused_small_nchw(
    const T* __restrict__ x, const T* __restrict__ res, T* __restrict__ y,
    int N, int64_t C, int S, float eps, float momentum,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    float* __restrict__ count_out, float* __restrict__ rmean,
    float* __restrict__ rvar, const float* __restrict__ w,
    const float* __restrict__ b, float* __restrict__ scale_out,
    float* __restrict__ shift_out) {
  const int64_t c = blockIdx.x;
  const int plane = N * S;
  const int packs = plane / V;
  double a = 0.0, bb = 0.0;
  for (int pp = threadIdx.x; pp < packs; pp += blockDim.x) {
    const int e = pp * V;
    const int n = e / S, s = e - n * S;
    const T* row = x + ((int64_t)n * C + c) * S + s;
    Pack<T, V> pk = *reinterpret_cast<const Pack<T, V>*>(row);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      const float v = to_f(pk.v[k]);
      a += v;
      bb += (double)v * v;
    }
  }
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, bb, lds);
  __shared__ float sc_sh[2];
  if (threadIdx.x == 0) {
    const double cnt = (double)plane;
    const double m = a / cnt;
    double var = bb / cnt - m * m;
    var = var > 0.0 ? var : 0.0;
    const float istd = (float)rsqrt(var + (double)eps);
    mean_out[c] = (float)m;
    invstd_out[c] = istd;
    if (c == 0 && count_out != nullptr) count_out[0] = (float)cnt;
    if (rmean != nullptr) {
      const double unbiased = cnt > 1.0 ? var * (cnt / (cnt - 1.0)) : var;
      rmean[c] = (1.f - momentum) * rmean[c] + momentum * (float)m;
      rvar[c] = (1.f - momentum) * rvar[c] + momentum * (float)unbiased;
    }
    const float sc = istd * (w != nullptr ? w[c] : 1.f);
    const float sh = -(float)m * sc + (b != nullptr ? b[c] : 0.f);
    scale_out[c] = sc;
    shift_out[c] = sh;
    sc_sh[0] = sc;
    sc_sh[1] = sh;
  }
  __syncthreads();
  const float sc = sc_sh[0], sh = sc_sh[1];
  for (int pp = threadIdx.x; pp < packs; pp += blockDim.x) {
    const int e = pp * V;
    const int n = e / S, s = e - n * S;
    const int64_t base = ((int64_t)n * C + c) * S + s;
    Pack<T, V> px = *reinterpret_cast<const Pack<T, V>*>(&x[base]);
    Pack<T, V> pr, py;
    if (RES) pr = *reinterpret_cast<const Pack<T, V>*>(&res[base]);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float z = to_f(px.v[k]) * sc + sh;
      if (RES) z += to_f(pr.v[k]);
      if (RELU) z = fmaxf(z, 0.f);
      py.v[k] = from_f<T>(z);
    }
    *reinterpret_cast<Pack<T, V>*>(&y[base]) = py;
  }
}

template <typename T, int V, bool MASK, bool RES, bool RESG>
__global__ void bn_bwd_fused_small_nchw(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ res, T* __restrict__ dx, T* __restrict__ dres,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    const float* __restrict__ scale, const float* __restrict__ shift,
    const float* __restrict__ w, float* __restrict__ gw,
    float* __restrict__ gb, int N, int64_t C, int S) {
  const int64_t c = blockIdx.x;
  const int plane = N * S;
  const int packs = plane / V;
  const float m = mean[c];
  const float sc = MASK ? scale[c] : 0.f;
  const float sh = MASK ? shift[c] : 0.f;
  double a = 0.0, bb = 0.0;
  for (int pp = threadIdx.x; pp < packs; pp += blockDim.x) {
    const int e = pp * V;
    const int n = e / S, s = e - n * S;
    const int64_t base = ((int64_t)n * C + c) * S + s;
    Pack<T, V> pg = *reinterpret_cast<const Pack<T, V>*>(&dy[base]);
    Pack<T, V> px = *reinterpret_cast<const Pack<T, V>*>(&x[base]);
    Pack<T, V> pr;
    if (MASK && RES) pr = *reinterpret_cast<const Pack<T, V>*>(&res[base]);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float g = to_f(pg.v[k]);
      const float xv = to_f(px.v[k]);
      if (MASK) {
        float z = sc * xv + sh;
        if (RES) z += to_f(pr.v[k]);
        if (z <= 0.f) g = 0.f;
      }
      a += g;
      bb += (double)g * (xv - m);
    }
  }
  __shared__ double lds[2 * (MSBN_BLOCK / MSBN_WAVE)];
  block_reduce_pair(a, bb, lds);
  __shared__ float abd[3];
  if (threadIdx.x == 0) {
    const float istd = invstd[c];
    const float n_inv = 1.f / (float)plane;
    if (gw != nullptr) gw[c] = (float)(bb * istd);
    if (gb != nullptr) gb[c] = (float)a;
    const float f1 = istd * (w != nullptr ? w[c] : 1.f);
    const float f2 = (float)a * n_inv;
    const float f3 = istd * istd * (float)bb * n_inv;
    abd[0] = f1;
    abd[1] = -f1 * f3;
    abd[2] = f1 * (f3 * m - f2);
  }
  __syncthreads();
  const float A = abd[0], B = abd[1], D = abd[2];
  for (int pp = threadIdx.x; pp < packs; pp += blockDim.x) {
    const int e = pp * V;
    const int n = e / S, s = e - n * S;
    const int64_t base = ((int64_t)n * C + c) * S + s;
    Pack<T, V> pg = *reinterpret_cast<const Pack<T, V>*>(&dy[base]);
    Pack<T, V> px = *reinterpret_cast<const Pack<T, V>*>(&x[base]);
    Pack<T, V> pr, po, pq;
    if (MASK && RES) pr = *reinterpret_cast<const Pack<T, V>*>(&res[base]);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      float g = to_f(pg.v[k]);
      const float xv = to_f(px.v[k]);
      if (MASK) {
        float z = sc * xv + sh;
        if (RES) z += to_f(pr.v[k]);
        if (z <= 0.f) g = 0.f;
      }
      po.v[k] = from_f<T>(A * g + B * xv + D);
      if (RESG) pq.v[k] = from_f<T>(g);
    }
    *reinterpret_cast<Pack<T, V>*>(&dx[base]) = po;
    if (RESG) *reinterpret_cast<Pack<T, V>*>(&dres[base]) = pq;
  }
}

// =====================================================================
// host-side helpers
// =====================================================================

struct Layout {
  bool nhwc;       // channels-last (rows x C contiguous)
  int64_t N, C, S; // NCHW view (S = spatial)
  int64_t rows;    // NHWC view (= N*S)
};

Layout get_layout(const at::Tensor& t) {
  Layout L{};
  L.C = t.size(1);
  L.N = t.size(0);
  L.S = t.numel() / std::max<int64_t>(L.N * L.C, 1);
  L.rows = L.N * L.S;
  const auto fmt = t.suggest_memory_format();
  if ((fmt == at::MemoryFormat::ChannelsLast && t.dim() == 4 &&
       t.is_contiguous(at::MemoryFormat::ChannelsLast)) ||
      (fmt == at::MemoryFormat::ChannelsLast3d && t.dim() == 5 &&
       t.is_contiguous(at::MemoryFormat::ChannelsLast3d))) {
    L.nhwc = true;
  } else {
    TORCH_CHECK(t.is_contiguous(), "msbn: input must be contiguous (NCHW) or "
                                   "channels-last");
    L.nhwc = false;
  }
  return L;
}

inline int64_t cdiv(int64_t a, int64_t b) { return (a + b - 1) / b; }
inline int64_t clamp64(int64_t v, int64_t lo, int64_t hi) {
  return v < lo ? lo : (v > hi ? hi : v);
}

// channels per finalize block (one wave each)
constexpr int kFinalizeWavesPerBlock = MSBN_BLOCK / MSBN_WAVE;

struct FlatGrid {
  dim3 grid;
  int64_t chunk_len;
  int nchunks;
};

// grid for the V==1 flat NCHW partial kernels: (C, nchunks)
FlatGrid flat_grid(int64_t C, int64_t NS, int64_t target_blocks) {
  FlatGrid g{};
  int64_t nchunks =
      clamp64(target_blocks / std::max<int64_t>(C, 1), 1, cdiv(NS, 1024));
  g.chunk_len = cdiv(NS, nchunks);
  nchunks = cdiv(NS, g.chunk_len);
  g.grid = dim3((unsigned)C, (unsigned)nchunks);
  g.nchunks = (int)nchunks;
  return g;
}

// pick vector width: 16B/lane when layout & alignment allow
template <typename T>
int pick_v(const void* p0, const void* p1, const void* p2, int64_t inner) {
  const int vmax = 16 / (int)sizeof(T);
  auto ok = [&](int v) {
    const size_t bytes = (size_t)v * sizeof(T);
    auto aligned = [&](const void* p) {
      return p == nullptr || ((uintptr_t)p % bytes) == 0;
    };
    return inner % v == 0 && aligned(p0) && aligned(p1) && aligned(p2);
  };
  if (ok(vmax)) return vmax;
  if (vmax >= 4 && ok(vmax / 2)) return vmax / 2;
  return 1;
}

constexpr int kTargetBlocks = 2048;  // 256 CUs x 8 blocks (guide G11)

struct NchwGrid {
  dim3 grid;
  int64_t chunkN, chunkS;
  int nchunks;
};

NchwGrid nchw_grid(int64_t N, int64_t C, int64_t S, int V) {
  NchwGrid g{};
  const int64_t perC = std::max<int64_t>(1, kTargetBlocks / std::max<int64_t>(C, 1));
  int64_t nchunkN = std::min<int64_t>(N, perC);
  nchunkN = std::max<int64_t>(nchunkN, 1);
  g.chunkN = cdiv(N, nchunkN);
  nchunkN = cdiv(N, g.chunkN);
  int64_t want_s = cdiv(perC, nchunkN);
  // each S-chunk should be a multiple of V and >= ~4096 elements of work
  int64_t max_s = cdiv(S, std::max<int64_t>((int64_t)MSBN_BLOCK * V, 1024));
  int64_t nchunkS = std::max<int64_t>(1, std::min(want_s, std::max<int64_t>(max_s, 1)));
  g.chunkS = cdiv(cdiv(S, nchunkS), V) * V;
  nchunkS = cdiv(S, g.chunkS);
  g.grid = dim3((unsigned)C, (unsigned)nchunkN, (unsigned)nchunkS);
  g.nchunks = (int)(nchunkN * nchunkS);
  return g;
}

struct NhwcGrid {
  dim3 grid;
  int64_t chunk_rows;
  int nchunks;
  int lpr;
};

NhwcGrid nhwc_grid(int64_t rows, int64_t C, int V) {
  NhwcGrid g{};
  const int64_t cv = cdiv(C, V);
  // Workspace traffic is C * nchunks * 16 B, written by the partial kernel
  // and read back by finalize.  The old lpr = min(cv, 256) choice made
  // ctiles = 1 for C >= 2048 (bf16), forcing nchunks ~ kTargetBlocks and a
  // workspace up to HALF the payload (measured 1.4 TB/s effective at
  // 512x2048x7x7 vs 6.3 TB/s ceiling).  Tile CHANNELS instead: pick lpr so
  // ~16 channel tiles exist, which caps nchunks near kTargetBlocks/16 and
  // keeps the workspace at a few % of the payload.  Lower bound lpr at 8
  // lanes so each row segment stays >= 128 B (one full memory granule).
  int64_t lpr = cdiv(cv, 16);
  // round up to a power of two that divides the block
  int64_t p2 = 8;
  while (p2 < lpr) p2 <<= 1;
  lpr = std::min<int64_t>(std::max<int64_t>(p2, 8), MSBN_BLOCK);
  lpr = std::min(lpr, [&] {  // never exceed what C needs
    int64_t q = 8;
    while (q < cv) q <<= 1;
    return std::min<int64_t>(q, MSBN_BLOCK);
  }());
  g.lpr = (int)lpr;
  const int64_t ctiles = cdiv(C, lpr * V);
  const int64_t per_tile =
      std::max<int64_t>(1, kTargetBlocks / std::max<int64_t>(ctiles, 1));
  const int rpi = MSBN_BLOCK / g.lpr;
  int64_t max_chunks = cdiv(rows, std::max<int64_t>(4 * rpi, 16));
  int64_t nchunks = std::max<int64_t>(1, std::min(per_tile, std::max<int64_t>(max_chunks, 1)));
  g.chunk_rows = cdiv(rows, nchunks);
  nchunks = cdiv(rows, g.chunk_rows);
  g.grid = dim3((unsigned)ctiles, (unsigned)nchunks);
  g.nchunks = (int)nchunks;
  return g;
}

int elemt_grid(int64_t total, int V) {
  return (int)std::min<int64_t>(cdiv(total, (int64_t)MSBN_BLOCK * V),
                                2 * kTargetBlocks);
}

#define MSBN_DISPATCH_FLOAT_TYPES(TYPE, NAME, ...)                          \
  [&] {                                                                     \
    switch (TYPE) {                                                         \
      case at::kFloat: {                                                    \
        using scalar_t = float;                                             \
        using native_t = float;                                             \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kBFloat16: {                                                 \
        using scalar_t = at::BFloat16;                                      \
        using native_t = __hip_bfloat16;                                    \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kHalf: {                                                     \
        using scalar_t = at::Half;                                          \
        using native_t = __half;                                            \
        return __VA_ARGS__();                                               \
      }                                                                     \
      default:                                                              \
        TORCH_CHECK(false, NAME, ": unsupported dtype ", TYPE);             \
    }                                                                       \
  }()

#define MSBN_DISPATCH_V(VSEL, VMAX, ...)                                    \
  [&] {                                                                     \
    if constexpr (VMAX == 8) {                                              \
      switch (VSEL) {                                                       \
        case 8: {                                                           \
          constexpr int VV = 8;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
        case 4: {                                                           \
          constexpr int VV = 4;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
        default: {                                                          \
          constexpr int VV = 1;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
      }                                                                     \
    } else {                                                                \
      switch (VSEL) {                                                       \
        case 4: {                                                           \
          constexpr int VV = 4;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
        case 2: {                                                           \
          constexpr int VV = 2;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
        default: {                                                          \
          constexpr int VV = 1;                                             \
          return __VA_ARGS__();                                             \
        }                                                                   \
      }                                                                     \
    }                                                                       \
  }()

#define MSBN_DISPATCH_RSTAT(TYPE, NAME, ...)                                \
  [&] {                                                                     \
    switch (TYPE) {                                                         \
      case at::kFloat: {                                                    \
        using rstat_t = float;                                              \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kBFloat16: {                                                 \
        using rstat_t = __hip_bfloat16;                                     \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kHalf: {                                                     \
        using rstat_t = __half;                                             \
        return __VA_ARGS__();                                               \
      }                                                                     \
      default:                                                              \
        TORCH_CHECK(false, NAME, ": unsupported running-stat dtype ", TYPE); \
    }                                                                       \
  }()

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// Launch the two-stage stats reduction; writes mean/invstd/count into the
// given fp32 pointers (which may alias a packed buffer).
#define MSBN_DISPATCH_WTYPE(TYPE, NAME, ...)                                \
  [&] {                                                                     \
    switch (TYPE) {                                                         \
      case at::kFloat: {                                                    \
        using wt_t = float;                                                 \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kBFloat16: {                                                 \
        using wt_t = __hip_bfloat16;                                        \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::kHalf: {                                                     \
        using wt_t = __half;                                                \
        return __VA_ARGS__();                                               \
      }                                                                     \
      default:                                                              \
        TORCH_CHECK(false, NAME, ": unsupported weight dtype ", TYPE);      \
    }                                                                       \
  }()

void stats_into(const at::Tensor& input, double eps, float* mean_p,
                float* invstd_p, float* count_p, const at::Tensor* rmean,
                const at::Tensor* rvar, double momentum,
                const at::Tensor* wpar = nullptr,
                const at::Tensor* bpar = nullptr, float* scale_p = nullptr,
                float* shift_p = nullptr) {
  const Layout L = get_layout(input);
  const int64_t count = L.rows;
  auto stream = cur_stream();
  const auto rstat_type = rmean ? rmean->scalar_type() : at::kFloat;
  const auto w_type = wpar ? wpar->scalar_type()
                     : bpar ? bpar->scalar_type()
                            : at::kFloat;

  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "batch_norm_stats", [&] {
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    at::Tensor ws;
    int nchunks = 0;
    if (!L.nhwc) {
      const int v = pick_v<native_t>(x, nullptr, nullptr, L.S);
      if (v == 1) {
        auto g = flat_grid(L.C, L.rows, kTargetBlocks);
        nchunks = g.nchunks;
        ws = at::empty({(int64_t)nchunks * L.C * 2},
                       input.options().dtype(at::kDouble));
        hipLaunchKernelGGL((bn_stats_partial_nchw_flat<native_t>), g.grid,
                           dim3(MSBN_BLOCK), 0, stream, x,
                           ws.data_ptr<double>(), L.N, L.C, L.S, g.chunk_len);
      } else {
      auto g = nchw_grid(L.N, L.C, L.S, v);
      nchunks = g.nchunks;
      ws = at::empty({(int64_t)nchunks * L.C * 2},
                     input.options().dtype(at::kDouble));
      MSBN_DISPATCH_V(v, VMAX, [&] {
        hipLaunchKernelGGL((bn_stats_partial_nchw<native_t, VV>), g.grid,
                           dim3(MSBN_BLOCK), 0, stream, x,
                           ws.data_ptr<double>(), L.N, L.C, L.S, g.chunkN,
                           g.chunkS);
      });
      }
    } else {
      const int v = pick_v<native_t>(x, nullptr, nullptr, L.C);
      auto g = nhwc_grid(L.rows, L.C, v);
      nchunks = g.nchunks;
      ws = at::empty({(int64_t)nchunks * L.C * 2},
                     input.options().dtype(at::kDouble));
      MSBN_DISPATCH_V(v, VMAX, [&] {
        hipLaunchKernelGGL((bn_stats_partial_nhwc<native_t, VV>), g.grid,
                           dim3(MSBN_BLOCK), 0, stream, x,
                           ws.data_ptr<double>(), L.rows, L.C, g.chunk_rows,
                           g.lpr);
      });
    }
    const int fgrid = (int)cdiv(L.C, kFinalizeWavesPerBlock);
    MSBN_DISPATCH_RSTAT(rstat_type, "batch_norm_stats", [&] {
      MSBN_DISPATCH_WTYPE(w_type, "batch_norm_stats", [&] {
        rstat_t* rm = rmean ? reinterpret_cast<rstat_t*>(rmean->data_ptr())
                            : nullptr;
        rstat_t* rv = rvar ? reinterpret_cast<rstat_t*>(rvar->data_ptr())
                           : nullptr;
        const wt_t* wp =
            wpar ? reinterpret_cast<const wt_t*>(wpar->data_ptr()) : nullptr;
        const wt_t* bp =
            bpar ? reinterpret_cast<const wt_t*>(bpar->data_ptr()) : nullptr;
        hipLaunchKernelGGL((bn_stats_finalize<rstat_t, wt_t>), dim3(fgrid),
                           dim3(MSBN_BLOCK), 0, stream, ws.data_ptr<double>(),
                           nchunks, L.C, (double)count, (float)eps, mean_p,
                           invstd_p, count_p, rm, rv, (float)momentum, wp, bp,
                           scale_p, shift_p);
      });
    });
  });
}

}  // namespace

// =====================================================================
// public ops
// =====================================================================

std::tuple<at::Tensor, at::Tensor> batch_norm_stats(const at::Tensor& input,
                                                    double eps) {
  const int64_t C = input.size(1);
  auto opts = input.options().dtype(at::kFloat);
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  if (input.numel() == 0) {
    mean.zero_();
    invstd.zero_();
    return {mean, invstd};
  }
  stats_into(input, eps, mean.data_ptr<float>(), invstd.data_ptr<float>(),
             nullptr, nullptr, nullptr, 0.0);
  return {mean, invstd};
}

void batch_norm_stats_packed(const at::Tensor& input, double eps,
                             at::Tensor& out) {
  const int64_t C = input.size(1);
  TORCH_CHECK(out.is_contiguous() && out.numel() == 2 * C + 1 &&
                  out.scalar_type() == at::kFloat,
              "packed stats buffer must be contiguous fp32 of size 2C+1");
  if (input.numel() == 0) {
    out.zero_();
    return;
  }
  float* p = out.data_ptr<float>();
  stats_into(input, eps, p, p + C, p + 2 * C, nullptr, nullptr, 0.0);
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_gather_stats_packed_coefs(
    const at::Tensor& packed_all, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, bool want_coefs) {
  TORCH_CHECK(packed_all.dim() == 2 && packed_all.is_contiguous() &&
                  packed_all.scalar_type() == at::kFloat,
              "packed_all must be contiguous fp32 [W, 2C+1]");
  const int W = (int)packed_all.size(0);
  const int64_t C = (packed_all.size(1) - 1) / 2;
  auto opts = packed_all.options();
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  auto count_sum = at::empty({1}, opts);
  at::Tensor coefs;
  float* scale_p = nullptr;
  float* shift_p = nullptr;
  if (want_coefs) {
    coefs = at::empty({2 * C}, opts);
    scale_p = coefs.data_ptr<float>();
    shift_p = scale_p + C;
  }
  auto stream = cur_stream();
  const auto rstat_type =
      running_mean.has_value() ? running_mean->scalar_type() : at::kFloat;
  const auto w_type = weight.has_value() ? weight->scalar_type()
                      : bias.has_value() ? bias->scalar_type()
                                         : at::kFloat;
  const int fgrid = (int)cdiv(C, MSBN_BLOCK);
  MSBN_DISPATCH_RSTAT(rstat_type, "gather_stats", [&] {
    MSBN_DISPATCH_WTYPE(w_type, "gather_stats", [&] {
      rstat_t* rm = running_mean.has_value()
                        ? reinterpret_cast<rstat_t*>(running_mean->data_ptr())
                        : nullptr;
      rstat_t* rv = running_var.has_value()
                        ? reinterpret_cast<rstat_t*>(running_var->data_ptr())
                        : nullptr;
      const wt_t* wp =
          weight.has_value()
              ? reinterpret_cast<const wt_t*>(weight->data_ptr())
              : nullptr;
      const wt_t* bp = bias.has_value()
                           ? reinterpret_cast<const wt_t*>(bias->data_ptr())
                           : nullptr;
      hipLaunchKernelGGL((bn_gather_stats<rstat_t, wt_t>), dim3(fgrid),
                         dim3(MSBN_BLOCK), 0, stream,
                         packed_all.data_ptr<float>(), W, C, (float)eps,
                         (float)momentum, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), count_sum.data_ptr<float>(),
                         rm, rv, wp, bp, scale_p, shift_p);
    });
  });
  return {mean, invstd, count_sum, coefs};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> batch_norm_gather_stats_packed(
    const at::Tensor& packed_all, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps) {
  auto r = batch_norm_gather_stats_packed_coefs(
      packed_all, running_mean, running_var, momentum, eps, c10::nullopt,
      c10::nullopt, false);
  return {std::get<0>(r), std::get<1>(r), std::get<2>(r)};
}

// Fused local (world_size==1) stats: partial+finalize with running-stats
// update AND [scale|shift] coefs in ONE finalize launch.
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_stats_local(const at::Tensor& input, double eps,
                       const c10::optional<at::Tensor>& running_mean,
                       const c10::optional<at::Tensor>& running_var,
                       double momentum,
                       const c10::optional<at::Tensor>& weight,
                       const c10::optional<at::Tensor>& bias,
                       bool want_coefs) {
  const int64_t C = input.size(1);
  auto opts = input.options().dtype(at::kFloat);
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  auto count_sum = at::empty({1}, opts);
  at::Tensor coefs;
  float* scale_p = nullptr;
  float* shift_p = nullptr;
  if (want_coefs) {
    coefs = at::empty({2 * C}, opts);
    scale_p = coefs.data_ptr<float>();
    shift_p = scale_p + C;
  }
  TORCH_CHECK(input.numel() > 0, "batch_norm_stats_local: empty input");
  const at::Tensor* rm =
      running_mean.has_value() ? &running_mean.value() : nullptr;
  const at::Tensor* rv =
      running_var.has_value() ? &running_var.value() : nullptr;
  const at::Tensor* wp = weight.has_value() ? &weight.value() : nullptr;
  const at::Tensor* bp = bias.has_value() ? &bias.value() : nullptr;
  stats_into(input, eps, mean.data_ptr<float>(), invstd.data_ptr<float>(),
             count_sum.data_ptr<float>(), rm, rv, momentum, wp, bp, scale_p,
             shift_p);
  return {mean, invstd, count_sum, coefs};
}

std::tuple<at::Tensor, at::Tensor> batch_norm_gather_stats_with_counts(
    const at::Tensor& mean_all, const at::Tensor& invstd_all,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    const at::Tensor& counts) {
  // API-parity wrapper: assemble the packed layout, then run the packed path.
  const int64_t W = mean_all.size(0);
  const int64_t C = mean_all.size(1);
  auto packed = at::empty({W, 2 * C + 1}, mean_all.options().dtype(at::kFloat));
  packed.narrow(1, 0, C).copy_(mean_all);
  packed.narrow(1, C, C).copy_(invstd_all);
  packed.narrow(1, 2 * C, 1).copy_(
      counts.to(at::kFloat).reshape({W, 1}));
  auto out = batch_norm_gather_stats_packed(packed, running_mean, running_var,
                                            momentum, eps);
  return {std::get<0>(out), std::get<1>(out)};
}

namespace {
// compute per-channel (scale, shift) into one [2C] fp32 buffer
at::Tensor make_fwd_coefs(const at::Tensor& mean, const at::Tensor& invstd,
                          const c10::optional<at::Tensor>& weight,
                          const c10::optional<at::Tensor>& bias, int64_t C) {
  auto coefs = at::empty({2 * C}, mean.options().dtype(at::kFloat));
  float* scale = coefs.data_ptr<float>();
  float* shift = scale + C;
  auto stream = cur_stream();
  const int cgrid = (int)cdiv(C, MSBN_BLOCK);
  const auto wtype = weight.has_value() ? weight->scalar_type()
                     : bias.has_value() ? bias->scalar_type()
                                        : at::kFloat;
  MSBN_DISPATCH_RSTAT(wtype, "bn_affine", [&] {
    const rstat_t* w =
        weight.has_value()
            ? reinterpret_cast<const rstat_t*>(weight->data_ptr())
            : nullptr;
    const rstat_t* b = bias.has_value()
                           ? reinterpret_cast<const rstat_t*>(bias->data_ptr())
                           : nullptr;
    hipLaunchKernelGGL((bn_affine_fwd<rstat_t>), dim3(cgrid), dim3(MSBN_BLOCK),
                       0, stream, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), w, b, C, scale, shift);
  });
  return coefs;
}

#define MSBN_DISPATCH_BOOL(FLAG, NAME, ...)                                  \
  [&] {                                                                      \
    if (FLAG) {                                                              \
      constexpr bool NAME = true;                                            \
      return __VA_ARGS__();                                                  \
    } else {                                                                 \
      constexpr bool NAME = false;                                           \
      return __VA_ARGS__();                                                  \
    }                                                                        \
  }()
}  // namespace

at::Tensor bn_make_coefs(const at::Tensor& mean, const at::Tensor& invstd,
                         const c10::optional<at::Tensor>& weight,
                         const c10::optional<at::Tensor>& bias) {
  return make_fwd_coefs(mean, invstd, weight, bias, mean.numel());
}

at::Tensor batch_norm_elemt_act(const at::Tensor& input,
                                const c10::optional<at::Tensor>& residual,
                                const c10::optional<at::Tensor>& weight,
                                const c10::optional<at::Tensor>& bias,
                                const at::Tensor& mean,
                                const at::Tensor& invstd, bool relu,
                                const c10::optional<at::Tensor>& coefs_in) {
  const Layout L = get_layout(input);
  auto out = at::empty_like(input);
  if (input.numel() == 0) return out;
  auto stream = cur_stream();
  auto coefs = coefs_in.has_value()
                   ? *coefs_in
                   : make_fwd_coefs(mean, invstd, weight, bias, L.C);
  float* scale = coefs.data_ptr<float>();
  float* shift = scale + L.C;
  const bool has_res = residual.has_value();
  if (has_res) {
    TORCH_CHECK(residual->sizes() == input.sizes() &&
                    residual->strides() == input.strides() &&
                    residual->scalar_type() == input.scalar_type(),
                "residual must match input shape/strides/dtype");
  }

  const int64_t total = input.numel();
  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "bn_elemt", [&] {
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    const native_t* res =
        has_res ? reinterpret_cast<const native_t*>(residual->data_ptr())
                : nullptr;
    native_t* y = reinterpret_cast<native_t*>(out.data_ptr<scalar_t>());
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    const int64_t inner = L.nhwc ? L.C : L.S;
    const int v = pick_v<native_t>(x, y, res, inner);
    const int grid = elemt_grid(total, v);
    MSBN_DISPATCH_V(v, VMAX, [&] {
      MSBN_DISPATCH_BOOL(relu, RELU_, [&] {
        MSBN_DISPATCH_BOOL(has_res, RES_, [&] {
          if (!L.nhwc) {
            hipLaunchKernelGGL((bn_elemt_nchw<native_t, VV, RELU_, RES_>),
                               dim3(grid), dim3(MSBN_BLOCK), 0, stream, x, res,
                               y, scale, shift, total, L.C, L.S);
          } else {
            hipLaunchKernelGGL((bn_elemt_nhwc<native_t, VV, RELU_, RES_>),
                               dim3(grid), dim3(MSBN_BLOCK), 0, stream, x, res,
                               y, scale, shift, total, L.C);
          }
        });
      });
    });
  });
  return out;
}

at::Tensor batch_norm_elemt(const at::Tensor& input,
                            const c10::optional<at::Tensor>& weight,
                            const c10::optional<at::Tensor>& bias,
                            const at::Tensor& mean, const at::Tensor& invstd,
                            double eps) {
  (void)eps;  // invstd already folds eps
  return batch_norm_elemt_act(input, c10::nullopt, weight, bias, mean, invstd,
                              false, c10::nullopt);
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_backward_reduce_act(
    const at::Tensor& grad_out, const at::Tensor& input,
    const c10::optional<at::Tensor>& residual, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, bool relu_mask, bool input_g,
    bool weight_g, bool bias_g, const c10::optional<at::Tensor>& coefs_in,
    const c10::optional<at::Tensor>& gm_out) {
  const Layout L = get_layout(input);
  if (input.numel() == 0) {
    auto z32 = input.options().dtype(at::kFloat);
    auto combined0 = at::zeros({2 * L.C}, z32);
    const auto wt = weight.has_value() ? weight->scalar_type()
                                       : input.scalar_type();
    at::Tensor gw0, gb0;
    if (weight_g) gw0 = at::zeros({L.C}, input.options().dtype(wt));
    if (bias_g) gb0 = at::zeros({L.C}, input.options().dtype(wt));
    return {combined0.narrow(0, 0, L.C), combined0.narrow(0, L.C, L.C), gw0,
            gb0};
  }
  const bool want_gm = gm_out.has_value();
  if (want_gm) {
    TORCH_CHECK(gm_out->sizes() == input.sizes() &&
                    gm_out->strides() == input.strides() &&
                    gm_out->scalar_type() == input.scalar_type(),
                "gm_out must match input shape/strides/dtype");
  }
  TORCH_CHECK(grad_out.sizes() == input.sizes() &&
                  grad_out.strides() == input.strides(),
              "grad_out must match input shape and layout");
  TORCH_CHECK(!residual.has_value() ||
                  (residual->sizes() == input.sizes() &&
                   residual->strides() == input.strides()),
              "residual must match input shape and layout");
  auto f32 = input.options().dtype(at::kFloat);
  // ONE contiguous buffer for [sum_dy | sum_dy_xmu] -> single all_reduce.
  auto combined = at::empty({2 * L.C}, f32);
  auto sum_dy = combined.narrow(0, 0, L.C);
  auto sum_dy_xmu = combined.narrow(0, L.C, L.C);
  const auto wtype =
      weight.has_value() ? weight->scalar_type() : input.scalar_type();
  auto wopts = input.options().dtype(wtype);
  at::Tensor grad_weight, grad_bias;
  if (weight_g) grad_weight = at::empty({L.C}, wopts);
  if (bias_g) grad_bias = at::empty({L.C}, wopts);

  const bool has_res = residual.has_value();
  float* scale = nullptr;
  float* shift = nullptr;
  at::Tensor coefs;
  if (relu_mask) {
    coefs = coefs_in.has_value()
                ? *coefs_in
                : make_fwd_coefs(mean, invstd, weight, bias, L.C);
    scale = coefs.data_ptr<float>();
    shift = scale + L.C;
  }

  auto stream = cur_stream();
  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "bn_bwd_reduce", [&] {
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    const native_t* dy =
        reinterpret_cast<const native_t*>(grad_out.data_ptr<scalar_t>());
    const native_t* res =
        has_res ? reinterpret_cast<const native_t*>(residual->data_ptr())
                : nullptr;
    native_t* gm =
        want_gm ? reinterpret_cast<native_t*>(gm_out->data_ptr()) : nullptr;
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    at::Tensor ws;
    int nchunks = 0;
    MSBN_DISPATCH_BOOL(relu_mask, MASK_, [&] {
      MSBN_DISPATCH_BOOL(has_res, RES_, [&] {
        MSBN_DISPATCH_BOOL(want_gm, GM_, [&] {
        if (!L.nhwc) {
          const int v = pick_v<native_t>(x, dy, res, L.S);
          if (v == 1) {
            auto g = flat_grid(L.C, L.rows, kTargetBlocks);
            nchunks = g.nchunks;
            ws = at::empty({(int64_t)nchunks * L.C * 2},
                           input.options().dtype(at::kDouble));
            hipLaunchKernelGGL(
                (bn_bwd_reduce_partial_nchw_flat<native_t, MASK_, RES_, GM_>),
                g.grid, dim3(MSBN_BLOCK), 0, stream, dy, x, res, gm,
                mean.data_ptr<float>(), scale, shift, ws.data_ptr<double>(),
                L.N, L.C, L.S, g.chunk_len);
          } else {
            auto g = nchw_grid(L.N, L.C, L.S, v);
            nchunks = g.nchunks;
            ws = at::empty({(int64_t)nchunks * L.C * 2},
                           input.options().dtype(at::kDouble));
            MSBN_DISPATCH_V(v, VMAX, [&] {
              hipLaunchKernelGGL(
                  (bn_bwd_reduce_partial_nchw<native_t, VV, MASK_, RES_, GM_>),
                  g.grid, dim3(MSBN_BLOCK), 0, stream, dy, x, res, gm,
                  mean.data_ptr<float>(), scale, shift, ws.data_ptr<double>(),
                  L.N, L.C, L.S, g.chunkN, g.chunkS);
            });
          }
        } else {
          const int v = pick_v<native_t>(x, dy, res, L.C);
          auto g = nhwc_grid(L.rows, L.C, v);
          nchunks = g.nchunks;
          ws = at::empty({(int64_t)nchunks * L.C * 2},
                         input.options().dtype(at::kDouble));
          MSBN_DISPATCH_V(v, VMAX, [&] {
            hipLaunchKernelGGL(
                (bn_bwd_reduce_partial_nhwc<native_t, VV, MASK_, RES_, GM_>),
                g.grid, dim3(MSBN_BLOCK), 0, stream, dy, x, res, gm,
                mean.data_ptr<float>(), scale, shift, ws.data_ptr<double>(),
                L.rows, L.C, g.chunk_rows, g.lpr);
          });
        }
        });
      });
    });
    const int fgrid = (int)cdiv(L.C, kFinalizeWavesPerBlock);
    MSBN_DISPATCH_RSTAT(wtype, "bn_bwd_reduce", [&] {
      rstat_t* gw = weight_g ? reinterpret_cast<rstat_t*>(grad_weight.data_ptr())
                             : nullptr;
      rstat_t* gb =
          bias_g ? reinterpret_cast<rstat_t*>(grad_bias.data_ptr()) : nullptr;
      hipLaunchKernelGGL((bn_bwd_reduce_finalize<rstat_t>), dim3(fgrid),
                         dim3(MSBN_BLOCK), 0, stream, ws.data_ptr<double>(),
                         nchunks, L.C, invstd.data_ptr<float>(),
                         sum_dy.data_ptr<float>(),
                         sum_dy_xmu.data_ptr<float>(), gw, gb);
    });
  });
  (void)input_g;
  return {sum_dy, sum_dy_xmu, grad_weight, grad_bias};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_backward_reduce(const at::Tensor& grad_out, const at::Tensor& input,
                           const at::Tensor& mean, const at::Tensor& invstd,
                           const c10::optional<at::Tensor>& weight, bool input_g,
                           bool weight_g, bool bias_g) {
  return batch_norm_backward_reduce_act(grad_out, input, c10::nullopt, mean,
                                        invstd, weight, c10::nullopt, false,
                                        input_g, weight_g, bias_g,
                                        c10::nullopt, c10::nullopt);
}

std::tuple<at::Tensor, at::Tensor> batch_norm_backward_elemt_act(
    const at::Tensor& grad_out, const at::Tensor& input,
    const c10::optional<at::Tensor>& residual, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const c10::optional<at::Tensor>& bias, const at::Tensor& sum_dy,
    const at::Tensor& sum_dy_xmu, const at::Tensor& count, bool relu_mask,
    bool want_res_grad, const c10::optional<at::Tensor>& coefs_in) {
  const Layout L = get_layout(input);
  auto dx = at::empty_like(grad_out);
  at::Tensor dres;
  if (want_res_grad) dres = at::empty_like(grad_out);
  if (input.numel() == 0) return {dx, dres};
  // total count as a device fp32 scalar (no host sync)
  at::Tensor count_sum;
  if (count.numel() == 1 && count.scalar_type() == at::kFloat) {
    count_sum = count;
  } else {
    count_sum = count.to(at::kFloat).sum().reshape({1});
  }
  auto stream = cur_stream();
  auto f32 = input.options().dtype(at::kFloat);
  auto coefs = at::empty({3 * L.C}, f32);
  float* ca = coefs.data_ptr<float>();
  float* cb = ca + L.C;
  float* cd = cb + L.C;
  const int cgrid = (int)cdiv(L.C, MSBN_BLOCK);
  const auto wtype =
      weight.has_value() ? weight->scalar_type() : at::kFloat;
  MSBN_DISPATCH_RSTAT(wtype, "bn_bwd_elemt", [&] {
    const rstat_t* w =
        weight.has_value()
            ? reinterpret_cast<const rstat_t*>(weight->data_ptr())
            : nullptr;
    hipLaunchKernelGGL((bn_affine_bwd<rstat_t>), dim3(cgrid), dim3(MSBN_BLOCK),
                       0, stream, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), w, sum_dy.data_ptr<float>(),
                       sum_dy_xmu.data_ptr<float>(),
                       count_sum.data_ptr<float>(), L.C, ca, cb, cd);
  });
  const bool has_res = residual.has_value();
  float* scale = nullptr;
  float* shift = nullptr;
  at::Tensor fcoefs;
  if (relu_mask) {
    fcoefs = coefs_in.has_value()
                 ? *coefs_in
                 : make_fwd_coefs(mean, invstd, weight, bias, L.C);
    scale = fcoefs.data_ptr<float>();
    shift = scale + L.C;
  }

  const int64_t total = input.numel();
  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "bn_bwd_elemt", [&] {
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    const native_t* dy =
        reinterpret_cast<const native_t*>(grad_out.data_ptr<scalar_t>());
    const native_t* res =
        has_res ? reinterpret_cast<const native_t*>(residual->data_ptr())
                : nullptr;
    native_t* o = reinterpret_cast<native_t*>(dx.data_ptr<scalar_t>());
    native_t* dr = want_res_grad
                       ? reinterpret_cast<native_t*>(dres.data_ptr<scalar_t>())
                       : nullptr;
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    const int64_t inner = L.nhwc ? L.C : L.S;
    const int v = pick_v<native_t>(x, dy, o, inner);
    const int grid = elemt_grid(total, v);
    MSBN_DISPATCH_V(v, VMAX, [&] {
      MSBN_DISPATCH_BOOL(relu_mask, MASK_, [&] {
        MSBN_DISPATCH_BOOL(has_res, RES_, [&] {
          MSBN_DISPATCH_BOOL(want_res_grad, RESG_, [&] {
            if (!L.nhwc) {
              hipLaunchKernelGGL(
                  (bn_bwd_elemt_nchw<native_t, VV, MASK_, RES_, RESG_>),
                  dim3(grid), dim3(MSBN_BLOCK), 0, stream, dy, x, res, o, dr,
                  ca, cb, cd, scale, shift, total, L.C, L.S);
            } else {
              hipLaunchKernelGGL(
                  (bn_bwd_elemt_nhwc<native_t, VV, MASK_, RES_, RESG_>),
                  dim3(grid), dim3(MSBN_BLOCK), 0, stream, dy, x, res, o, dr,
                  ca, cb, cd, scale, shift, total, L.C);
            }
          });
        });
      });
    });
  });
  return {dx, dres};
}

at::Tensor batch_norm_backward_elemt(
    const at::Tensor& grad_out, const at::Tensor& input, const at::Tensor& mean,
    const at::Tensor& invstd, const c10::optional<at::Tensor>& weight,
    const at::Tensor& sum_dy, const at::Tensor& sum_dy_xmu,
    const at::Tensor& count) {
  return std::get<0>(batch_norm_backward_elemt_act(
      grad_out, input, c10::nullopt, mean, invstd, weight, c10::nullopt,
      sum_dy, sum_dy_xmu, count, false, false, c10::nullopt));
}

// ---------------------------------------------------------------------------
// small-plane world-1 fused path (single-launch forward / backward)
// ---------------------------------------------------------------------------
namespace {
// Measured crossover (tools/fused_local_bench.py, gpurun_out/flb_*.log):
// block-per-channel wins up to plane ~8K elems (128bs x 8x8 and smaller);
// beyond that a single block serializes too much plane work and the
// two-stage chip-filling pipeline wins (3.4x at plane 32K).
constexpr int64_t kFusedSmallPlaneMax = 8192;
constexpr int64_t kFusedSmallMinC = 64;  // >= 64 blocks on the grid

bool fp32_or_absent(const c10::optional<at::Tensor>& t) {
  return !t.has_value() || !t->defined() || t->scalar_type() == at::kFloat;
}
}  // namespace

bool bn_fused_local_eligible(const at::Tensor& input,
                             const c10::optional<at::Tensor>& weight,
                             const c10::optional<at::Tensor>& bias,
                             const c10::optional<at::Tensor>& running_mean,
                             const c10::optional<at::Tensor>& running_var) {
  if (!input.is_cuda() || input.dim() < 2 || input.numel() == 0) return false;
  const auto st = input.scalar_type();
  if (st != at::kFloat && st != at::kBFloat16 && st != at::kHalf) return false;
  if (!input.is_contiguous()) return false;  // NCHW only
  // a 4-D tensor that REPORTS channels-last gets the NHWC kernels instead
  if (input.dim() >= 4 &&
      input.suggest_memory_format() == at::MemoryFormat::ChannelsLast)
    return false;
  const int64_t C = input.size(1);
  const int64_t N = input.size(0);
  const int64_t plane = input.numel() / C;
  const int64_t S = plane / std::max<int64_t>(N, 1);
  const int vmax = 16 / (int)input.element_size();
  const bool vec = (S % vmax) == 0;
  // vectorized blocks chew 4-8x more plane per cycle -> higher crossover
  // (measured: win at 16K, slight loss at 32K — gpurun_out/flb2.log)
  const int64_t lim = vec ? 2 * kFusedSmallPlaneMax : kFusedSmallPlaneMax;
  if (C < kFusedSmallMinC || plane > lim) return false;
  return fp32_or_absent(weight) && fp32_or_absent(bias) &&
         fp32_or_absent(running_mean) && fp32_or_absent(running_var);
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_fwd_fused_local(const at::Tensor& input,
                           const c10::optional<at::Tensor>& residual,
                           const c10::optional<at::Tensor>& weight,
                           const c10::optional<at::Tensor>& bias, double eps,
                           double momentum,
                           const c10::optional<at::Tensor>& running_mean,
                           const c10::optional<at::Tensor>& running_var,
                           bool relu) {
  const int64_t C = input.size(1);
  const int64_t N = input.size(0);
  const int64_t S = input.numel() / std::max<int64_t>(N * C, 1);
  auto opts = input.options().dtype(at::kFloat);
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  auto count = at::empty({1}, opts);
  auto coefs = at::empty({2 * C}, opts);
  auto y = at::empty_like(input);
  auto stream = cur_stream();
  const bool has_res = residual.has_value() && residual->defined();
  TORCH_CHECK(!has_res || residual->is_contiguous(),
              "fused local: residual must match NCHW layout");
  float* rm = running_mean.has_value() && running_mean->defined()
                  ? running_mean->data_ptr<float>()
                  : nullptr;
  float* rv = running_var.has_value() && running_var->defined()
                  ? running_var->data_ptr<float>()
                  : nullptr;
  const float* w = weight.has_value() && weight->defined()
                       ? weight->data_ptr<float>()
                       : nullptr;
  const float* b =
      bias.has_value() && bias->defined() ? bias->data_ptr<float>() : nullptr;
  float* scale_p = coefs.data_ptr<float>();
  float* shift_p = scale_p + C;
  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "bn_fwd_fused_local", [&] {
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    const native_t* res =
        has_res ? reinterpret_cast<const native_t*>(
                      residual->data_ptr<scalar_t>())
                : nullptr;
    native_t* yp = reinterpret_cast<native_t*>(y.data_ptr<scalar_t>());
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    const int v = pick_v<native_t>(x, res, yp, S);
    MSBN_DISPATCH_V(v, VMAX, [&] {
      auto launch = [&](auto relu_c, auto res_c) {
        hipLaunchKernelGGL(
            (bn_fwd_fused_small_nchw<native_t, VV, decltype(relu_c)::value,
                                     decltype(res_c)::value>),
            dim3((unsigned)C), dim3(MSBN_BLOCK), 0, stream, x, res, yp,
            (int)N, C, (int)S, (float)eps, (float)momentum,
            mean.data_ptr<float>(), invstd.data_ptr<float>(),
            count.data_ptr<float>(), rm, rv, w, b, scale_p, shift_p);
      };
      using T = std::true_type;
      using F = std::false_type;
      if (relu && has_res) launch(T{}, T{});
      else if (relu) launch(T{}, F{});
      else if (has_res) launch(F{}, T{});
      else launch(F{}, F{});
    });
  });
  return {y, mean, invstd, count, coefs};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor>
batch_norm_bwd_fused_local(const at::Tensor& grad_out, const at::Tensor& input,
                           const c10::optional<at::Tensor>& residual,
                           const at::Tensor& mean, const at::Tensor& invstd,
                           const c10::optional<at::Tensor>& weight,
                           const c10::optional<at::Tensor>& coefs,
                           bool relu_mask, bool want_res_grad, bool weight_g,
                           bool bias_g) {
  const int64_t C = input.size(1);
  const int64_t N = input.size(0);
  const int64_t S = input.numel() / std::max<int64_t>(N * C, 1);
  TORCH_CHECK(grad_out.is_contiguous(), "fused local bwd: grad layout");
  TORCH_CHECK(!relu_mask || (coefs.has_value() && coefs->defined()),
              "fused local bwd: relu mask needs coefs");
  auto opts = input.options().dtype(at::kFloat);
  auto dx = at::empty_like(input);
  at::Tensor gw, gb, dres;
  if (weight_g) gw = at::empty({C}, opts);
  if (bias_g) gb = at::empty({C}, opts);
  if (want_res_grad) dres = at::empty_like(grad_out);
  auto stream = cur_stream();
  const bool has_res = residual.has_value() && residual->defined();
  const float* scale_p = nullptr;
  const float* shift_p = nullptr;
  if (relu_mask) {
    scale_p = coefs->data_ptr<float>();
    shift_p = scale_p + C;
  }
  const float* w = weight.has_value() && weight->defined()
                       ? weight->data_ptr<float>()
                       : nullptr;
  MSBN_DISPATCH_FLOAT_TYPES(input.scalar_type(), "bn_bwd_fused_local", [&] {
    const native_t* dy =
        reinterpret_cast<const native_t*>(grad_out.data_ptr<scalar_t>());
    const native_t* x =
        reinterpret_cast<const native_t*>(input.data_ptr<scalar_t>());
    const native_t* res =
        (relu_mask && has_res)
            ? reinterpret_cast<const native_t*>(residual->data_ptr<scalar_t>())
            : nullptr;
    native_t* dxp = reinterpret_cast<native_t*>(dx.data_ptr<scalar_t>());
    native_t* drp = want_res_grad
                        ? reinterpret_cast<native_t*>(dres.data_ptr<scalar_t>())
                        : nullptr;
    constexpr int VMAX = 16 / (int)sizeof(native_t);
    const int v0 = pick_v<native_t>(dy, x, res, S);
    const int v1 = pick_v<native_t>(dxp, drp, nullptr, S);
    const int v = std::min(v0, v1);
    MSBN_DISPATCH_V(v, VMAX, [&] {
      auto launch = [&](auto mask_c, auto res_c, auto resg_c) {
        hipLaunchKernelGGL(
            (bn_bwd_fused_small_nchw<native_t, VV, decltype(mask_c)::value,
                                     decltype(res_c)::value,
                                     decltype(resg_c)::value>),
            dim3((unsigned)C), dim3(MSBN_BLOCK), 0, stream, dy, x, res, dxp,
            drp, mean.data_ptr<float>(), invstd.data_ptr<float>(), scale_p,
            shift_p, w, weight_g ? gw.data_ptr<float>() : nullptr,
            bias_g ? gb.data_ptr<float>() : nullptr, (int)N, C, (int)S);
      };
      using T = std::true_type;
      using F = std::false_type;
      const bool mres = relu_mask && has_res;
      if (relu_mask && mres && want_res_grad) launch(T{}, T{}, T{});
      else if (relu_mask && mres) launch(T{}, T{}, F{});
      else if (relu_mask && want_res_grad) launch(T{}, F{}, T{});
      else if (relu_mask) launch(T{}, F{}, F{});
      else if (want_res_grad) launch(F{}, F{}, T{});
      else launch(F{}, F{}, F{});
    });
  });
  return {dx, gw, gb, dres};
}

}  // namespace msbn
