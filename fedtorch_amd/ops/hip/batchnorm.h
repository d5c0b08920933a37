// Fused spatial BatchNorm (NCHW) for gfx950 — training fwd/bwd.
//
// Replaces MIOpen's 3-kernel fwd / 3-kernel bwd sequences (profile:
// profiles/r01_bench_notes.md — MIOpenBatchNorm* is ~40 % of ResNet-20
// step kernel time) with 2+2 streaming kernels:
//   fwd:  bn_stats  (per-channel partial sum/sumsq, one partial per block)
//         bn_norm   (finalizes the partials, elementwise normalize + affine
//                    [+ReLU], saves mean/ivar, optionally updates running
//                    stats)
//   bwd:  bn_bwd_stats (per-channel partial sum(dy), sum(dy*xhat))
//         bn_bwd_dx    (finalize + elementwise dx [+ReLU mask], writes
//                       dgamma/dbeta)
//
// Reduction scheme: NO global atomics and NO zero-init kernel — each block
// writes its partial to part[C, B, 2] with a plain store and the consumer
// kernel sums the B (≤64) partials per channel (uniform, L2-hot loads).
// The earlier atomic variant needed a torch::zeros fill kernel per call and
// 2048-block launches with ~1 vector-load per thread; this version launches
// ~512 blocks with 2-8 iterations per thread (measured: stats kernels drop
// from ~21 us to memory-bound).
//
// Layout: NCHW contiguous; element (n, c, i) at ((n*C + c)*HW + i).
// All reductions fp32; index math in u32 (host asserts N*HW < 2^31).
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

// ---- 16-byte vectorized element types ------------------------------------
template <typename T> struct BnVec;
template <> struct BnVec<float> {
  struct alignas(16) V { float d[4]; };
  static constexpr int N = 4;
};
template <> struct BnVec<__hip_bfloat16> {
  struct alignas(16) V { __hip_bfloat16 d[8]; };
  static constexpr int N = 8;
};
template <> struct BnVec<double> {
  struct alignas(16) V { double d[2]; };
  static constexpr int N = 2;
};
// scalar fallback "vector" of 1
template <typename T> struct Bn1 {
  struct V { T d[1]; };
  static constexpr int N = 1;
};

// sum the B partials of channel c (uniform across the block; B ≤ 64)
__device__ __forceinline__ void bn_finalize(const float* __restrict__ part,
                                            long c, int B, float* s,
                                            float* q) {
  float a = 0.f, b = 0.f;
  const float* p = part + c * (long)B * 2;
  for (int i = 0; i < B; ++i) {
    a += p[2 * i];
    b += p[2 * i + 1];
  }
  *s = a;
  *q = b;
}

// grid: (B, C); block (b, c) reduces its slice of channel c and stores one
// partial {sum, sumsq} to part[c, b].
template <typename T, typename VT, int VN>
__global__ void bn_stats_k(const T* __restrict__ x, long N, long C, long HW,
                           float* __restrict__ part /*[C,B,2]*/) {
  const long c = blockIdx.y;
  const unsigned hw = (unsigned)HW;
  const unsigned per_v = (unsigned)((N * HW) / VN);
  const long chw = C * HW;
  float sum = 0.f, sumsq = 0.f;
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned t = blockIdx.x * blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const unsigned e = t * VN;
    const unsigned n = e / hw, i = e % hw;
    VT v = *reinterpret_cast<const VT*>(x + n * chw + c * HW + i);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float f = (float)v.d[j];
      sum += f;
      sumsq = fmaf(f, f, sumsq);
    }
  }
  float bs = block_reduce<0>(sum);
  __syncthreads();
  float bq = block_reduce<0>(sumsq);
  if (threadIdx.x == 0) {
    float* p = part + (c * gridDim.x + blockIdx.x) * 2;
    p[0] = bs;
    p[1] = bq;
  }
}

// elementwise normalize; finalizes mean/ivar from the partials (cheap,
// uniform) and lane 0 of block (0, c) writes the saved mean/ivar and
// running stats.
// `res` (optional): residual input added before the ReLU — fuses the
// ResNet block tail `relu(bn(conv(x)) + identity)` into this kernel.
template <typename T, typename VT, int VN>
__global__ void bn_norm_k(const T* __restrict__ x, T* __restrict__ y,
                          const float* __restrict__ part, int B,
                          const float* __restrict__ weight,
                          const float* __restrict__ bias,
                          float* __restrict__ save_mean,
                          float* __restrict__ save_ivar,
                          float* __restrict__ running_mean,
                          float* __restrict__ running_var,
                          const T* __restrict__ res, long N, long C,
                          long HW, float eps, float momentum, int relu) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  float s, q;
  bn_finalize(part, c, B, &s, &q);
  const float mean = s * inv_n;
  const float var = fmaxf(q * inv_n - mean * mean, 0.f);
  const float ivar = rsqrtf(var + eps);
  const float w = weight ? weight[c] : 1.f;
  const float b = bias ? bias[c] : 0.f;
  const float scale = w * ivar;
  const float shift = b - mean * scale;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    save_mean[c] = mean;
    save_ivar[c] = ivar;
    if (running_mean) {
      // torch uses the UNBIASED variance for running stats
      float ub = var * (float)per_ch / (float)(per_ch > 1 ? per_ch - 1 : 1);
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * ub;
    }
  }
  const unsigned hw = (unsigned)HW;
  const unsigned per_v = (unsigned)(per_ch / VN);
  const long chw = C * HW;
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned t = blockIdx.x * blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const unsigned e = t * VN;
    const unsigned n = e / hw, i = e % hw;
    const long base = n * chw + c * HW + i;
    VT v = *reinterpret_cast<const VT*>(x + base);
    if (res) {
      VT r = *reinterpret_cast<const VT*>(res + base);
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        float f = fmaf((float)v.d[j], scale, shift) + (float)r.d[j];
        v.d[j] = (T)(relu ? fmaxf(f, 0.f) : f);
      }
    } else {
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        float f = fmaf((float)v.d[j], scale, shift);
        v.d[j] = (T)(relu ? fmaxf(f, 0.f) : f);
      }
    }
    *reinterpret_cast<VT*>(y + base) = v;
  }
}

// bwd partials: {sum(dy), sum(dy * xhat)} per (channel, block) (+ReLU mask
// on dy via the saved output y).
template <typename T, typename VT, int VN>
__global__ void bn_bwd_stats_k(const T* __restrict__ dy,
                               const T* __restrict__ x,
                               const T* __restrict__ yv,
                               const float* __restrict__ save_mean,
                               const float* __restrict__ save_ivar, long N,
                               long C, long HW,
                               float* __restrict__ part /*[C,B,2]*/,
                               int relu) {
  const long c = blockIdx.y;
  const float mean = save_mean[c], ivar = save_ivar[c];
  float s1 = 0.f, s2 = 0.f;
  const unsigned hw = (unsigned)HW;
  const unsigned per_v = (unsigned)((N * HW) / VN);
  const long chw = C * HW;
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned t = blockIdx.x * blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const unsigned e = t * VN;
    const unsigned n = e / hw, i = e % hw;
    const long base = n * chw + c * HW + i;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT xv = *reinterpret_cast<const VT*>(x + base);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float gg = (float)g.d[j];
      if (relu && (float)yv[base + j] <= 0.f) gg = 0.f;
      float xh = ((float)xv.d[j] - mean) * ivar;
      s1 += gg;
      s2 = fmaf(gg, xh, s2);
    }
  }
  float b1 = block_reduce<0>(s1);
  __syncthreads();
  float b2 = block_reduce<0>(s2);
  if (threadIdx.x == 0) {
    float* p = part + (c * gridDim.x + blockIdx.x) * 2;
    p[0] = b1;
    p[1] = b2;
  }
}

// dx = (gamma*ivar) * (dy - sum(dy)/n - xhat * sum(dy*xhat)/n)
// block (0, c) lane 0 writes dgamma[c] = sum(dy*xhat), dbeta[c] = sum(dy).
// `dres` (optional): gradient of the fused residual input = ReLU-masked dy.
template <typename T, typename VT, int VN>
__global__ void bn_bwd_dx_k(const T* __restrict__ dy, const T* __restrict__ x,
                            const T* __restrict__ yv,
                            const float* __restrict__ part, int B,
                            const float* __restrict__ save_mean,
                            const float* __restrict__ save_ivar,
                            const float* __restrict__ weight,
                            T* __restrict__ dx, T* __restrict__ dres,
                            float* __restrict__ dweight,
                            float* __restrict__ dbias, long N, long C,
                            long HW, int relu) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  const float mean = save_mean[c], ivar = save_ivar[c];
  const float w = weight ? weight[c] : 1.f;
  float sum_dy, sum_dyxh;
  bn_finalize(part, c, B, &sum_dy, &sum_dyxh);
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    dweight[c] = sum_dyxh;
    dbias[c] = sum_dy;
  }
  const float k1 = w * ivar;
  const float m_dy = sum_dy * inv_n;
  const float m_dyxh = sum_dyxh * inv_n;
  const unsigned hw = (unsigned)HW;
  const unsigned per_v = (unsigned)(per_ch / VN);
  const long chw = C * HW;
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned t = blockIdx.x * blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const unsigned e = t * VN;
    const unsigned n = e / hw, i = e % hw;
    const long base = n * chw + c * HW + i;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT xv = *reinterpret_cast<const VT*>(x + base);
    VT gr;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float gg = (float)g.d[j];
      if (relu && (float)yv[base + j] <= 0.f) gg = 0.f;
      if (dres) gr.d[j] = (T)gg;
      float xh = ((float)xv.d[j] - mean) * ivar;
      g.d[j] = (T)(k1 * (gg - m_dy - xh * m_dyxh));
    }
    *reinterpret_cast<VT*>(dx + base) = g;
    if (dres) *reinterpret_cast<VT*>(dres + base) = gr;
  }
}
