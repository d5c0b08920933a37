// Fused spatial BatchNorm (NCHW, fp32) for gfx950 — training fwd/bwd.
//
// Replaces MIOpen's 3-kernel fwd / 3-kernel bwd sequences (profile:
// profiles/r01_bench_notes.md — MIOpenBatchNorm* is ~40 % of ResNet-20
// step kernel time) with 2+2 streaming kernels:
//   fwd:  bn_stats  (per-channel sum/sumsq, grid-split + atomics)
//         bn_norm   (elementwise normalize + affine [+ReLU], saves mean/ivar
//                    and optionally updates running stats)
//   bwd:  bn_bwd_stats (per-channel sum(dy), sum(dy*xhat))
//         bn_bwd_dx    (elementwise dx [+ReLU mask], writes dgamma/dbeta)
//
// Layout: NCHW contiguous; element (n, c, i) at ((n*C + c)*HW + i).
// All reductions fp32; one channel's data is reduced by multiple blocks via
// global atomics on a small [C,2] scratch (Guideline 12: per-wave partials
// first, one atomic per block).
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

template <typename T>
__device__ __forceinline__ float bn_ld(const T* p, long i) {
  return (float)p[i];
}
template <>
__device__ __forceinline__ float bn_ld<__hip_bfloat16>(
    const __hip_bfloat16* p, long i) {
  return __bfloat162float(p[i]);
}
template <typename T>
__device__ __forceinline__ void bn_st(T* p, long i, float v) {
  p[i] = (T)v;
}
template <>
__device__ __forceinline__ void bn_st<__hip_bfloat16>(
    __hip_bfloat16* p, long i, float v) {
  p[i] = __float2bfloat16(v);
}

// grid: (spatial_chunks, C); each block reduces a chunk of one channel.
template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x, long N, long C,
                                long HW, float* __restrict__ stats /*[C,2]*/) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  float sum = 0.f, sumsq = 0.f;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_ch;
       t += stride) {
    const long n = t / HW, i = t % HW;
    float v = bn_ld(x, (n * C + c) * HW + i);
    sum += v;
    sumsq = fmaf(v, v, sumsq);
  }
  float bs = block_reduce<0>(sum);
  __syncthreads();
  float bq = block_reduce<0>(sumsq);
  if (threadIdx.x == 0) {
    atomicAdd(&stats[2 * c], bs);
    atomicAdd(&stats[2 * c + 1], bq);
  }
}

// elementwise normalize; also finalizes mean/ivar from stats once per
// channel (cheap recompute per block) and lane 0 of block (0, c) updates
// the saved mean/ivar and running stats.
template <typename T>
__global__ void bn_norm_kernel(const T* __restrict__ x,
                               T* __restrict__ y,
                               const float* __restrict__ stats,
                               const float* __restrict__ weight,
                               const float* __restrict__ bias,
                               float* __restrict__ save_mean,
                               float* __restrict__ save_ivar,
                               float* __restrict__ running_mean,
                               float* __restrict__ running_var,
                               long N, long C, long HW, float eps,
                               float momentum, int relu) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  const float mean = stats[2 * c] * inv_n;
  const float var = fmaxf(stats[2 * c + 1] * inv_n - mean * mean, 0.f);
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
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_ch;
       t += stride) {
    const long n = t / HW, i = t % HW;
    const long idx = (n * C + c) * HW + i;
    float v = fmaf(bn_ld(x, idx), scale, shift);
    bn_st(y, idx, relu ? fmaxf(v, 0.f) : v);
  }
}

// bwd reductions: sum(dy), sum(dy * xhat) per channel (+ReLU mask on dy).
template <typename T>
__global__ void bn_bwd_stats_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    const float* __restrict__ save_mean,
                                    const float* __restrict__ save_ivar,
                                    long N, long C, long HW,
                                    float* __restrict__ red /*[C,2]*/,
                                    int relu) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float mean = save_mean[c], ivar = save_ivar[c];
  float s1 = 0.f, s2 = 0.f;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_ch;
       t += stride) {
    const long n = t / HW, i = t % HW;
    const long idx = (n * C + c) * HW + i;
    float g = bn_ld(dy, idx);
    if (relu && bn_ld(y, idx) <= 0.f) g = 0.f;
    float xh = (bn_ld(x, idx) - mean) * ivar;
    s1 += g;
    s2 = fmaf(g, xh, s2);
  }
  float b1 = block_reduce<0>(s1);
  __syncthreads();
  float b2 = block_reduce<0>(s2);
  if (threadIdx.x == 0) {
    atomicAdd(&red[2 * c], b1);
    atomicAdd(&red[2 * c + 1], b2);
  }
}

// dx = (gamma*ivar) * (dy - sum(dy)/n - xhat * sum(dy*xhat)/n)
// block (0, c) lane 0 writes dgamma[c] = sum(dy*xhat), dbeta[c] = sum(dy).
template <typename T>
__global__ void bn_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const T* __restrict__ y,
                                 const float* __restrict__ red,
                                 const float* __restrict__ save_mean,
                                 const float* __restrict__ save_ivar,
                                 const float* __restrict__ weight,
                                 T* __restrict__ dx,
                                 float* __restrict__ dweight,
                                 float* __restrict__ dbias,
                                 long N, long C, long HW, int relu) {
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  const float mean = save_mean[c], ivar = save_ivar[c];
  const float w = weight ? weight[c] : 1.f;
  const float sum_dy = red[2 * c], sum_dyxh = red[2 * c + 1];
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    dweight[c] = sum_dyxh;
    dbias[c] = sum_dy;
  }
  const float k1 = w * ivar;
  const float m_dy = sum_dy * inv_n;
  const float m_dyxh = sum_dyxh * inv_n;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_ch;
       t += stride) {
    const long n = t / HW, i = t % HW;
    const long idx = (n * C + c) * HW + i;
    float g = bn_ld(dy, idx);
    if (relu && bn_ld(y, idx) <= 0.f) g = 0.f;
    float xh = (bn_ld(x, idx) - mean) * ivar;
    bn_st(dx, idx, k1 * (g - m_dy - xh * m_dyxh));
  }
}

// ---- 16-byte vectorized variants (HW % VEC == 0; G13: scalar bf16 loads
// cost ~2-2.5x on memory-bound streaming kernels) -------------------------
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

template <typename T>
__global__ void bn_stats_kernel_v(const T* __restrict__ x, long N, long C,
                                  long HW, float* __restrict__ stats) {
  using VT = typename BnVec<T>::V;
  constexpr int VN = BnVec<T>::N;
  const long c = blockIdx.y;
  const long per_v = (N * HW) / VN;
  float sum = 0.f, sumsq = 0.f;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const long e = t * VN;
    const long n = e / HW, i = e % HW;
    VT v = *reinterpret_cast<const VT*>(x + (n * C + c) * HW + i);
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
    atomicAdd(&stats[2 * c], bs);
    atomicAdd(&stats[2 * c + 1], bq);
  }
}

template <typename T>
__global__ void bn_norm_kernel_v(const T* __restrict__ x, T* __restrict__ y,
                                 const float* __restrict__ stats,
                                 const float* __restrict__ weight,
                                 const float* __restrict__ bias,
                                 float* __restrict__ save_mean,
                                 float* __restrict__ save_ivar,
                                 float* __restrict__ running_mean,
                                 float* __restrict__ running_var,
                                 long N, long C, long HW, float eps,
                                 float momentum, int relu) {
  using VT = typename BnVec<T>::V;
  constexpr int VN = BnVec<T>::N;
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  const float mean = stats[2 * c] * inv_n;
  const float var = fmaxf(stats[2 * c + 1] * inv_n - mean * mean, 0.f);
  const float ivar = rsqrtf(var + eps);
  const float w = weight ? weight[c] : 1.f;
  const float b = bias ? bias[c] : 0.f;
  const float scale = w * ivar;
  const float shift = b - mean * scale;
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    save_mean[c] = mean;
    save_ivar[c] = ivar;
    if (running_mean) {
      float ub = var * (float)per_ch / (float)(per_ch > 1 ? per_ch - 1 : 1);
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * ub;
    }
  }
  const long per_v = per_ch / VN;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const long e = t * VN;
    const long n = e / HW, i = e % HW;
    const long base = (n * C + c) * HW + i;
    VT v = *reinterpret_cast<const VT*>(x + base);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float f = fmaf((float)v.d[j], scale, shift);
      v.d[j] = (T)(relu ? fmaxf(f, 0.f) : f);
    }
    *reinterpret_cast<VT*>(y + base) = v;
  }
}

template <typename T>
__global__ void bn_bwd_stats_kernel_v(const T* __restrict__ dy,
                                      const T* __restrict__ x,
                                      const T* __restrict__ yv,
                                      const float* __restrict__ save_mean,
                                      const float* __restrict__ save_ivar,
                                      long N, long C, long HW,
                                      float* __restrict__ red, int relu) {
  using VT = typename BnVec<T>::V;
  constexpr int VN = BnVec<T>::N;
  const long c = blockIdx.y;
  const float mean = save_mean[c], ivar = save_ivar[c];
  float s1 = 0.f, s2 = 0.f;
  const long per_v = (N * HW) / VN;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const long e = t * VN;
    const long n = e / HW, i = e % HW;
    const long base = (n * C + c) * HW + i;
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
    atomicAdd(&red[2 * c], b1);
    atomicAdd(&red[2 * c + 1], b2);
  }
}

template <typename T>
__global__ void bn_bwd_dx_kernel_v(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ yv,
                                   const float* __restrict__ red,
                                   const float* __restrict__ save_mean,
                                   const float* __restrict__ save_ivar,
                                   const float* __restrict__ weight,
                                   T* __restrict__ dx,
                                   float* __restrict__ dweight,
                                   float* __restrict__ dbias,
                                   long N, long C, long HW, int relu) {
  using VT = typename BnVec<T>::V;
  constexpr int VN = BnVec<T>::N;
  const long c = blockIdx.y;
  const long per_ch = N * HW;
  const float inv_n = 1.0f / (float)per_ch;
  const float mean = save_mean[c], ivar = save_ivar[c];
  const float w = weight ? weight[c] : 1.f;
  const float sum_dy = red[2 * c], sum_dyxh = red[2 * c + 1];
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    dweight[c] = sum_dyxh;
    dbias[c] = sum_dy;
  }
  const float k1 = w * ivar;
  const float m_dy = sum_dy * inv_n;
  const float m_dyxh = sum_dyxh * inv_n;
  const long per_v = per_ch / VN;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < per_v;
       t += stride) {
    const long e = t * VN;
    const long n = e / HW, i = e % HW;
    const long base = (n * C + c) * HW + i;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT xv = *reinterpret_cast<const VT*>(x + base);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float gg = (float)g.d[j];
      if (relu && (float)yv[base + j] <= 0.f) gg = 0.f;
      float xh = ((float)xv.d[j] - mean) * ivar;
      g.d[j] = (T)(k1 * (gg - m_dy - xh * m_dyxh));
    }
    *reinterpret_cast<VT*>(dx + base) = g;
  }
}
