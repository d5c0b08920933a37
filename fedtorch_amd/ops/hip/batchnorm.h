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

// sum the B (≤64) partials of channel c: lane-parallel loads + wave64
// xor-butterfly, so the ~300-cycle L2 latencies overlap instead of forming
// a B-deep dependent chain (a serial loop here measured ~10 us per call).
// Every lane of every wave ends up with the full sums.
__device__ __forceinline__ void bn_finalize(const float* __restrict__ part,
                                            long c, int B, float* s,
                                            float* q) {
  const int lane = threadIdx.x & (WAVE - 1);
  float a = 0.f, b = 0.f;
  const float* p = part + c * (long)B * 2;
  for (int i = lane; i < B; i += WAVE) {
    a += p[2 * i];
    b += p[2 * i + 1];
  }
#pragma unroll
  for (int off = WAVE / 2; off; off >>= 1) {
    a += __shfl_xor(a, off, WAVE);
    b += __shfl_xor(b, off, WAVE);
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

// ==========================================================================
// NHWC (channels_last) variants — the end-to-end NHWC path runs MIOpen's
// native-NHWC igemm conv solvers WITHOUT the batched_transpose /
// SubTensorOp wrapper kernels that NCHW tensors force around them
// (~45 % of ResNet-20 step kernel time at b256, profiles/r01_bench_notes.md).
//
// Layout: element (n, i=h*W+w, c) at (n*HW + i)*C + c — channels contiguous.
// Thread task split: each thread owns one channel GROUP (VN consecutive
// channels = one 16 B vector) of one row and keeps VN fp32 partials in
// registers; the cross-row reduction is a wave64 xor-butterfly over lanes
// with the same channel-group residue (offsets CGC..32, CGC = C/VN, a power
// of two ≤ 64), then one LDS pass across the 4 waves.  No atomics, no
// zero-fill: per-block partials land in part[B, C, 2] and the consumer
// kernel finalizes per channel (same scheme as the NCHW pack above).
// Host guarantees: C % VN == 0, CGC = C/VN is a power of two ≤ 64, C ≤ 256.
// ==========================================================================

// lgc = log2(C/VN); rows NI = N*HW.
template <typename T, typename VT, int VN>
__global__ void bnh_stats_k(const T* __restrict__ x, long NI, long C,
                            int lgc, float* __restrict__ part /*[B,C,2]*/) {
  const int cgc = 1 << lgc;
  const unsigned t = blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned cg = t & (cgc - 1);
  const unsigned rstride = (gridDim.x * blockDim.x) >> lgc;
  float s[VN], q[VN];
#pragma unroll
  for (int j = 0; j < VN; ++j) { s[j] = 0.f; q[j] = 0.f; }
  for (unsigned r = t >> lgc; r < NI; r += rstride) {
    VT v = *reinterpret_cast<const VT*>(x + (long)r * C + cg * VN);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float f = (float)v.d[j];
      s[j] += f;
      q[j] = fmaf(f, f, q[j]);
    }
  }
  // wave butterfly over same-residue lanes
  for (int off = WAVE / 2; off >= cgc; off >>= 1) {
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      s[j] += __shfl_xor(s[j], off, WAVE);
      q[j] += __shfl_xor(q[j], off, WAVE);
    }
  }
  __shared__ float lds[2][FT_BLOCK / WAVE][256];  // [s|q][wave][channel]
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  if (lane < cgc) {
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      lds[0][wid][lane * VN + j] = s[j];
      lds[1][wid][lane * VN + j] = q[j];
    }
  }
  __syncthreads();
  const int c = threadIdx.x;
  if (c < C) {
    float as = 0.f, aq = 0.f;
#pragma unroll
    for (int w = 0; w < FT_BLOCK / WAVE; ++w) {
      as += lds[0][w][c];
      aq += lds[1][w][c];
    }
    float* p = part + ((long)blockIdx.x * C + c) * 2;
    p[0] = as;
    p[1] = aq;
  }
}

// finalize per-channel scale/shift into LDS, then stream normalize
// [+res add][+ReLU].  Block 0 threads c<C write saved/running stats.
template <typename T, typename VT, int VN>
__global__ void bnh_norm_k(const T* __restrict__ x, T* __restrict__ y,
                           const float* __restrict__ part, int B,
                           const float* __restrict__ weight,
                           const float* __restrict__ bias,
                           float* __restrict__ save_mean,
                           float* __restrict__ save_ivar,
                           float* __restrict__ running_mean,
                           float* __restrict__ running_var,
                           const T* __restrict__ res,
                           unsigned char* __restrict__ mask, long NI, long C,
                           int lgc, float eps, float momentum, int relu) {
  // block-parallel finalize: thread (p, c) sums a slice of the B partials
  // so the L2 latencies overlap (a per-channel serial loop here is a
  // B-deep dependent chain, ~10 us); LDS tree collapses the P slices.
  __shared__ float lsc[256], lsh[256], fa[FT_BLOCK], fb[FT_BLOCK];
  const float inv_n = 1.0f / (float)NI;
  {
    const int P = blockDim.x / (int)C;
    const int c0 = threadIdx.x % (int)C, p0 = threadIdx.x / (int)C;
    float a = 0.f, bb = 0.f;
    for (int i = p0; i < B; i += P) {
      const float* p = part + ((long)i * C + c0) * 2;
      a += p[0];
      bb += p[1];
    }
    fa[threadIdx.x] = a;
    fb[threadIdx.x] = bb;
  }
  __syncthreads();
  const int c = threadIdx.x;
  if (c < C) {
    float s = 0.f, q = 0.f;
    const int P = blockDim.x / (int)C;
    for (int i = 0; i < P; ++i) {
      s += fa[i * C + c];
      q += fb[i * C + c];
    }
    const float mean = s * inv_n;
    const float var = fmaxf(q * inv_n - mean * mean, 0.f);
    const float ivar = rsqrtf(var + eps);
    const float w = weight ? weight[c] : 1.f;
    const float bb = bias ? bias[c] : 0.f;
    lsc[c] = w * ivar;
    lsh[c] = bb - mean * (w * ivar);
    if (blockIdx.x == 0) {
      save_mean[c] = mean;
      save_ivar[c] = ivar;
      if (running_mean) {
        float ub = var * (float)NI / (float)(NI > 1 ? NI - 1 : 1);
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * ub;
      }
    }
  }
  __syncthreads();
  const int cgc = 1 << lgc;
  const unsigned t = blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned cg = t & (cgc - 1);
  const unsigned rstride = (gridDim.x * blockDim.x) >> lgc;
  float sc[VN], sh[VN];
#pragma unroll
  for (int j = 0; j < VN; ++j) {
    sc[j] = lsc[cg * VN + j];
    sh[j] = lsh[cg * VN + j];
  }
  for (unsigned r = t >> lgc; r < NI; r += rstride) {
    const long base = (long)r * C + cg * VN;
    VT v = *reinterpret_cast<const VT*>(x + base);
    unsigned mbits = 0;
    if (res) {
      VT rv = *reinterpret_cast<const VT*>(res + base);
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        float f = fmaf((float)v.d[j], sc[j], sh[j]) + (float)rv.d[j];
        if (f > 0.f) mbits |= 1u << j;
        v.d[j] = (T)(relu ? fmaxf(f, 0.f) : f);
      }
    } else {
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        float f = fmaf((float)v.d[j], sc[j], sh[j]);
        if (f > 0.f) mbits |= 1u << j;
        v.d[j] = (T)(relu ? fmaxf(f, 0.f) : f);
      }
    }
    *reinterpret_cast<VT*>(y + base) = v;
    // ReLU bitmask (VN==8): the bwd kernels read 1 byte per vector instead
    // of re-reading the 16 B output vector for the mask
    if (mask) mask[((long)r * C >> 3) + cg] = (unsigned char)mbits;
  }
}

template <typename T, typename VT, int VN>
__global__ void bnh_bwd_stats_k(const T* __restrict__ dy,
                                const T* __restrict__ x,
                                const T* __restrict__ yv,
                                const unsigned char* __restrict__ mask,
                                const float* __restrict__ save_mean,
                                const float* __restrict__ save_ivar, long NI,
                                long C, int lgc,
                                float* __restrict__ part /*[B,C,2]*/,
                                int relu) {
  const int cgc = 1 << lgc;
  const unsigned t = blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned cg = t & (cgc - 1);
  const unsigned rstride = (gridDim.x * blockDim.x) >> lgc;
  float mean[VN], ivar[VN];
#pragma unroll
  for (int j = 0; j < VN; ++j) {
    mean[j] = save_mean[cg * VN + j];
    ivar[j] = save_ivar[cg * VN + j];
  }
  float s1[VN], s2[VN];
#pragma unroll
  for (int j = 0; j < VN; ++j) { s1[j] = 0.f; s2[j] = 0.f; }
  for (unsigned r = t >> lgc; r < NI; r += rstride) {
    const long base = (long)r * C + cg * VN;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT xv = *reinterpret_cast<const VT*>(x + base);
    if (relu) {
      if (mask) {
        const unsigned m = mask[((long)r * C >> 3) + cg];
#pragma unroll
        for (int j = 0; j < VN; ++j)
          if (!((m >> j) & 1u)) g.d[j] = (T)0.f;
      } else {
        VT yy = *reinterpret_cast<const VT*>(yv + base);
#pragma unroll
        for (int j = 0; j < VN; ++j)
          if ((float)yy.d[j] <= 0.f) g.d[j] = (T)0.f;
      }
    }
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float gg = (float)g.d[j];
      float xh = ((float)xv.d[j] - mean[j]) * ivar[j];
      s1[j] += gg;
      s2[j] = fmaf(gg, xh, s2[j]);
    }
  }
  for (int off = WAVE / 2; off >= cgc; off >>= 1) {
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      s1[j] += __shfl_xor(s1[j], off, WAVE);
      s2[j] += __shfl_xor(s2[j], off, WAVE);
    }
  }
  __shared__ float lds[2][FT_BLOCK / WAVE][256];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  if (lane < cgc) {
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      lds[0][wid][lane * VN + j] = s1[j];
      lds[1][wid][lane * VN + j] = s2[j];
    }
  }
  __syncthreads();
  const int c = threadIdx.x;
  if (c < C) {
    float a1 = 0.f, a2 = 0.f;
#pragma unroll
    for (int w = 0; w < FT_BLOCK / WAVE; ++w) {
      a1 += lds[0][w][c];
      a2 += lds[1][w][c];
    }
    float* p = part + ((long)blockIdx.x * C + c) * 2;
    p[0] = a1;
    p[1] = a2;
  }
}

template <typename T, typename VT, int VN>
__global__ void bnh_bwd_dx_k(const T* __restrict__ dy,
                             const T* __restrict__ x,
                             const T* __restrict__ yv,
                             const unsigned char* __restrict__ mask,
                             const float* __restrict__ part, int B,
                             const float* __restrict__ save_mean,
                             const float* __restrict__ save_ivar,
                             const float* __restrict__ weight,
                             T* __restrict__ dx, T* __restrict__ dres,
                             float* __restrict__ dweight,
                             float* __restrict__ dbias, long NI, long C,
                             int lgc, int relu) {
  __shared__ float lk1[256], lmdy[256], lmdyxh[256], lmean[256], livar[256];
  __shared__ float fa[FT_BLOCK], fb[FT_BLOCK];
  const float inv_n = 1.0f / (float)NI;
  {
    const int P = blockDim.x / (int)C;
    const int c0 = threadIdx.x % (int)C, p0 = threadIdx.x / (int)C;
    float a = 0.f, bb = 0.f;
    for (int i = p0; i < B; i += P) {
      const float* p = part + ((long)i * C + c0) * 2;
      a += p[0];
      bb += p[1];
    }
    fa[threadIdx.x] = a;
    fb[threadIdx.x] = bb;
  }
  __syncthreads();
  const int c = threadIdx.x;
  if (c < C) {
    float s1 = 0.f, s2 = 0.f;
    const int P = blockDim.x / (int)C;
    for (int i = 0; i < P; ++i) {
      s1 += fa[i * C + c];
      s2 += fb[i * C + c];
    }
    const float w = weight ? weight[c] : 1.f;
    const float iv = save_ivar[c];
    lk1[c] = w * iv;
    lmdy[c] = s1 * inv_n;
    lmdyxh[c] = s2 * inv_n;
    lmean[c] = save_mean[c];
    livar[c] = iv;
    if (blockIdx.x == 0) {
      dweight[c] = s2;
      dbias[c] = s1;
    }
  }
  __syncthreads();
  const int cgc = 1 << lgc;
  const unsigned t = blockIdx.x * blockDim.x + threadIdx.x;
  const unsigned cg = t & (cgc - 1);
  const unsigned rstride = (gridDim.x * blockDim.x) >> lgc;
  float k1[VN], mdy[VN], mdyxh[VN], mean[VN], ivar[VN];
#pragma unroll
  for (int j = 0; j < VN; ++j) {
    const int cc = cg * VN + j;
    k1[j] = lk1[cc];
    mdy[j] = lmdy[cc];
    mdyxh[j] = lmdyxh[cc];
    mean[j] = lmean[cc];
    ivar[j] = livar[cc];
  }
  for (unsigned r = t >> lgc; r < NI; r += rstride) {
    const long base = (long)r * C + cg * VN;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT xv = *reinterpret_cast<const VT*>(x + base);
    if (relu) {
      if (mask) {
        const unsigned m = mask[((long)r * C >> 3) + cg];
#pragma unroll
        for (int j = 0; j < VN; ++j)
          if (!((m >> j) & 1u)) g.d[j] = (T)0.f;
      } else {
        VT yy = *reinterpret_cast<const VT*>(yv + base);
#pragma unroll
        for (int j = 0; j < VN; ++j)
          if ((float)yy.d[j] <= 0.f) g.d[j] = (T)0.f;
      }
    }
    if (dres) *reinterpret_cast<VT*>(dres + base) = g;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float gg = (float)g.d[j];
      float xh = ((float)xv.d[j] - mean[j]) * ivar[j];
      g.d[j] = (T)(k1[j] * (gg - mdy[j] - xh * mdyxh[j]));
    }
    *reinterpret_cast<VT*>(dx + base) = g;
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


// finalize bnh_bwd_stats partials into the per-channel AFFINE transform
// of the deferred BN backward (consumed by conv3x3_dgrad_k<TRF=true>):
//   dx_bn = A[c]*g_masked + B[c] + D[c]*x
// with A = gamma*ivar, B = A*(-mdy + mean*ivar*mdyxh), D = -A*ivar*mdyxh
// (algebraic rearrangement of bnh_bwd_dx_k's formula).  Also emits
// dgamma = sum(dy*xhat), dbeta = sum(dy).
__global__ void __launch_bounds__(FT_BLOCK) bnh_bwd_coef_k(
    const float* __restrict__ part, int B, long NI,
    const float* __restrict__ save_mean,
    const float* __restrict__ save_ivar,
    const float* __restrict__ weight, long C,
    float* __restrict__ coefs /* [3, C] */, float* __restrict__ dweight,
    float* __restrict__ dbias) {
  __shared__ float fa[FT_BLOCK], fb[FT_BLOCK];
  const float inv_n = 1.0f / (float)NI;
  const int P = blockDim.x / (int)C;
  const int c0 = threadIdx.x % (int)C, p0 = threadIdx.x / (int)C;
  float a = 0.f, bb = 0.f;
  for (int i = p0; i < B; i += P) {
    const float* p = part + ((long)i * C + c0) * 2;
    a += p[0];
    bb += p[1];
  }
  fa[threadIdx.x] = a;
  fb[threadIdx.x] = bb;
  __syncthreads();
  const int c = threadIdx.x;
  if (c < C) {
    float s1 = 0.f, s2 = 0.f;
    for (int i = 0; i < P; ++i) {
      s1 += fa[i * C + c];
      s2 += fb[i * C + c];
    }
    const float w = weight ? weight[c] : 1.f;
    const float iv = save_ivar[c];
    const float A = w * iv;
    const float mdy = s1 * inv_n, mdyxh = s2 * inv_n;
    coefs[c] = A;
    coefs[C + c] = A * (fmaf(save_mean[c] * iv, mdyxh, -mdy));
    coefs[2 * C + c] = -A * iv * mdyxh;
    dweight[c] = s2;
    dbias[c] = s1;
  }
}

// dres for the deferred BN backward: the ReLU-masked upstream gradient
// (the fused residual branch needs it immediately; the conv transform
// recomputes the same mask internally for its own staging)
template <typename T, typename VT, int VN>
__global__ void __launch_bounds__(FT_BLOCK) bnh_mask_dres_k(
    const T* __restrict__ dy, const T* __restrict__ yv,
    T* __restrict__ dres, long total_v) {
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned t = blockIdx.x * blockDim.x + threadIdx.x; t < total_v;
       t += stride) {
    const long base = (long)t * VN;
    VT g = *reinterpret_cast<const VT*>(dy + base);
    VT yy = *reinterpret_cast<const VT*>(yv + base);
#pragma unroll
    for (int j = 0; j < VN; ++j)
      if ((float)yy.d[j] <= 0.f) g.d[j] = (T)0.f;
    *reinterpret_cast<VT*>(dres + base) = g;
  }
}
