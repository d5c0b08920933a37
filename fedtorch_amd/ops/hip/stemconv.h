// Custom NHWC stem convolution (gfx950) — 3x3 / stride 1 / pad 1, tiny
// input-channel count (CIFAR stems: 3 -> 16).
//
// Why: MIOpen's NHWC bf16 solvers fall back to naive_conv_* for the
// 3-channel stem (measured 275 us fwd / 9.5 ms wrw per call at b128 —
// profiles/r01_bench_notes.md), which is what forced the NCHW layout and
// its batched_transpose wrapper costs in the first place.  These two
// kernels keep the whole model channels_last.
//
// Layout: x [N, H, W, Ci] channels_last; w [Co, kh, kw, Ci] (the memory
// order of a channels_last conv weight), fp32 straight from the parameter
// arena (no per-step autocast cast of the weight); y [N, H, W, Co].
// No bwd-data kernel: the stem input is the data batch (requires_grad is
// false); the Python wrapper falls back to eager conv if it ever isn't.
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

#define STEM_MAX_CI 4
#define STEM_MAX_CO 32

// grid-stride over output pixels; each thread produces all Co channels of
// one pixel.  Weights staged in LDS as [kh][kw][ci][co] so the co loop is
// an LDS broadcast.
template <typename TX, typename TY>
__global__ void stem_fwd_k(const TX* __restrict__ x,
                           const float* __restrict__ w,
                           TY* __restrict__ y, int N, int H, int W, int Ci,
                           int Co) {
  __shared__ float wl[3][3][STEM_MAX_CI][STEM_MAX_CO];
  const int wn = Co * 9 * Ci;
  for (int t = threadIdx.x; t < wn; t += blockDim.x) {
    // w memory order: [co][kh][kw][ci]
    const int ci = t % Ci, kw = (t / Ci) % 3, kh = (t / (3 * Ci)) % 3,
              co = t / (9 * Ci);
    wl[kh][kw][ci][co] = w[t];
  }
  __syncthreads();
  const int HW = H * W;
  const long total = (long)N * HW;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long pix = blockIdx.x * (long)blockDim.x + threadIdx.x; pix < total;
       pix += stride) {
    const int hw = (int)(pix % HW);
    const int n = (int)(pix / HW);
    const int h = hw / W, ww_ = hw % W;
    float acc[STEM_MAX_CO];
#pragma unroll
    for (int co = 0; co < STEM_MAX_CO; ++co) acc[co] = 0.f;
    const TX* xn = x + (long)n * HW * Ci;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int hh = h + kh - 1;
      if (hh < 0 || hh >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int wwp = ww_ + kw - 1;
        if (wwp < 0 || wwp >= W) continue;
        const TX* xp = xn + ((long)hh * W + wwp) * Ci;
        for (int ci = 0; ci < Ci; ++ci) {
          const float xv = (float)xp[ci];
          for (int co = 0; co < Co; ++co)
            acc[co] = fmaf(xv, wl[kh][kw][ci][co], acc[co]);
        }
      }
    }
    TY* yp = y + pix * Co;
    for (int co = 0; co < Co; ++co) yp[co] = (TY)acc[co];
  }
}

// weight gradient: each thread owns one co and a strided set of pixels,
// accumulating the 9*Ci tap products in registers; per-block LDS reduction
// then one partial row per block; stem_wrw_final_k sums the partials.
template <typename TY, typename TX>
__global__ void stem_wrw_k(const TY* __restrict__ dy,
                           const TX* __restrict__ x,
                           float* __restrict__ part /*[B, Co*9*Ci]*/, int N,
                           int H, int W, int Ci, int Co) {
  __shared__ float lds[STEM_MAX_CO * 9 * STEM_MAX_CI];
  const int wn = Co * 9 * Ci;
  for (int t = threadIdx.x; t < wn; t += blockDim.x) lds[t] = 0.f;
  __syncthreads();
  const int co = threadIdx.x % Co;  // Co divides 256 for Co in {16, 32}
  const int HW = H * W;
  const long total = (long)N * HW;
  const long pix0 = (blockIdx.x * (long)blockDim.x + threadIdx.x) / Co;
  const long pstride = ((long)gridDim.x * blockDim.x) / Co;
  float acc[9 * STEM_MAX_CI];
#pragma unroll
  for (int j = 0; j < 9 * STEM_MAX_CI; ++j) acc[j] = 0.f;
  for (long pix = pix0; pix < total; pix += pstride) {
    const int hw = (int)(pix % HW);
    const int n = (int)(pix / HW);
    const int h = hw / W, ww_ = hw % W;
    const float g = (float)dy[pix * Co + co];
    const TX* xn = x + (long)n * HW * Ci;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int hh = h + kh - 1;
      if (hh < 0 || hh >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int wwp = ww_ + kw - 1;
        if (wwp < 0 || wwp >= W) continue;
        const TX* xp = xn + ((long)hh * W + wwp) * Ci;
        for (int ci = 0; ci < Ci; ++ci)
          acc[(kh * 3 + kw) * Ci + ci] =
              fmaf(g, (float)xp[ci], acc[(kh * 3 + kw) * Ci + ci]);
      }
    }
  }
  for (int j = 0; j < 9 * Ci; ++j)
    atomicAdd(&lds[co * 9 * Ci + j], acc[j]);
  __syncthreads();
  for (int t = threadIdx.x; t < wn; t += blockDim.x)
    part[(long)blockIdx.x * wn + t] = lds[t];
}

__global__ void stem_wrw_final_k(const float* __restrict__ part, int B,
                                 int wn, float* __restrict__ dw) {
  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= wn) return;
  float s = 0.f;
  for (int b = 0; b < B; ++b) s += part[(long)b * wn + t];
  dw[t] = s;
}
