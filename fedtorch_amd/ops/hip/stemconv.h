// Custom NHWC stem convolution (gfx950) — 3x3 / stride 1 / pad 1, tiny
// input-channel count (CIFAR stems: 3 -> 16/32).
//
// Why: MIOpen's NHWC bf16 solvers fall back to naive_conv_* for the
// 3-channel stem (measured 275 us fwd / 9.5 ms wrw per call at b128 —
// profiles/r01_bench_notes.md), which is what forced the NCHW layout and
// its batched_transpose wrapper costs in the first place.  These kernels
// keep the whole model channels_last.
//
// CI/CO are COMPILE-TIME template parameters: the accumulator arrays are
// indexed by the (now fully unrolled) tap/channel loops — with runtime
// bounds the compiler demotes them to scratch memory and the kernel runs
// ~70x slower (measured 1.28 ms/call for the first runtime-CI version).
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

// grid-stride over output pixels; each thread produces all CO channels of
// one pixel.  Weights staged in LDS as [kh][kw][ci][co] so the co loop is
// an LDS broadcast.
template <typename TX, typename TY, int CI, int CO>
__global__ void __launch_bounds__(FT_BLOCK) stem_fwd_k(const TX* __restrict__ x,
                           const float* __restrict__ w,
                           TY* __restrict__ y, int N, int H, int W) {
  __shared__ float wl[3][3][CI][CO];
  for (int t = threadIdx.x; t < CO * 9 * CI; t += blockDim.x) {
    // w memory order: [co][kh][kw][ci]
    const int ci = t % CI, kw = (t / CI) % 3, kh = (t / (3 * CI)) % 3,
              co = t / (9 * CI);
    wl[kh][kw][ci][co] = w[t];
  }
  __syncthreads();
  const unsigned HW = H * W;
  const unsigned total = (unsigned)N * HW;
  const unsigned stride = gridDim.x * blockDim.x;
  for (unsigned pix = blockIdx.x * blockDim.x + threadIdx.x; pix < total;
       pix += stride) {
    const unsigned hw = pix % HW;
    const unsigned n = pix / HW;
    const int h = hw / (unsigned)W, ww_ = hw % (unsigned)W;
    float acc[CO];
#pragma unroll
    for (int co = 0; co < CO; ++co) acc[co] = 0.f;
    const TX* xn = x + (long)n * HW * CI;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int hh = h + kh - 1;
      if (hh < 0 || hh >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int wwp = ww_ + kw - 1;
        if (wwp < 0 || wwp >= W) continue;
        const TX* xp = xn + ((long)hh * W + wwp) * CI;
#pragma unroll
        for (int ci = 0; ci < CI; ++ci) {
          const float xv = (float)xp[ci];
#pragma unroll
          for (int co = 0; co < CO; ++co)
            acc[co] = fmaf(xv, wl[kh][kw][ci][co], acc[co]);
        }
      }
    }
    TY* yp = y + (long)pix * CO;
#pragma unroll
    for (int co = 0; co < CO; ++co) yp[co] = (TY)acc[co];
  }
}

// weight gradient: each thread owns one co and a strided set of pixels,
// accumulating the 9*CI tap products in registers (fully unrolled); LDS
// reduction per block, then one partial row per block; stem_wrw_final_k
// sums the partials.
template <typename TY, typename TX, int CI, int CO>
__global__ void __launch_bounds__(FT_BLOCK) stem_wrw_k(const TY* __restrict__ dy,
                           const TX* __restrict__ x,
                           float* __restrict__ part /*[B, CO*9*CI]*/, int N,
                           int H, int W) {
  __shared__ float lds[CO * 9 * CI];
  for (int t = threadIdx.x; t < CO * 9 * CI; t += blockDim.x) lds[t] = 0.f;
  __syncthreads();
  const int co = threadIdx.x % CO;  // CO divides 256
  const unsigned HW = H * W;
  const unsigned total = (unsigned)N * HW;
  const unsigned pix0 = (blockIdx.x * blockDim.x + threadIdx.x) / CO;
  const unsigned pstride = (gridDim.x * blockDim.x) / CO;
  float acc[9 * CI];
#pragma unroll
  for (int j = 0; j < 9 * CI; ++j) acc[j] = 0.f;
  for (unsigned pix = pix0; pix < total; pix += pstride) {
    const unsigned hw = pix % HW;
    const unsigned n = pix / HW;
    const int h = hw / (unsigned)W, ww_ = hw % (unsigned)W;
    const float g = (float)dy[(long)pix * CO + co];
    const TX* xn = x + (long)n * HW * CI;
#pragma unroll
    for (int kh = 0; kh < 3; ++kh) {
      const int hh = h + kh - 1;
      if (hh < 0 || hh >= H) continue;
#pragma unroll
      for (int kw = 0; kw < 3; ++kw) {
        const int wwp = ww_ + kw - 1;
        if (wwp < 0 || wwp >= W) continue;
        const TX* xp = xn + ((long)hh * W + wwp) * CI;
#pragma unroll
        for (int ci = 0; ci < CI; ++ci)
          acc[(kh * 3 + kw) * CI + ci] =
              fmaf(g, (float)xp[ci], acc[(kh * 3 + kw) * CI + ci]);
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 9 * CI; ++j)
    atomicAdd(&lds[co * 9 * CI + j], acc[j]);
  __syncthreads();
  for (int t = threadIdx.x; t < CO * 9 * CI; t += blockDim.x)
    part[(long)blockIdx.x * (CO * 9 * CI) + t] = lds[t];
}

// one wave per output element: lane-parallel loads over the B partials +
// xor-butterfly (a serial B-loop is a latency chain at B=1024).
__global__ void __launch_bounds__(FT_BLOCK) stem_wrw_final_k(
    const float* __restrict__ part, int B, int wn,
    float* __restrict__ dw) {
  const int wave = threadIdx.x / WAVE, lane = threadIdx.x & (WAVE - 1);
  const int t = blockIdx.x * (FT_BLOCK / WAVE) + wave;
  if (t >= wn) return;
  float s = 0.f;
  for (int b = lane; b < B; b += WAVE) s += part[(long)b * wn + t];
#pragma unroll
  for (int off = WAVE / 2; off; off >>= 1) s += __shfl_xor(s, off, WAVE);
  if (lane == 0) dw[t] = s;
}
