// Common device helpers for the fedtorch_amd CDNA4 kernel pack (gfx950).
//
// All arena ops are memory-bound streaming kernels; per the CDNA4 rules
// (cdna_hip_programming.md Appendix B / Guideline 11+13):
//   * block = 256 threads (4 wave64), grid capped at 2048 blocks +
//     grid-stride loop;
//   * 16 B per lane (float4) vectorized access — arenas are 64-element
//     aligned so float4 is always safe;
//   * wave64 shuffle reductions (NOT 32-wide warp idioms).
#pragma once
#include <hip/hip_runtime.h>

#define FT_BLOCK 256
#define FT_MAX_BLOCKS 2048
#define WAVE 64

static inline int ft_grid(long n_items) {
  long b = (n_items + FT_BLOCK - 1) / FT_BLOCK;
  if (b > FT_MAX_BLOCKS) b = FT_MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}

// wave64 reductions ---------------------------------------------------------
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_min(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fminf(v, __shfl_down(v, off, WAVE));
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// block-level reduce: each of the 4 waves reduces, lane0 writes LDS, wave 0
// finishes. `op`: 0 sum, 1 min, 2 max.
template <int OP>
__device__ __forceinline__ float block_reduce(float v) {
  __shared__ float lds[FT_BLOCK / WAVE];
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  if (OP == 0) v = wave_sum(v);
  if (OP == 1) v = wave_min(v);
  if (OP == 2) v = wave_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = (lane < FT_BLOCK / WAVE) ? lds[lane]
        : (OP == 1 ? 3.4e38f : (OP == 2 ? -3.4e38f : 0.f));
    if (OP == 0) v = wave_sum(v);
    if (OP == 1) v = wave_min(v);
    if (OP == 2) v = wave_max(v);
  }
  return v;  // valid in wave 0 lane 0
}

// ordered-uint mapping of |x| for radix select: IEEE bits of a non-negative
// float are monotonic as unsigned.
__device__ __forceinline__ unsigned int abs_key(float x) {
  union { float f; unsigned int u; } c;
  c.f = fabsf(x);
  return c.u;
}
