// MFMA 3x3/s1/p1 NHWC bf16 conv weight-gradient, v2 (gfx950).
//
// v1 (convwrw.h) stages 32 positions per wave-private slab and reads
// every MFMA fragment element with a SCALAR ds_read (k = the position
// axis, memory is channel-contiguous): it measured 77/28/33 us per call
// vs MIOpen's 33/25/22 (incl. its SubTensorOp/cast wrappers).  v2 fixes
// the layout, not the schedule:
//
//   * staging TRANSPOSES both operands into LDS — s_dyT[co][pos] and
//     s_xT[ci][pos-with-halo] — so every fragment is ONE contiguous
//     ds_read_b128 (the transpose costs scalar ds_writes, which are
//     cheap and cooperative across the 4 waves);
//   * the 3 tap COLUMN shifts never touch LDS again: each (dh) row is
//     read once as a 16-position aligned window (2x b128) and the
//     dw = 0/1/2 fragments come out of registers (dw=1 via v_alignbit
//     by 16 bits, dw=2 is a free dword shuffle);
//   * one supertile (128 positions; 64 = one image for W=8) is staged
//     cooperatively per workgroup per iteration, single-buffered
//     (stage -> barrier -> MFMAs -> barrier), overlap from 4+ resident
//     workgroups per CU — the structure conv3x3_bn_fwd measured at
//     5-7 us/call.
//
// Work split per shape (Q = (CO/16)*(CI/16) 16x16 output quadrants):
//   C16 (Q=1):  wave w owns k-chunk w of the supertile (4 chunks x 9 taps).
//   C32 (Q=4):  wave w owns quadrant w, loops all 4 chunks.
//   C64 (Q=16): blocks come in groups of 4 (qg = blockIdx%4); wave w owns
//               quadrant qg*4+w, loops the image's 2 chunks. dy/x are
//               re-read 4x (~2 us of extra HBM traffic at these sizes).
//
// Partials: one row per (block, wave) in QUADRANT-LOCAL layout
// part[q*RQ + r][9*16*16]; conv3x3_wrw2_final_k sums each quadrant's rows
// and scatters into dw[co][kh][kw][ci] (channels_last memory order).
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

// bf16x8 / f32x4 from convwrw.h (same TU)

template <int CO, int CI, int W>
struct Wrw2Cfg {
  static constexpr int SPOS = (W == 8) ? 64 : 128;   // positions/supertile
  static constexpr int SR = SPOS / W;                // image rows
  static constexpr int CH = SPOS / 32;               // 32-pos k-chunks
  static constexpr int XR = SR + 2;
  static constexpr int XCP = (W + 2 + 7) & ~7;       // padded row stride
  // plane strides: multiples of 8 elements (aligned b128) with
  // (stride/2) % 8 == 4 so 16 consecutive planes hit distinct banks
  static constexpr int SXR = XR * XCP;
  static constexpr int SX = (SXR % 16 == 8) ? SXR : ((SXR & ~15) + 8 +
                            ((SXR % 16 > 8) ? 16 : 0));
  static constexpr int SD = (SPOS % 16 == 8) ? SPOS : SPOS + 8;
  static constexpr int Q = (CO / 16) * (CI / 16);
  static constexpr int QG = (Q + 3) / 4;             // block-group size
};

template <int CO, int CI, int W>
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_wrw2_k(
    const __hip_bfloat16* __restrict__ dy, const __hip_bfloat16* __restrict__ x,
    float* __restrict__ part, int N, int H, int RQ) {
  using C = Wrw2Cfg<CO, CI, W>;
  __shared__ __hip_bfloat16 s_dyT[CO * C::SD];
  __shared__ __hip_bfloat16 s_xT[CI * C::SX];
  __shared__ float s_merge[C::Q == 1 ? 9 * 256 : 1];  // C16 wave merge

  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int fm = lane & 15, kg = lane >> 4;

  const int qg = (C::QG > 1) ? (blockIdx.x % C::QG) : 0;
  const int bstream = blockIdx.x / C::QG;
  const int nstreams = gridDim.x / C::QG;

  // quadrant of this wave (C16: all waves quadrant 0)
  const int q = (C::Q == 1) ? 0 : qg * 4 + wave;
  const int qco = (C::Q == 1) ? 0 : (q / (CI / 16)) * 16;
  const int qci = (C::Q == 1) ? 0 : (q % (CI / 16)) * 16;

  // zero the x halo/padding once: body writes cover cols [1..W] of rows
  // [0..XR); everything else must read as 0 for every supertile
  for (int e = tid; e < CI * C::SX / 8; e += FT_BLOCK)
    *reinterpret_cast<uint4*>(&s_xT[(long)e * 8]) = uint4{0, 0, 0, 0};

  f32x4 acc[9];
#pragma unroll
  for (int t = 0; t < 9; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const int tiles_per_img = H / C::SR;
  const long tiles = (long)N * tiles_per_img;
  __syncthreads();

  for (long tg = bstream; tg < tiles; tg += nstreams) {
    const int n = (int)(tg / tiles_per_img);
    const int h0 = (int)(tg % tiles_per_img) * C::SR;

    // ---- stage dy transposed: [pos][co] -> s_dyT[co][pos] --------------
    {
      const __hip_bfloat16* dyp = dy + (((long)n * H + h0) * W) * CO;
      constexpr int CHK = C::SPOS * CO / 8;
      for (int e = tid; e < CHK; e += FT_BLOCK) {
        const int pos = e * 8 / CO, c8 = (e * 8) % CO;
        uint4 v = *reinterpret_cast<const uint4*>(dyp + (long)pos * CO + c8);
        const __hip_bfloat16* vp =
            reinterpret_cast<const __hip_bfloat16*>(&v);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          s_dyT[(long)(c8 + j) * C::SD + pos] = vp[j];
      }
    }
    // ---- stage x transposed with halo: -> s_xT[ci][row*XCP + col+1] ----
    {
      constexpr int CHK = C::XR * W * CI / 8;
      for (int e = tid; e < CHK; e += FT_BLOCK) {
        const int rowe = e * 8 / (W * CI);
        const int rem = (e * 8) % (W * CI);
        const int col = rem / CI, c8 = rem % CI;
        const int hh = h0 - 1 + rowe;
        uint4 v = {0, 0, 0, 0};
        if (hh >= 0 && hh < H)
          v = *reinterpret_cast<const uint4*>(
              x + (((long)n * H + hh) * W + col) * CI + c8);
        const __hip_bfloat16* vp =
            reinterpret_cast<const __hip_bfloat16*>(&v);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          s_xT[(long)(c8 + j) * C::SX + rowe * C::XCP + col + 1] = vp[j];
      }
    }
    __syncthreads();

    // ---- MFMAs ----------------------------------------------------------
#pragma unroll
    for (int ch = 0; ch < ((C::Q == 1) ? 1 : C::CH); ++ch) {
      const int chunk = (C::Q == 1) ? wave : ch;
      const int pb = chunk * 32 + kg * 8;       // this lane's 8 positions
      const int r0 = pb / W, w0 = pb % W;
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &s_dyT[(long)(qco + fm) * C::SD + pb]);
#pragma unroll
      for (int dh = 0; dh < 3; ++dh) {
        // 16-position aligned window of this (ci, row): the dw=0/1/2
        // fragments come out of these 8 dwords without further LDS reads
        const __hip_bfloat16* bp =
            &s_xT[(long)(qci + fm) * C::SX + (r0 + dh) * C::XCP + w0];
        typedef __attribute__((ext_vector_type(4))) unsigned u32x4;
        const u32x4 lo = *reinterpret_cast<const u32x4*>(bp);
        const u32x4 hi = *reinterpret_cast<const u32x4*>(bp + 8);
        u32x4 f0 = lo;
        u32x4 f1, f2 = {lo.y, lo.z, lo.w, hi.x};
        f1.x = __builtin_amdgcn_alignbit(lo.y, lo.x, 16);
        f1.y = __builtin_amdgcn_alignbit(lo.z, lo.y, 16);
        f1.z = __builtin_amdgcn_alignbit(lo.w, lo.z, 16);
        f1.w = __builtin_amdgcn_alignbit(hi.x, lo.w, 16);
        acc[dh * 3 + 0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, __builtin_bit_cast(bf16x8, f0), acc[dh * 3 + 0], 0, 0, 0);
        acc[dh * 3 + 1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, __builtin_bit_cast(bf16x8, f1), acc[dh * 3 + 1], 0, 0, 0);
        acc[dh * 3 + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, __builtin_bit_cast(bf16x8, f2), acc[dh * 3 + 2], 0, 0, 0);
      }
    }
    __syncthreads();  // slab consumed; next supertile may overwrite
  }

  // ---- partial row: quadrant-local [9][16][16] ---------------------------
  // D of mfma(a=dy, b=x): col = lane&15 maps the B operand's n index
  // (ci), rows map A's m (co): row = kg*4 + reg.
  if (C::Q == 1) {
    // the 4 waves hold CHUNK partials of the SAME quadrant: merge them
    // through LDS so the block emits ONE row (4x less partial traffic —
    // the reduce kernel was reading 37.7 MB/call for C16)
    float* buf = s_merge;
    for (int w = 1; w < 4; ++w) {
      __syncthreads();
      if (wave == w) {
#pragma unroll
        for (int t = 0; t < 9; ++t)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            buf[t * 256 + (kg * 4 + r) * 16 + fm] = acc[t][r];
      }
      __syncthreads();
      if (wave == 0) {
#pragma unroll
        for (int t = 0; t < 9; ++t)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            acc[t][r] += buf[t * 256 + (kg * 4 + r) * 16 + fm];
      }
    }
    if (wave == 0) {
      float* pr = part + (long)blockIdx.x * (9 * 16 * 16);
#pragma unroll
      for (int t = 0; t < 9; ++t)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pr[t * 256 + (kg * 4 + r) * 16 + fm] = acc[t][r];
    }
    return;
  }
  const int rr = (C::Q == 4) ? q * RQ + (int)blockIdx.x
                             : q * RQ + (int)bstream;
  float* pr = part + (long)rr * (9 * 16 * 16);
#pragma unroll
  for (int t = 0; t < 9; ++t) {
    const int row = kg * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r)
      pr[t * 256 + (row + r) * 16 + fm] = acc[t][r];
  }
}

// part[q*RQ + r][9*16*16] -> [stripe][Q*9*256] (stripe-parallel, no
// atomics — the atomic+acc-zero version cost a fill kernel + atomic
// round per call); conv3x3_wrw2_cast_k folds the stripes and casts.
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_wrw2_reduce_k(
    const float* __restrict__ part, int RQ, int stripes,
    float* __restrict__ out /* [stripes][Q*9*256] */, int Q) {
  __shared__ float lds[4][64];
  const int c = threadIdx.x & 63, rs = threadIdx.x >> 6;
  const int blocks_per_q = (9 * 256) / 64 * stripes;
  const int q = blockIdx.x / blocks_per_q;
  const int rem = blockIdx.x % blocks_per_q;
  const int stripe = rem % stripes;
  const int lc = (rem / stripes) * 64 + c;
  const int r0 = (int)((long)stripe * RQ / stripes);
  const int r1 = (int)((long)(stripe + 1) * RQ / stripes);
  float s = 0.f;
  for (int r = r0 + rs; r < r1; r += 4)
    s += part[((long)q * RQ + r) * (9 * 256) + lc];
  lds[rs][c] = s;
  __syncthreads();
  if (rs == 0) {
    s = lds[0][c] + lds[1][c] + lds[2][c] + lds[3][c];
    out[(long)stripe * (Q * 9 * 256) + (long)q * (9 * 256) + lc] = s;
  }
}

// fold stripes + cast: [stripes][Q*9*256] fp32 -> dw bf16 [co][kh][kw][ci]
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_wrw2_cast_k(
    const float* __restrict__ out, int stripes,
    __hip_bfloat16* __restrict__ dw, int CO, int CI) {
  const int total = 9 * CO * CI;
  const int QC = ((CO / 16) * (CI / 16)) * 9 * 256;
  for (int e = blockIdx.x * FT_BLOCK + threadIdx.x; e < total;
       e += gridDim.x * FT_BLOCK) {
    const int q = e / (9 * 256), lc = e % (9 * 256);
    float s = 0.f;
    for (int st = 0; st < stripes; ++st)
      s += out[(long)st * QC + (long)q * (9 * 256) + lc];
    const int tap = lc / 256, mrow = (lc % 256) / 16, ncol = lc % 16;
    const int co = (q / (CI / 16)) * 16 + mrow;
    const int ci = (q % (CI / 16)) * 16 + ncol;
    dw[((long)co * 9 + tap) * CI + ci] = __float2bfloat16(s);
  }
}
