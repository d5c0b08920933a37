// MFMA-based 3x3/s1/p1 NHWC bf16 convolution weight-gradient (gfx950).
//
// MIOpen's wrw solvers for these shapes (igemm / CK batched-GEMM) carry
// SubTensorOp workspace-zero + fp32->bf16 cast wrapper kernels (~280 us of
// a 1.65 ms ResNet-20/b256 step); this kernel computes all 9 taps of
// dW[Co, 3, 3, Ci] as MFMA 16x16x32 tile-GEMMs over the flattened
// position axis with row-wise LDS staging (halo columns zero-padded, so
// tap shifts never wrap across rows/images) and needs no workspace pass.
//
//   dW_t[co, ci] = sum_{n,h,w} dY[n,h,w,co] * X[n, h+dh, w+dw, ci]
//
// Fragment layout (verified on hardware by mfma_probe_gemm / the
// test_mfma_tile_gemm GPU test):
//   A (16x32, M=co):  m = lane&15, k = (lane>>4)*8 + i   (8 bf16/lane)
//   B (32x16, N=ci):  n = lane&15, k = (lane>>4)*8 + i
//   D (16x16):        col = lane&15, row = (lane>>4)*4 + reg
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// ==========================================================================
// wrw kernel
//
// K axis = flattened positions, processed in 32-position tiles = R = 32/W
// whole image rows (so a tap shift of ±1 row / ±1 col indexes a staged
// (R+2) x (W+2) x CI LDS slab whose halo ring is zero — no wrap-around,
// no masks).  Per 32-tile each wave issues 9 MFMA (one per tap) into 9
// resident f32x4 accumulators.
//
// Work split (template Q = (CO/16)*(CI/16)):
//   Q == 1  : each of the 4 waves owns its OWN tile stream (own LDS slab,
//             no barriers), partial rows indexed per wave.
//   Q == 4  : the 4 waves share one tile (one staging, __syncthreads) and
//             each owns one 16x16 quadrant of the 32x32 output.
//   Q == 16 : like Q == 4 with 4 quadrant passes (K re-read; the whole
//             dY/X working set is LLC-resident at these sizes).
// Partials land row-major in part[rows][9*CO*CI] fp32 (writes coalesce:
// 16 consecutive-ci lanes store 64 B bursts); wrw_final_k sums the rows
// of 64-column tiles (coalesced reads) and emits bf16 dW in channels_last
// memory order [co][kh][kw][ci].
// ==========================================================================
template <int CO, int CI, int W>
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_wrw_k(
    const __hip_bfloat16* __restrict__ dy, const __hip_bfloat16* __restrict__ x,
    float* __restrict__ part, int N, int H) {
  constexpr int R = 32 / W;            // image rows per 32-position K-tile
  constexpr int Q = (CO / 16) * (CI / 16);
  constexpr int QC = CI / 16;
  constexpr int GW = (Q == 1) ? 1 : 4;  // waves per tile-group
  constexpr int GROUPS = 4 / GW;        // tile-groups per block
  constexpr int PASSES = (Q + 3) / 4;   // quadrant passes per tile stream
  constexpr int XROWS = R + 2, XCOLS = W + 2;

  // double-buffered slabs: tile t+1's global loads are issued (into
  // registers) while tile t's fragments/MFMAs run, so the waves are not
  // parked on the load->ds_write->ds_read chain every tile (PMC: the
  // single-buffer version spent 68 % of wave cycles in SQ_WAIT_ANY).
  __shared__ __hip_bfloat16 s_dy[GROUPS][2][32 * CO];
  __shared__ __hip_bfloat16 s_x[GROUPS][2][XROWS * XCOLS * CI];

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int grp = (GW == 1) ? wave : 0;
  const int gtid = (GW == 1) ? lane : threadIdx.x;  // tid within the group
  constexpr int GTH = GW * WAVE;

  const int rows_per_img = H / R;
  const long tiles = (long)N * rows_per_img;
  const long group_id0 = (long)blockIdx.x * GROUPS + grp;
  const long gstride = (long)gridDim.x * GROUPS;

  // lane -> fragment coordinates (A row / B col = lane&15; 8 consecutive
  // positions kbase..kbase+7 live in ONE image row: w0..w0+7 of row r0)
  const int fm = lane & 15;
  const int kbase = (lane >> 4) * 8;
  const int r0 = kbase / W, w0 = kbase % W;

  constexpr int DYC = 32 * CO / 8;   // 16 B chunks of the dY tile
  constexpr int BCH = W * CI / 8;    // 16 B chunks per X row body
  constexpr int HCH = CI / 8;        // 16 B chunks per halo column
  constexpr int DYI = (DYC + GTH - 1) / GTH;
  constexpr int XBI = (XROWS * BCH + GTH - 1) / GTH;

  uint4 dyr[DYI], xr[XBI];

  auto load_tile = [&](long tg) {
    const int n = (int)(tg / rows_per_img);
    const int h0 = (int)(tg % rows_per_img) * R;
    const __hip_bfloat16* dyp = dy + (((long)n * H + h0) * W) * CO;
#pragma unroll
    for (int j = 0; j < DYI; ++j) {
      const int e = gtid + j * GTH;
      if (e < DYC)
        dyr[j] = reinterpret_cast<const uint4*>(dyp)[e];
    }
#pragma unroll
    for (int j = 0; j < XBI; ++j) {
      const int e = gtid + j * GTH;
      uint4 v = {0, 0, 0, 0};
      if (e < XROWS * BCH) {
        const int row = e / BCH, c = e % BCH;
        const int hh = h0 - 1 + row;
        if (hh >= 0 && hh < H)
          v = reinterpret_cast<const uint4*>(
              x + (((long)n * H + hh) * W) * CI)[c];
      }
      xr[j] = v;
    }
  };

  auto write_slab = [&](int buf) {
#pragma unroll
    for (int j = 0; j < DYI; ++j) {
      const int e = gtid + j * GTH;
      if (e < DYC)
        reinterpret_cast<uint4*>(&s_dy[grp][buf][0])[e] = dyr[j];
    }
    for (int e = gtid; e < XROWS * 2 * HCH; e += GTH) {
      const int row = e / (2 * HCH), half = (e / HCH) & 1, c = e % HCH;
      const int col = half ? (W + 1) : 0;
      reinterpret_cast<uint4*>(
          &s_x[grp][buf][(row * XCOLS + col) * CI])[c] = uint4{0, 0, 0, 0};
    }
#pragma unroll
    for (int j = 0; j < XBI; ++j) {
      const int e = gtid + j * GTH;
      if (e < XROWS * BCH) {
        const int row = e / BCH, c = e % BCH;
        reinterpret_cast<uint4*>(
            &s_x[grp][buf][(row * XCOLS + 1) * CI])[c] = xr[j];
      }
    }
  };

  for (int pass = 0; pass < PASSES; ++pass) {
    const int q = pass * 4 + ((GW == 1) ? 0 : wave);
    const int qco = (Q == 1) ? 0 : (q / QC) * 16;
    const int qci = (Q == 1) ? 0 : (q % QC) * 16;
    f32x4 acc[9];
#pragma unroll
    for (int t = 0; t < 9; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

    int cur = 0;
    if (group_id0 < tiles) {
      load_tile(group_id0);
      write_slab(cur);
    }
    if (GW > 1) __syncthreads();
    for (long tg = group_id0; tg < tiles; tg += gstride) {
      const long nxt = tg + gstride;
      if (nxt < tiles) load_tile(nxt);  // loads in flight during compute
      // ---- fragments + 9 MFMA from slab `cur` -----------------------------
      bf16x8 a;
#pragma unroll
      for (int i = 0; i < 8; ++i)
        a[i] = *reinterpret_cast<const __bf16*>(
            &s_dy[grp][cur][(kbase + i) * CO + qco + fm]);
#pragma unroll
      for (int dh = 0; dh < 3; ++dh) {
#pragma unroll
        for (int dw = 0; dw < 3; ++dw) {
          bf16x8 b;
#pragma unroll
          for (int i = 0; i < 8; ++i)
            b[i] = *reinterpret_cast<const __bf16*>(
                &s_x[grp][cur][((r0 + dh) * XCOLS + (w0 + dw + i)) * CI
                               + qci + fm]);
          acc[dh * 3 + dw] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[dh * 3 + dw], 0, 0, 0);
        }
      }
      if (nxt < tiles) {
        write_slab(cur ^ 1);
        cur ^= 1;
      }
      if (GW > 1) __syncthreads();
    }
    // ---- write partials (row-major [rows][9*CO*CI]: coalesced) -----------
    const long out_row = (long)blockIdx.x * GROUPS + grp;
    float* pr = part + out_row * (long)(9 * CO * CI);
#pragma unroll
    for (int t = 0; t < 9; ++t) {
      const int row = (lane >> 4) * 4;  // + reg
      const int col = lane & 15;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pr[((long)t * CO + qco + row + r) * CI + qci + col] = acc[t][r];
    }
  }
}

// reduce part[rows][9*CO*CI] -> dw bf16 [co][kh][kw][ci] (channels_last
// conv-weight memory order).  Each block owns 64 consecutive part-columns
// and walks the rows with 4 row-threads per column: every row visit is a
// 256 B contiguous read, so the reduction streams the partial buffer at
// full bandwidth instead of one 4 B element per 64 B line.
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_wrw_final_k(
    const float* __restrict__ part, long rows, int wn,
    __hip_bfloat16* __restrict__ dw, int CO, int CI) {
  __shared__ float lds[4][64];
  const int c = threadIdx.x & 63, rs = threadIdx.x >> 6;
  const long off = (long)blockIdx.x * 64 + c;
  float s = 0.f;
  if (off < wn)
    for (long r = rs; r < rows; r += 4) s += part[r * wn + off];
  lds[rs][c] = s;
  __syncthreads();
  if (rs == 0 && off < wn) {
    s = lds[0][c] + lds[1][c] + lds[2][c] + lds[3][c];
    // part column (tap, co, ci) -> dw memory index (co, tap, ci)
    const int ci = (int)(off % CI);
    const int co = (int)((off / CI) % CO);
    const int tap = (int)(off / ((long)CO * CI));
    dw[((long)co * 9 + tap) * CI + ci] = __float2bfloat16(s);
  }
}

// ---- layout probe: D[16,16] = A[16,32] x B[32,16], one wave ---------------
__global__ void __launch_bounds__(64) mfma_probe_gemm(
    const __hip_bfloat16* __restrict__ A, const __hip_bfloat16* __restrict__ B,
    float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  bf16x8 a, b;
  const int kbase = (lane >> 4) * 8;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    // A row-major [16, 32]; B row-major [32, 16]
    a[i] = *reinterpret_cast<const __bf16*>(A + (lane & 15) * 32 + kbase + i);
    b[i] = *reinterpret_cast<const __bf16*>(B + (kbase + i) * 16 + (lane & 15));
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = (lane >> 4) * 4 + r, col = lane & 15;
    D[row * 16 + col] = acc[r];
  }
}
