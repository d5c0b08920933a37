// MFMA direct NHWC 3x3/s1/p1 convolution FORWARD with fused BatchNorm
// epilogue/prologue (gfx950) — the CIFAR ResNet body shapes.
//
// Why (VERDICT r1 "win the conv stack"): the default flagship step spends
// ~56% in MIOpen conv solvers + their SubTensorOp/cast wrappers and ~34%
// in the BN pack; every BN pass re-reads the conv output from HBM.  This
// kernel computes y = conv3x3(x~, w) in ONE launch where
//   * the INPUT transform  x~ = relu(a[c]*x + b[c] (+ res))  is applied
//     while staging x into LDS (the previous layer's BN-normalize+ReLU
//     pass never materializes), and
//   * the OUTPUT epilogue accumulates per-channel sum/sumsq of y (the
//     next BN's statistics) with per-WG wave reductions + one atomicAdd
//     per channel — the separate bnh_stats pass dies.
//
// GEMM mapping (fragment layouts hardware-verified by mfma_probe_gemm,
// convwrw.h:12-16):
//   D[m, n] : m = output pixel (16/M-tile), n = output channel
//   A[m, k] : k = (dh*3+dw)*CI + ci -> x~[n, y+dh-1, x+dw-1, ci]
//   B[k, n] : w[co][dh][dw][ci] (channels_last conv-weight memory order)
//
// One workgroup = R(=8) image rows x W pixels x CO_TILE channels of ONE
// image: x slab (R+2 rows, W+2 halo cols, per-pixel channel pad +8 so the
// 16-lane A-fragment ds_read_b128 group walks distinct banks) + the
// weight tile as [K/8][CO_TILE][8] both in LDS; no K loop over tiles —
// the whole GEMM-K (9*CI <= 576) is resident, so the kernel is stage ->
// barrier -> MFMAs -> epilogue with zero further syncs.
#pragma once
#include <hip/hip_runtime.h>
#include "common.h"
#include <hip/hip_bf16.h>

// bf16x8 / f32x4 come from convwrw.h (same TU)

template <int CI, int CO, int CO_TILE, int W, bool FUSE_IN, bool HAS_RES,
          int S = 1>
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_bn_fwd_k(
    const __hip_bfloat16* __restrict__ x,   // [N, H*S, W*S, CI]
    const __hip_bfloat16* __restrict__ w,   // [CO, 3, 3, CI]
    __hip_bfloat16* __restrict__ y,         // [N, H, W, CO]  (H, W = OUT)
    float* __restrict__ ysum,               // [2*CO] (sum, sumsq) or null
    const float* __restrict__ in_a,         // [CI] scale or null
    const float* __restrict__ in_b,         // [CI] shift
    const __hip_bfloat16* __restrict__ res, // residual or null
    int N, int H, int relu_in) {
  constexpr int R = 8;                    // output rows per workgroup
  constexpr int WI = W * S;               // input row width
  constexpr int KTOT = 9 * CI;
  constexpr int KPAD = (KTOT + 31) & ~31; // MFMA K granularity
  constexpr int K8 = KPAD / 8;
  constexpr int CIP = CI + 8;             // padded pixel slot (elements)
  constexpr int XR = R * S + 2, XC = WI + 2;
  constexpr int NSLOT = XR * XC + 1;      // +1 zero slot for K padding
  constexpr int P = R * W;                // output pixels per WG
  constexpr int MT = P / 16;              // M-tiles
  constexpr int MTW = MT / 4;             // M-tiles per wave
  constexpr int NT = CO_TILE / 16;        // N-tiles
  static_assert(P % 64 == 0, "4 waves x 16-pixel tiles");

  __shared__ __hip_bfloat16 xs[NSLOT * CIP];
  __shared__ __hip_bfloat16 bw[K8 * CO_TILE * 8];
  __shared__ float sa[CI], sb[CI];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int fm = lane & 15, kg = lane >> 4;

  // grid: [N * (H/R) * (CO/CO_TILE)]
  const int cob = blockIdx.x % (CO / CO_TILE);
  const int rb = (blockIdx.x / (CO / CO_TILE)) % (H / R);
  const int n = blockIdx.x / ((CO / CO_TILE) * (H / R));
  const int co0 = cob * CO_TILE;
  const int h0 = rb * R;                  // OUTPUT row base
  const int HI = H * S;                   // input height

  if (FUSE_IN && tid < CI) {
    sa[tid] = in_a[tid];
    sb[tid] = in_b[tid];
  }

  // ---- stage weights: bw[k/8][co][j] = w[co0+co][tap][ci], k=tap*CI+ci --
  for (int e = tid; e < K8 * CO_TILE; e += FT_BLOCK) {
    const int k8 = e / CO_TILE, co = e % CO_TILE;
    const int k = k8 * 8;
    uint4 v = {0, 0, 0, 0};
    if (k < KTOT) {
      // tap*CI+ci runs of 8 never cross a tap boundary (CI % 8 == 0), and
      // [co][tap][ci] is exactly k-major per co: one 16 B global read.
      v = *reinterpret_cast<const uint4*>(w + (long)(co0 + co) * KTOT + k);
    }
    *reinterpret_cast<uint4*>(&bw[((long)k8 * CO_TILE + co) * 8]) = v;
  }

  // ---- stage x slab: rows h0-1..h0+R, halo cols + pad zeroed ------------
  // zero the halo columns, channel pad and the K-pad slot
  for (int e = tid; e < NSLOT; e += FT_BLOCK) {
    // zero pad tail of every slot (8 elements) + whole zero-slot
    *reinterpret_cast<uint4*>(&xs[(long)e * CIP + CI]) = uint4{0, 0, 0, 0};
  }
  for (int e = tid; e < XR * 2 + (CIP / 8); e += FT_BLOCK) {
    if (e < XR * 2) {
      const int row = e >> 1, col = (e & 1) ? (XC - 1) : 0;
      __hip_bfloat16* p = &xs[(long)(row * XC + col) * CIP];
      for (int j = 0; j < CI; j += 8)
        *reinterpret_cast<uint4*>(p + j) = uint4{0, 0, 0, 0};
    } else {
      *reinterpret_cast<uint4*>(
          &xs[(long)(NSLOT - 1) * CIP + (e - XR * 2) * 8]) = uint4{0, 0, 0, 0};
    }
  }
  __syncthreads();  // zero-fill visible before body writes land below

  // body: XR input rows x WI*CI contiguous elements each (16 B chunks)
  constexpr int BCH = WI * CI / 8;
  const float zero = 0.f;
  for (int e = tid; e < XR * BCH; e += FT_BLOCK) {
    const int row = e / BCH, c = e % BCH;
    const int hh = h0 * S - 1 + row;
    uint4 v = {0, 0, 0, 0};
    if (hh >= 0 && hh < HI)
      v = *reinterpret_cast<const uint4*>(
          x + (((long)n * HI + hh) * WI) * CI + c * 8);
    const int pix = c * 8 / CI;          // pixel within the row
    const int ci0 = c * 8 % CI;
    if (FUSE_IN) {
      // x~ = [relu](a*x + b [+ res]) fused into the staging store.
      // Out-of-bounds rows stay ZERO: conv zero-pads the TRANSFORMED
      // input, so the transform must not touch the halo.  FUSE_IN and
      // HAS_RES are COMPILE-TIME: a runtime branch here makes hipcc
      // branch around each staging load and drain vmcnt per element
      // (guide §5 trap (c); measured +12 us per call).
      if (hh >= 0 && hh < HI) {
        uint4 rv = {0, 0, 0, 0};
        if (HAS_RES)
          rv = *reinterpret_cast<const uint4*>(
              res + (((long)n * HI + hh) * WI) * CI + c * 8);
        const __hip_bfloat16* xv =
            reinterpret_cast<const __hip_bfloat16*>(&v);
        const __hip_bfloat16* rr =
            reinterpret_cast<const __hip_bfloat16*>(&rv);
        __hip_bfloat16 out[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float t = fmaf(sa[ci0 + j], (float)xv[j], sb[ci0 + j]);
          if (HAS_RES) t += (float)rr[j];
          if (relu_in) t = fmaxf(t, zero);
          out[j] = (__hip_bfloat16)t;
        }
        v = *reinterpret_cast<const uint4*>(out);
      }
    }
    *reinterpret_cast<uint4*>(
        &xs[(long)(row * XC + 1 + pix) * CIP + ci0]) = v;
  }
  __syncthreads();

  // ---- MFMAs ------------------------------------------------------------
  f32x4 acc[MTW][NT];
#pragma unroll
  for (int mt = 0; mt < MTW; ++mt)
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) acc[mt][nt] = {0.f, 0.f, 0.f, 0.f};

  // this lane's A-fragment pixel per M-tile: p = (wave*MTW + mt)*16 + fm
#pragma unroll
  for (int ks = 0; ks < KPAD / 32; ++ks) {
    const int kk = ks * 32 + kg * 8;     // 8 contiguous k for this lane
    const int tap = kk / CI, ci0 = kk % CI;
    const int dh = tap / 3, dw = tap % 3;
#pragma unroll
    for (int mt = 0; mt < MTW; ++mt) {
      const int p = (wave * MTW + mt) * 16 + fm;
      const int r = p / W, c = p % W;
      const int slot = (tap < 9) ? ((r * S + dh) * XC + (c * S + dw))
                                 : (NSLOT - 1);
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &xs[(long)slot * CIP + ci0]);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &bw[((long)(ks * 4 + kg) * CO_TILE + nt * 16 + fm) * 8]);
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, acc[mt][nt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: store y + per-channel sum/sumsq ------------------------
  // D: col(channel) = fm, row(pixel-in-tile) = kg*4 + reg
  float s[NT], ss[NT];
#pragma unroll
  for (int nt = 0; nt < NT; ++nt) { s[nt] = 0.f; ss[nt] = 0.f; }
#pragma unroll
  for (int mt = 0; mt < MTW; ++mt) {
    const int ptile = (wave * MTW + mt) * 16;
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        const float v = acc[mt][nt][r4];
        s[nt] += v;
        ss[nt] = fmaf(v, v, ss[nt]);
        const int p = ptile + kg * 4 + r4;   // pixel in WG tile
        const int hh = h0 + p / W, cc = p % W;
        y[(((long)n * H + hh) * W + cc) * CO + co0 + nt * 16 + fm] =
            (__hip_bfloat16)v;
      }
    }
  }
  if (ysum != nullptr) {
    // per-WG partial row in bnh_norm_k's [B, C, 2] layout (B = gridDim.x):
    // NO atomics — 1024 WGs atomically adding to 2*CO words measured
    // +12 us/call; bnh_norm_k's block-parallel finalize eats the rows.
    // reduce lanes that share fm (kg = 0..3): xor over bits 4,5 of lane
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
#pragma unroll
      for (int off = 16; off <= 32; off <<= 1) {
        s[nt] += __shfl_xor(s[nt], off, WAVE);
        ss[nt] += __shfl_xor(ss[nt], off, WAVE);
      }
    }
    __shared__ float red[4][NT > 0 ? NT : 1][16][2];
    if (kg == 0) {
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        red[wave][nt][fm][0] = s[nt];
        red[wave][nt][fm][1] = ss[nt];
      }
    }
    __syncthreads();
    float* prow = ysum + (long)blockIdx.x * CO * 2;
    for (int c0 = tid; c0 < CO; c0 += FT_BLOCK) {
      float ts = 0.f, tss = 0.f;
      if (c0 >= co0 && c0 < co0 + CO_TILE) {
        const int nt = (c0 - co0) / 16, ch = (c0 - co0) % 16;
#pragma unroll
        for (int wv = 0; wv < 4; ++wv) {
          ts += red[wv][nt][ch][0];
          tss += red[wv][nt][ch][1];
        }
      }
      prow[c0 * 2] = ts;
      prow[c0 * 2 + 1] = tss;
    }
  }
}


// ==========================================================================
// MFMA direct NHWC 3x3/s1/p1 conv BACKWARD-DATA (gfx950) — the fwd
// kernel's mirror: dx[p, ci] = sum_{tap, co} dy[p - shift(tap)][co] *
// w[co][tap][ci].  GEMM: A[m=pixel][k=(tap, co)] = dy slab reads
// (channels contiguous, same staging as fwd's x), B[k][n=ci] = weights
// gathered as [k/8][CI_TILE][8] with the tap REVERSED (full correlation
// <-> convolution flip), K = 9*CO resident in LDS.
// ==========================================================================
template <int CI, int CO, int CI_TILE, int W, bool TRF,
          bool HAS_DRES = false>
__global__ void __launch_bounds__(FT_BLOCK) conv3x3_dgrad_k(
    const __hip_bfloat16* __restrict__ dy,  // [N, H, W, CO] (dz when TRF)
    const __hip_bfloat16* __restrict__ w,   // [CO, 3, 3, CI]
    __hip_bfloat16* __restrict__ dx,        // [N, H, W, CI]
    const __hip_bfloat16* __restrict__ xbn, // conv output (BN input)
    const __hip_bfloat16* __restrict__ zv,  // BN output (ReLU mask src)
    const float* __restrict__ coefs,        // [3, CO] A, B, D
    __hip_bfloat16* __restrict__ dyc,       // transformed dy out (for wrw)
    __hip_bfloat16* __restrict__ dres,      // masked dz out (residual)
    int N, int H, int relu) {
  constexpr int R = 8;
  constexpr int KTOT = 9 * CO;
  constexpr int KPAD = (KTOT + 31) & ~31;
  constexpr int K8 = KPAD / 8;
  constexpr int COP = CO + 8;
  constexpr int XR = R + 2, XC = W + 2;
  constexpr int NSLOT = XR * XC + 1;
  constexpr int P = R * W;
  constexpr int MT = P / 16;
  constexpr int MTW = MT / 4;
  constexpr int NT = CI_TILE / 16;

  __shared__ __hip_bfloat16 ys[NSLOT * COP];     // dy slab (+halo, pad)
  __shared__ __hip_bfloat16 bw[K8 * CI_TILE * 8];
  __shared__ float sA[TRF ? CO : 1], sB[TRF ? CO : 1], sD[TRF ? CO : 1];

  const int tid = threadIdx.x;
  const int wave = tid / WAVE, lane = tid & (WAVE - 1);
  const int fm = lane & 15, kg = lane >> 4;

  const int cib = blockIdx.x % (CI / CI_TILE);
  const int rb = (blockIdx.x / (CI / CI_TILE)) % (H / R);
  const int n = blockIdx.x / ((CI / CI_TILE) * (H / R));
  const int ci0 = cib * CI_TILE;
  const int h0 = rb * R;

  if (TRF && tid < CO) {
    sA[tid] = coefs[tid];
    sB[tid] = coefs[CO + tid];
    sD[tid] = coefs[2 * CO + tid];
  }

  // ---- stage weights: bw[k/8][ci][j] = w[co][2-dh][2-dw][ci0+ci],
  //      k = tap*CO + co (co runs of 8 within one tap: CO % 8 == 0) ----
  for (int e = tid; e < K8 * CI_TILE; e += FT_BLOCK) {
    const int k8 = e / CI_TILE, ci = e % CI_TILE;
    const int k = k8 * 8;
    __hip_bfloat16 vals[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_bfloat16 v; *reinterpret_cast<unsigned short*>(&v) = 0;
      if (k + j < KTOT) {
        const int tap = (k + j) / CO, co = (k + j) % CO;
        const int rev = 8 - tap;                   // (2-dh)*3 + (2-dw)
        v = w[((long)co * 9 + rev) * CI + ci0 + ci];
      }
      vals[j] = v;
    }
    *reinterpret_cast<uint4*>(&bw[((long)k8 * CI_TILE + ci) * 8]) =
        *reinterpret_cast<const uint4*>(vals);
  }

  // ---- stage dy slab (identical structure to the fwd x slab) ----------
  for (int e = tid; e < NSLOT; e += FT_BLOCK)
    *reinterpret_cast<uint4*>(&ys[(long)e * COP + CO]) = uint4{0, 0, 0, 0};
  for (int e = tid; e < XR * 2 + (COP / 8); e += FT_BLOCK) {
    if (e < XR * 2) {
      const int row = e >> 1, col = (e & 1) ? (XC - 1) : 0;
      __hip_bfloat16* p = &ys[(long)(row * XC + col) * COP];
      for (int j = 0; j < CO; j += 8)
        *reinterpret_cast<uint4*>(p + j) = uint4{0, 0, 0, 0};
    } else {
      *reinterpret_cast<uint4*>(
          &ys[(long)(NSLOT - 1) * COP + (e - XR * 2) * 8]) =
          uint4{0, 0, 0, 0};
    }
  }
  __syncthreads();
  constexpr int BCH = W * CO / 8;
  for (int e = tid; e < XR * BCH; e += FT_BLOCK) {
    const int row = e / BCH, c = e % BCH;
    const int hh = h0 - 1 + row;
    uint4 v = {0, 0, 0, 0};
    const long gbase = (((long)n * H + hh) * W) * CO + c * 8;
    if (hh >= 0 && hh < H)
      v = *reinterpret_cast<const uint4*>(dy + gbase);
    const int pix = c * 8 / CO, co8 = c * 8 % CO;
    if (TRF && hh >= 0 && hh < H) {
      // deferred BN backward: dy_conv = A*relu_mask(dz) + B + D*x_bn,
      // staged AND written back out for the weight-gradient consumer
      const uint4 xv4 = *reinterpret_cast<const uint4*>(xbn + gbase);
      const uint4 zv4 = *reinterpret_cast<const uint4*>(zv + gbase);
      const __hip_bfloat16* gg =
          reinterpret_cast<const __hip_bfloat16*>(&v);
      const __hip_bfloat16* xx =
          reinterpret_cast<const __hip_bfloat16*>(&xv4);
      const __hip_bfloat16* zz =
          reinterpret_cast<const __hip_bfloat16*>(&zv4);
      __hip_bfloat16 out[8], gm[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = (float)gg[j];
        if (relu && (float)zz[j] <= 0.f) g = 0.f;
        if (HAS_DRES) gm[j] = (__hip_bfloat16)g;
        out[j] = (__hip_bfloat16)(
            fmaf(sA[co8 + j], g,
                 fmaf(sD[co8 + j], (float)xx[j], sB[co8 + j])));
      }
      if (HAS_DRES)
        *reinterpret_cast<uint4*>(dres + gbase) =
            *reinterpret_cast<const uint4*>(gm);
      v = *reinterpret_cast<const uint4*>(out);
      *reinterpret_cast<uint4*>(dyc + gbase) = v;
    }
    *reinterpret_cast<uint4*>(
        &ys[(long)(row * XC + 1 + pix) * COP + co8]) = v;
  }
  __syncthreads();

  // ---- MFMAs ----------------------------------------------------------
  f32x4 acc[MTW][NT];
#pragma unroll
  for (int mt = 0; mt < MTW; ++mt)
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) acc[mt][nt] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int ks = 0; ks < KPAD / 32; ++ks) {
    const int kk = ks * 32 + kg * 8;
    const int tap = kk / CO, co8 = kk % CO;
    // dx[p] needs dy[p + (dh-1, dw-1)] for REVERSED w tap; with tap
    // enumerating the REVERSED kernel, the dy offset is (dh, dw) of the
    // forward halo walk: same slot arithmetic as the fwd kernel.
    const int dh = tap / 3, dw = tap % 3;
#pragma unroll
    for (int mt = 0; mt < MTW; ++mt) {
      const int p = (wave * MTW + mt) * 16 + fm;
      const int r = p / W, c = p % W;
      const int slot = (tap < 9) ? ((r + dh) * XC + (c + dw)) : (NSLOT - 1);
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &ys[(long)slot * COP + co8]);
#pragma unroll
      for (int nt = 0; nt < NT; ++nt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &bw[((long)(ks * 4 + kg) * CI_TILE + nt * 16 + fm) * 8]);
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, acc[mt][nt], 0, 0, 0);
      }
    }
  }

  // ---- store dx --------------------------------------------------------
#pragma unroll
  for (int mt = 0; mt < MTW; ++mt) {
    const int ptile = (wave * MTW + mt) * 16;
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
#pragma unroll
      for (int r4 = 0; r4 < 4; ++r4) {
        const int p = ptile + kg * 4 + r4;
        const int hh = h0 + p / W, cc = p % W;
        dx[(((long)n * H + hh) * W + cc) * CI + ci0 + nt * 16 + fm] =
            (__hip_bfloat16)acc[mt][nt][r4];
      }
    }
  }
}
