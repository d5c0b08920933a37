// fedtorch_amd CDNA4 kernel pack (gfx950 / MI355X) — torch extension.
//
// Arena hot paths of the federated engine, each ONE kernel launch over the
// flat parameter arena (design rationale: SURVEY.md §2.3 — the reference
// fedtorch runs every one of these as P~65 per-parameter torch ops).
// Reference semantics cited per op in fedtorch_amd/ops/__init__.py.
//
// Build: hipcc --offload-arch=gfx950 via torch.utils.cpp_extension (see
// setup.py). No CUDA path, no hipify, no multi-backend dispatch.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include "common.h"
#include "batchnorm.h"
#include "stemconv.h"
#include "convwrw.h"
#include "convwrw2.h"
#include "convfwd.h"

#define CHK(x) TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous on GPU")
#define STREAM at::hip::getCurrentHIPStream().stream()

// grid caps overridable via env for on-box tuning sweeps (read once)
static inline int ft_env_int(const char* k, int d) {
  const char* v = getenv(k);
  return v ? atoi(v) : d;
}


// ==========================================================================
// fused dual-mode SGD step (+ per-algorithm corrections)
// ==========================================================================
enum {
  F_DELTA = 1, F_CTRL = 2, F_PROX = 4, F_WD = 8,
  F_IN = 16, F_OUT = 32, F_NESTEROV = 64, F_FIRST_IN = 128,
  F_FIRST_OUT = 256, F_HALF = 512
};

__global__ void fused_sgd_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ bin, float* __restrict__ bout,
    const float* __restrict__ delta, const float* __restrict__ cs,
    const float* __restrict__ cc, const float* __restrict__ server,
    __hip_bfloat16* __restrict__ half_p,
    long n4, long wd_n4, float wd, float m_in, float m_out,
    float omd_in, float omd_out, float step_scale, float prox_mu,
    int flags) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 dv = reinterpret_cast<const float4*>(g)[i];
    float4 pv = reinterpret_cast<float4*>(p)[i];
    float* d = reinterpret_cast<float*>(&dv);
    float* pp = reinterpret_cast<float*>(&pv);
    if (flags & F_DELTA) {
      float4 t = reinterpret_cast<const float4*>(delta)[i];
      const float* tt = reinterpret_cast<const float*>(&t);
#pragma unroll
      for (int j = 0; j < 4; ++j) d[j] -= tt[j];
    }
    if (flags & F_CTRL) {
      float4 a = reinterpret_cast<const float4*>(cs)[i];
      float4 b = reinterpret_cast<const float4*>(cc)[i];
      const float* aa = reinterpret_cast<const float*>(&a);
      const float* bb = reinterpret_cast<const float*>(&b);
#pragma unroll
      for (int j = 0; j < 4; ++j) d[j] += aa[j] - bb[j];
    }
    if (flags & F_PROX) {
      float4 s = reinterpret_cast<const float4*>(server)[i];
      const float* ss = reinterpret_cast<const float*>(&s);
#pragma unroll
      for (int j = 0; j < 4; ++j) d[j] += prox_mu * (pp[j] - ss[j]);
    }
    if ((flags & F_WD) && i < wd_n4) {
#pragma unroll
      for (int j = 0; j < 4; ++j) d[j] = fmaf(wd, pp[j], d[j]);
    }
    if (flags & F_IN) {
      float4 bv;
      if (flags & F_FIRST_IN) {
        bv = dv;
      } else {
        bv = reinterpret_cast<const float4*>(bin)[i];
        float* b = reinterpret_cast<float*>(&bv);
#pragma unroll
        for (int j = 0; j < 4; ++j) b[j] = fmaf(m_in, b[j], omd_in * d[j]);
      }
      reinterpret_cast<float4*>(bin)[i] = bv;
      const float* b = reinterpret_cast<const float*>(&bv);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        d[j] = (flags & F_NESTEROV) ? fmaf(m_in, b[j], d[j]) : b[j];
    }
    if (flags & F_OUT) {
      float4 bv;
      if (flags & F_FIRST_OUT) {
        bv = dv;
      } else {
        bv = reinterpret_cast<const float4*>(bout)[i];
        float* b = reinterpret_cast<float*>(&bv);
#pragma unroll
        for (int j = 0; j < 4; ++j) b[j] = fmaf(m_out, b[j], omd_out * d[j]);
      }
      reinterpret_cast<float4*>(bout)[i] = bv;
      const float* b = reinterpret_cast<const float*>(&bv);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        d[j] = (flags & F_NESTEROV) ? fmaf(m_out, b[j], d[j]) : b[j];
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) pp[j] = fmaf(-step_scale, d[j], pp[j]);
    reinterpret_cast<float4*>(p)[i] = pv;
    if (flags & F_HALF) {  // refresh the bf16 compute copy in the same pass
      struct alignas(8) H4 { __hip_bfloat16 d[4]; };
      H4 hv;
#pragma unroll
      for (int j = 0; j < 4; ++j) hv.d[j] = __float2bfloat16(pp[j]);
      reinterpret_cast<H4*>(half_p)[i] = hv;
    }
  }
}

void fused_sgd_step(torch::Tensor p, torch::Tensor g, torch::Tensor in_buf,
                    torch::Tensor out_buf, torch::Tensor delta,
                    torch::Tensor ctrl_server, torch::Tensor ctrl_client,
                    torch::Tensor server, double lr, double scale, double wd,
                    double m_in, double m_out, double damp, bool nesterov,
                    bool apply_lr, bool apply_in, bool apply_out,
                    bool first_in, bool first_out, double prox_mu,
                    long wd_numel, torch::Tensor half_p) {
  CHK(p); CHK(g);
  TORCH_CHECK(p.numel() % 4 == 0, "arena numel must be float4-aligned");
  long n4 = p.numel() / 4;
  int flags = 0;
  if (delta.numel()) flags |= F_DELTA;
  if (ctrl_server.numel()) flags |= F_CTRL;
  if (prox_mu != 0.0 && server.numel()) flags |= F_PROX;
  if (wd != 0.0 && apply_lr) flags |= F_WD;
  if (apply_in && m_in != 0.0) flags |= F_IN;
  if (apply_out && m_out != 0.0) flags |= F_OUT;
  if (nesterov) flags |= F_NESTEROV;
  if (first_in) flags |= F_FIRST_IN;
  if (first_out) flags |= F_FIRST_OUT;
  if (half_p.numel()) flags |= F_HALF;
  hipLaunchKernelGGL(fused_sgd_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0,
                     STREAM, p.data_ptr<float>(), g.data_ptr<float>(),
                     (flags & F_IN) ? in_buf.data_ptr<float>() : nullptr,
                     (flags & F_OUT) ? out_buf.data_ptr<float>() : nullptr,
                     (flags & F_DELTA) ? delta.data_ptr<float>() : nullptr,
                     (flags & F_CTRL) ? ctrl_server.data_ptr<float>() : nullptr,
                     (flags & F_CTRL) ? ctrl_client.data_ptr<float>() : nullptr,
                     (flags & F_PROX) ? server.data_ptr<float>() : nullptr,
                     (flags & F_HALF) ? reinterpret_cast<__hip_bfloat16*>(
                                            half_p.data_ptr())
                                      : nullptr,
                     n4, wd_numel / 4, (float)wd, (float)m_in, (float)m_out,
                     (float)(1.0 - damp), (float)(1.0 - damp),
                     (float)(apply_lr ? lr : scale), (float)prox_mu, flags);
}

// ==========================================================================
// elementwise arena ops
// ==========================================================================
__global__ void wdr_kernel(const float* __restrict__ s, float* __restrict__ c,
                           float* __restrict__ out, float w, long n4,
                           int restore) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 sv = reinterpret_cast<const float4*>(s)[i];
    float4 cv = reinterpret_cast<float4*>(c)[i];
    float4 ov;
    float* o = reinterpret_cast<float*>(&ov);
    const float* ss = reinterpret_cast<const float*>(&sv);
    const float* ccp = reinterpret_cast<const float*>(&cv);
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = (ss[j] - ccp[j]) * w;
    reinterpret_cast<float4*>(out)[i] = ov;
    if (restore) reinterpret_cast<float4*>(c)[i] = sv;
  }
}

void weighted_diff_restore(torch::Tensor s, torch::Tensor c, torch::Tensor out,
                           double w) {
  CHK(s); CHK(c); CHK(out);
  long n4 = s.numel() / 4;
  hipLaunchKernelGGL(wdr_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0, STREAM,
                     s.data_ptr<float>(), c.data_ptr<float>(),
                     out.data_ptr<float>(), (float)w, n4, 1);
}

void scaled_diff(torch::Tensor a, torch::Tensor b, torch::Tensor out,
                 double w) {
  CHK(a); CHK(b); CHK(out);
  long n4 = a.numel() / 4;
  hipLaunchKernelGGL(wdr_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0, STREAM,
                     a.data_ptr<float>(), b.data_ptr<float>(),
                     out.data_ptr<float>(), (float)w, n4, 0);
}

// y = b*y + a*x ; ef: mem += g*invw - d ; delta += (s-agg-c)*coef ;
// ctrl_new = cc - cs + (s-c)*coef ; blend out = alpha*a+(1-alpha)*b —
// all are 2-4-operand streaming FMAs; one generic kernel each.
__global__ void axpby_kernel(float* __restrict__ y, const float* __restrict__ x,
                             float a, float b, long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 yv = reinterpret_cast<float4*>(y)[i];
    float4 xv = reinterpret_cast<const float4*>(x)[i];
    float* yy = reinterpret_cast<float*>(&yv);
    const float* xx = reinterpret_cast<const float*>(&xv);
#pragma unroll
    for (int j = 0; j < 4; ++j) yy[j] = fmaf(b, yy[j], a * xx[j]);
    reinterpret_cast<float4*>(y)[i] = yv;
  }
}

void axpby(torch::Tensor y, torch::Tensor x, double a, double b) {
  CHK(y); CHK(x);
  long n4 = y.numel() / 4;
  hipLaunchKernelGGL(axpby_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0,
                     STREAM, y.data_ptr<float>(), x.data_ptr<float>(),
                     (float)a, (float)b, n4);
}

__global__ void ef_kernel(float* __restrict__ mem, const float* __restrict__ g,
                          const float* __restrict__ d, float invw, long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 mv = reinterpret_cast<float4*>(mem)[i];
    float4 gv = reinterpret_cast<const float4*>(g)[i];
    float4 dv = reinterpret_cast<const float4*>(d)[i];
    float* m = reinterpret_cast<float*>(&mv);
    const float* gg = reinterpret_cast<const float*>(&gv);
    const float* dd = reinterpret_cast<const float*>(&dv);
#pragma unroll
    for (int j = 0; j < 4; ++j) m[j] = fmaf(invw, gg[j], m[j]) - dd[j];
    reinterpret_cast<float4*>(mem)[i] = mv;
  }
}

void error_feedback_update(torch::Tensor mem, torch::Tensor g, torch::Tensor d,
                           double invw) {
  CHK(mem); CHK(g); CHK(d);
  long n4 = mem.numel() / 4;
  hipLaunchKernelGGL(ef_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0, STREAM,
                     mem.data_ptr<float>(), g.data_ptr<float>(),
                     d.data_ptr<float>(), (float)invw, n4);
}

__global__ void delta_kernel(float* __restrict__ delta,
                             const float* __restrict__ s,
                             const float* __restrict__ agg,
                             const float* __restrict__ c, float coef,
                             long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 dv = reinterpret_cast<float4*>(delta)[i];
    float4 sv = reinterpret_cast<const float4*>(s)[i];
    float4 av = reinterpret_cast<const float4*>(agg)[i];
    float4 cv = reinterpret_cast<const float4*>(c)[i];
    float* d = reinterpret_cast<float*>(&dv);
    const float* ss = reinterpret_cast<const float*>(&sv);
    const float* aa = reinterpret_cast<const float*>(&av);
    const float* cc2 = reinterpret_cast<const float*>(&cv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      d[j] = fmaf(coef, ss[j] - aa[j] - cc2[j], d[j]);
    reinterpret_cast<float4*>(delta)[i] = dv;
  }
}

void delta_update(torch::Tensor delta, torch::Tensor s, torch::Tensor agg,
                  torch::Tensor c, double coef) {
  CHK(delta);
  long n4 = delta.numel() / 4;
  hipLaunchKernelGGL(delta_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0,
                     STREAM, delta.data_ptr<float>(), s.data_ptr<float>(),
                     agg.data_ptr<float>(), c.data_ptr<float>(), (float)coef,
                     n4);
}

__global__ void ctrl_kernel(float* __restrict__ out,
                            const float* __restrict__ cc,
                            const float* __restrict__ cs,
                            const float* __restrict__ s,
                            const float* __restrict__ c, float coef, long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 a = reinterpret_cast<const float4*>(cc)[i];
    float4 b = reinterpret_cast<const float4*>(cs)[i];
    float4 sv = reinterpret_cast<const float4*>(s)[i];
    float4 cv = reinterpret_cast<const float4*>(c)[i];
    float4 ov;
    float* o = reinterpret_cast<float*>(&ov);
    const float* aa = reinterpret_cast<const float*>(&a);
    const float* bb = reinterpret_cast<const float*>(&b);
    const float* ss = reinterpret_cast<const float*>(&sv);
    const float* cc3 = reinterpret_cast<const float*>(&cv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      o[j] = aa[j] - bb[j] + coef * (ss[j] - cc3[j]);
    reinterpret_cast<float4*>(out)[i] = ov;
  }
}

void scaffold_control_update(torch::Tensor out, torch::Tensor cc,
                             torch::Tensor cs, torch::Tensor s,
                             torch::Tensor c, double coef) {
  CHK(out);
  long n4 = out.numel() / 4;
  hipLaunchKernelGGL(ctrl_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0,
                     STREAM, out.data_ptr<float>(), cc.data_ptr<float>(),
                     cs.data_ptr<float>(), s.data_ptr<float>(),
                     c.data_ptr<float>(), (float)coef, n4);
}

__global__ void blend_kernel(float* __restrict__ out,
                             const float* __restrict__ a,
                             const float* __restrict__ b, float alpha,
                             long n4) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 av = reinterpret_cast<const float4*>(a)[i];
    float4 bv = reinterpret_cast<const float4*>(b)[i];
    float4 ov;
    float* o = reinterpret_cast<float*>(&ov);
    const float* aa = reinterpret_cast<const float*>(&av);
    const float* bb = reinterpret_cast<const float*>(&bv);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      o[j] = fmaf(alpha, aa[j] - bb[j], bb[j]);
    reinterpret_cast<float4*>(out)[i] = ov;
  }
}

void blend(torch::Tensor out, torch::Tensor a, torch::Tensor b, double alpha) {
  CHK(out);
  long n4 = out.numel() / 4;
  hipLaunchKernelGGL(blend_kernel, dim3(ft_grid(n4)), dim3(FT_BLOCK), 0,
                     STREAM, out.data_ptr<float>(), a.data_ptr<float>(),
                     b.data_ptr<float>(), (float)alpha, n4);
}

// APFL alpha gradient: dot(p_f - l_f, alpha*p_g + (1-alpha)*l_g) ----------
__global__ void alpha_grad_kernel(const float* __restrict__ lf,
                                  const float* __restrict__ pf,
                                  const float* __restrict__ lg,
                                  const float* __restrict__ pg, float alpha,
                                  long n, float* __restrict__ out) {
  const long stride = (long)gridDim.x * blockDim.x;
  float acc = 0.f;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float dif = pf[i] - lf[i];
    float g = fmaf(alpha, pg[i] - lg[i], lg[i]);
    acc = fmaf(dif, g, acc);
  }
  acc = block_reduce<0>(acc);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

double alpha_grad(torch::Tensor lf, torch::Tensor pf, torch::Tensor lg,
                  torch::Tensor pg, double alpha) {
  CHK(lf);
  auto out = torch::zeros({1}, lf.options());
  long n = lf.numel();
  hipLaunchKernelGGL(alpha_grad_kernel, dim3(ft_grid(n)), dim3(FT_BLOCK), 0,
                     STREAM, lf.data_ptr<float>(), pf.data_ptr<float>(),
                     lg.data_ptr<float>(), pg.data_ptr<float>(), (float)alpha,
                     n, out.data_ptr<float>());
  return out.item<float>();
}

// ==========================================================================
// adaptive quantization (reference flow_utils.py:169-212 semantics)
// ==========================================================================
__global__ void minmaxsum_kernel(const float* __restrict__ x, long n,
                                 float* __restrict__ scratch) {
  // scratch layout: [gridDim] mins | [gridDim] maxs | [gridDim] sums
  const long stride = (long)gridDim.x * blockDim.x;
  float mn = 3.4e38f, mx = -3.4e38f, sm = 0.f;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float v = x[i];
    mn = fminf(mn, v);
    mx = fmaxf(mx, v);
    sm += v;
  }
  float bmn = block_reduce<1>(mn);
  __syncthreads();
  float bmx = block_reduce<2>(mx);
  __syncthreads();
  float bsm = block_reduce<0>(sm);
  if (threadIdx.x == 0) {
    scratch[blockIdx.x] = bmn;
    scratch[gridDim.x + blockIdx.x] = bmx;
    scratch[2 * gridDim.x + blockIdx.x] = bsm;
  }
}

__global__ void quant_info_kernel(const float* __restrict__ scratch,
                                  int nblocks, long n, float qmin, float qmax,
                                  float* __restrict__ info) {
  // single block finalize: info = [scale, zero_point, mean]
  float mn = 3.4e38f, mx = -3.4e38f, sm = 0.f;
  for (int i = threadIdx.x; i < nblocks; i += blockDim.x) {
    mn = fminf(mn, scratch[i]);
    mx = fmaxf(mx, scratch[nblocks + i]);
    sm += scratch[2 * nblocks + i];
  }
  mn = block_reduce<1>(mn);
  __syncthreads();
  mx = block_reduce<2>(mx);
  __syncthreads();
  sm = block_reduce<0>(sm);
  if (threadIdx.x == 0) {
    float mean = sm / (float)n;
    float scale = (mx - mn) / (qmax - qmin);
    if (scale == 0.f) scale = 0.001f;
    float izp = qmin - (mn - mean) / scale;
    // reference: clamp then python int() = truncation toward zero
    float zp = izp < qmin ? qmin : (izp > qmax ? qmax : truncf(izp));
    info[0] = scale;
    info[1] = zp;
    info[2] = mean;
  }
}

template <typename QT>
__global__ void quant_encode_kernel(const float* __restrict__ x, long n,
                                    const float* __restrict__ info,
                                    float qmin, float qmax,
                                    QT* __restrict__ q) {
  const float scale = info[0], zp = info[1], mean = info[2];
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float v = zp + (x[i] - mean) / scale;
    v = fminf(fmaxf(v, qmin), qmax);
    q[i] = (QT)rintf(v);  // round-half-even, matches torch round_()
  }
}

std::vector<torch::Tensor> quantize_adaptive(torch::Tensor x, long num_bits) {
  CHK(x);
  long n = x.numel();
  float qmin = -exp2f((float)(num_bits - 1));
  float qmax = exp2f((float)(num_bits - 1)) - 1.f;
  int blocks = ft_grid(n);
  auto scratch = torch::empty({3 * blocks}, x.options());
  auto info = torch::empty({3}, x.options());
  hipLaunchKernelGGL(minmaxsum_kernel, dim3(blocks), dim3(FT_BLOCK), 0,
                     STREAM, x.data_ptr<float>(), n,
                     scratch.data_ptr<float>());
  hipLaunchKernelGGL(quant_info_kernel, dim3(1), dim3(FT_BLOCK), 0, STREAM,
                     scratch.data_ptr<float>(), blocks, n, qmin, qmax,
                     info.data_ptr<float>());
  torch::Tensor q;
  if (num_bits == 8) {
    q = torch::empty({n}, x.options().dtype(torch::kChar));
    hipLaunchKernelGGL(quant_encode_kernel<int8_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, x.data_ptr<float>(), n,
                       info.data_ptr<float>(), qmin, qmax,
                       q.data_ptr<int8_t>());
  } else {
    q = torch::empty({n}, x.options().dtype(torch::kShort));
    hipLaunchKernelGGL(quant_encode_kernel<int16_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, x.data_ptr<float>(), n,
                       info.data_ptr<float>(), qmin, qmax,
                       q.data_ptr<int16_t>());
  }
  return {q, info};
}

template <typename QT>
__global__ void dequant_kernel(const QT* __restrict__ q, long n,
                               const float* __restrict__ info,
                               float* __restrict__ out) {
  const float scale = info[0], zp = info[1], mean = info[2];
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = fmaf(scale, (float)q[i] - zp, mean);
}

torch::Tensor dequantize(torch::Tensor q, torch::Tensor info) {
  CHK(q);
  long n = q.numel();
  auto out = torch::empty({n}, info.options());
  if (q.scalar_type() == torch::kChar)
    hipLaunchKernelGGL(dequant_kernel<int8_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, q.data_ptr<int8_t>(), n,
                       info.data_ptr<float>(), out.data_ptr<float>());
  else
    hipLaunchKernelGGL(dequant_kernel<int16_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, q.data_ptr<int16_t>(), n,
                       info.data_ptr<float>(), out.data_ptr<float>());
  return out;
}

template <typename QT>
__global__ void dequant_acc_kernel(const QT* __restrict__ qs, long k, long n,
                                   const float* __restrict__ infos,
                                   float* __restrict__ out) {
  // qs: [k, n]; infos: [k, 3]; out[i] = sum_j scale_j*(q_ji - zp_j) + mean_j
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float acc = 0.f;
    for (long j = 0; j < k; ++j) {
      const float scale = infos[3 * j], zp = infos[3 * j + 1],
                  mean = infos[3 * j + 2];
      acc += fmaf(scale, (float)qs[j * n + i] - zp, mean);
    }
    out[i] = acc;
  }
}

void dequant_accumulate(torch::Tensor qs, torch::Tensor infos,
                        torch::Tensor out) {
  CHK(qs); CHK(out);
  long k = qs.size(0), n = qs.size(1);
  if (qs.scalar_type() == torch::kChar)
    hipLaunchKernelGGL(dequant_acc_kernel<int8_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, qs.data_ptr<int8_t>(), k, n,
                       infos.data_ptr<float>(), out.data_ptr<float>());
  else
    hipLaunchKernelGGL(dequant_acc_kernel<int16_t>, dim3(ft_grid(n)),
                       dim3(FT_BLOCK), 0, STREAM, qs.data_ptr<int16_t>(), k,
                       n, infos.data_ptr<float>(), out.data_ptr<float>());
}

// ==========================================================================
// top-k |x| selection: 4-pass device-side radix select + compaction
// (the reference's `x.abs().topk(k)` per tensor, `flow_utils.py:218-230`).
// ==========================================================================
__global__ void radix_hist_kernel(const float* __restrict__ x, long n,
                                  const unsigned int* __restrict__ state,
                                  int shift, unsigned int* __restrict__ hist) {
  // state = {prefix, k_remaining}; count keys matching prefix above `shift`
  __shared__ unsigned int lhist[256];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) lhist[i] = 0;
  __syncthreads();
  const unsigned int prefix = state[0];
  const unsigned int pmask = (shift == 24) ? 0u
      : (0xFFFFFFFFu << (shift + 8));
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned int key = abs_key(x[i]);
    if ((key & pmask) == (prefix & pmask))
      atomicAdd(&lhist[(key >> shift) & 0xFF], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    if (lhist[i]) atomicAdd(&hist[i], lhist[i]);
}

__global__ void radix_pick_kernel(unsigned int* __restrict__ hist, int shift,
                                  unsigned int* __restrict__ state) {
  // ONE thread walks the 256 buckets from the top: find the bucket where the
  // k-th largest key lands, fold it into the prefix, update k_remaining.
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  unsigned int k = state[1];
  unsigned int cum = 0;
  int b = 255;
  for (; b >= 0; --b) {
    unsigned int c = hist[b];
    if (cum + c >= k) break;
    cum += c;
  }
  if (b < 0) b = 0;
  state[0] |= ((unsigned int)b) << shift;
  state[1] = k - cum;
  for (int i = 0; i < 256; ++i) hist[i] = 0;  // reset for next pass
}

__global__ void topk_compact_kernel(const float* __restrict__ x, long n,
                                    const unsigned int* __restrict__ state,
                                    long k, float* __restrict__ v,
                                    int* __restrict__ idx,
                                    unsigned int* __restrict__ counters) {
  // counters[0]: slots for keys > threshold; counters[1]: ties (== thr).
  const unsigned int thr = state[0];
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned int key = abs_key(x[i]);
    if (key > thr) {
      unsigned int slot = atomicAdd(&counters[0], 1u);
      v[slot] = x[i];
      idx[slot] = (int)i;
    }
  }
}

__global__ void topk_ties_kernel(const float* __restrict__ x, long n,
                                 const unsigned int* __restrict__ state,
                                 long k, float* __restrict__ v,
                                 int* __restrict__ idx,
                                 unsigned int* __restrict__ counters) {
  const unsigned int thr = state[0];
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    unsigned int key = abs_key(x[i]);
    if (key == thr) {
      unsigned int slot = atomicAdd(&counters[0], 1u);
      if (slot < (unsigned int)k) {
        v[slot] = x[i];
        idx[slot] = (int)i;
      }
    }
  }
}

std::vector<torch::Tensor> topk_compress(torch::Tensor x, long k) {
  CHK(x);
  long n = x.numel();
  TORCH_CHECK(k > 0 && k <= n, "invalid k");
  auto u32 = x.options().dtype(torch::kUInt32);
  auto state = torch::tensor({(int64_t)0, (int64_t)k},
                             torch::dtype(torch::kUInt32))
                   .to(x.device(), /*non_blocking=*/true);
  auto hist = torch::zeros({256}, u32);
  auto v = torch::empty({k}, x.options());
  auto idx = torch::empty({k}, x.options().dtype(torch::kInt32));
  auto counters = torch::zeros({2}, u32);
  for (int shift = 24; shift >= 0; shift -= 8) {
    hipLaunchKernelGGL(radix_hist_kernel, dim3(ft_grid(n)), dim3(FT_BLOCK), 0,
                       STREAM, x.data_ptr<float>(), n,
                       state.data_ptr<unsigned int>(), shift,
                       hist.data_ptr<unsigned int>());
    hipLaunchKernelGGL(radix_pick_kernel, dim3(1), dim3(1), 0, STREAM,
                       hist.data_ptr<unsigned int>(), shift,
                       state.data_ptr<unsigned int>());
  }
  hipLaunchKernelGGL(topk_compact_kernel, dim3(ft_grid(n)), dim3(FT_BLOCK), 0,
                     STREAM, x.data_ptr<float>(), n,
                     state.data_ptr<unsigned int>(), k, v.data_ptr<float>(),
                     idx.data_ptr<int>(), counters.data_ptr<unsigned int>());
  hipLaunchKernelGGL(topk_ties_kernel, dim3(ft_grid(n)), dim3(FT_BLOCK), 0,
                     STREAM, x.data_ptr<float>(), n,
                     state.data_ptr<unsigned int>(), k, v.data_ptr<float>(),
                     idx.data_ptr<int>(), counters.data_ptr<unsigned int>());
  return {v, idx};
}

// batched virtual-client aggregation: ONE pass over the [C, N] replica
// arena — out[i] = sum_c w[c] * (server[i] - replicas[c][i]).
// (packed mode, fedtorch_amd/parallel/multiclient.py; replaces C separate
// ==========================================================================
// FedAdam server normalizer (reference `federated/fedavg.py:81-85`, after
// arXiv:2003.00295): per-parameter-tensor v_p = beta*v_p + (1-beta)*||g_p||,
// g_p /= (sqrt(v_p)+tau).  ONE kernel, one workgroup per tensor segment,
// v resident on-device — the reference (and round-1 repo) ran a Python
// loop with a float(torch.norm(g)) host sync per tensor per sync round.
__global__ void fedadam_norm_kernel(float* __restrict__ g,
                                    const long* __restrict__ seg,
                                    float* __restrict__ v,
                                    float beta, float tau) {
  const int p = blockIdx.x;
  const long s = seg[2 * p], e = seg[2 * p + 1];
  __shared__ float red[FT_BLOCK / 64];
  __shared__ float inv_s;
  float acc = 0.f;
  for (long i = s + threadIdx.x; i < e; i += blockDim.x) {
    const float x = g[i];
    acc += x * x;
  }
#pragma unroll
  for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int i = 0; i < FT_BLOCK / 64; ++i) t += red[i];
    const float nv = beta * v[p] + (1.f - beta) * sqrtf(t);
    v[p] = nv;
    inv_s = 1.f / (sqrtf(nv) + tau);
  }
  __syncthreads();
  const float inv = inv_s;
  for (long i = s + threadIdx.x; i < e; i += blockDim.x) g[i] *= inv;
}

void fedadam_normalize(torch::Tensor g, torch::Tensor seg, torch::Tensor v,
                       double beta, double tau) {
  CHK(g); CHK(seg); CHK(v);
  TORCH_CHECK(seg.scalar_type() == torch::kLong && seg.numel() == 2 * v.numel(),
              "seg must be int64 [P,2]");
  hipLaunchKernelGGL(fedadam_norm_kernel, dim3((int)v.numel()),
                     dim3(FT_BLOCK), 0, STREAM, g.data_ptr<float>(),
                     seg.data_ptr<long>(), v.data_ptr<float>(),
                     (float)beta, (float)tau);
}

// diff+add launches.)
__global__ void multi_diff_acc_kernel(const float* __restrict__ server,
                                      const float* __restrict__ replicas,
                                      const float* __restrict__ w, long C,
                                      long n4, float wsum,
                                      float* __restrict__ out) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 sv = reinterpret_cast<const float4*>(server)[i];
    const float* ss = reinterpret_cast<const float*>(&sv);
    float acc[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[j] = wsum * ss[j];
    for (long c = 0; c < C; ++c) {
      float wc = w[c];
      if (wc == 0.f) continue;
      float4 rv = reinterpret_cast<const float4*>(replicas + c * n4 * 4)[i];
      const float* rr = reinterpret_cast<const float*>(&rv);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] = fmaf(-wc, rr[j], acc[j]);
    }
    float4 ov;
    float* oo = reinterpret_cast<float*>(&ov);
#pragma unroll
    for (int j = 0; j < 4; ++j) oo[j] = acc[j];
    reinterpret_cast<float4*>(out)[i] = ov;
  }
}

void multi_diff_accumulate(torch::Tensor server, torch::Tensor replicas,
                           torch::Tensor weights, torch::Tensor out,
                           double wsum) {
  CHK(server); CHK(replicas); CHK(out);
  long C = replicas.size(0);
  long n4 = server.numel() / 4;
  hipLaunchKernelGGL(multi_diff_acc_kernel, dim3(ft_grid(n4)),
                     dim3(FT_BLOCK), 0, STREAM, server.data_ptr<float>(),
                     replicas.data_ptr<float>(), weights.data_ptr<float>(),
                     C, n4, (float)wsum, out.data_ptr<float>());
}

// fused decompress + K-way sum: out = sum_k scatter(vs[k] @ idxs[k]) -------
__global__ void scatter_acc_kernel(const float* __restrict__ vs,
                                   const int* __restrict__ idxs, long total,
                                   float* __restrict__ out) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride)
    atomicAdd(&out[idxs[i]], vs[i]);
}

void scatter_accumulate(torch::Tensor out, torch::Tensor vs,
                        torch::Tensor idxs) {
  CHK(out); CHK(vs); CHK(idxs);
  out.zero_();
  long total = vs.numel();
  hipLaunchKernelGGL(scatter_acc_kernel, dim3(ft_grid(total)), dim3(FT_BLOCK),
                     0, STREAM, vs.data_ptr<float>(), idxs.data_ptr<int>(),
                     total, out.data_ptr<float>());
}

// ==========================================================================
// grad gather: scattered autograd-owned grad buffers -> flat grad arena.
//
// With p.grad = None at hipGraph-capture time, AccumulateGrad STEALS the
// produced tensor instead of launching one fp32 add per parameter per step
// (~65 CUDAFunctor_add kernels/step on ResNet-20, ~300 us at b256 —
// profiles/r01_bench_notes.md).  This single kernel then copies every
// stolen buffer into the contiguous grad arena that the fused SGD step and
// the grad-norm trackers read.  Chunk table is built once per capture on
// the host (shapes are static).
// ==========================================================================
// table row: (src_idx, src_off, dst_off, n, is_bf16).  bf16 sources (grads
// of bf16-compute params) are cast to fp32 during the copy — no separate
// cast kernel.
__global__ void gather_chunks_kernel(const unsigned long long* __restrict__
                                         srcs,
                                     const int* __restrict__ table /*[B,5]*/,
                                     float* __restrict__ dst) {
  const int* e = table + 5 * blockIdx.x;
  float* out = dst + e[2];
  const int n = e[3];
  if (e[4]) {  // bf16 source
    const __hip_bfloat16* src =
        reinterpret_cast<const __hip_bfloat16*>(srcs[e[0]]) + e[1];
    const int n8 = n >> 3;
    for (int i = threadIdx.x; i < n8; i += blockDim.x) {
      struct alignas(16) V8 { __hip_bfloat16 d[8]; };
      V8 v = reinterpret_cast<const V8*>(src)[i];
      float4 lo, hi;
      float* l = reinterpret_cast<float*>(&lo);
      float* h = reinterpret_cast<float*>(&hi);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        l[j] = __bfloat162float(v.d[j]);
        h[j] = __bfloat162float(v.d[4 + j]);
      }
      reinterpret_cast<float4*>(out)[2 * i] = lo;
      reinterpret_cast<float4*>(out)[2 * i + 1] = hi;
    }
    for (int i = (n8 << 3) + threadIdx.x; i < n; i += blockDim.x)
      out[i] = __bfloat162float(src[i]);
    return;
  }
  const float* src = reinterpret_cast<const float*>(srcs[e[0]]) + e[1];
  const int n4 = n >> 2;
  for (int i = threadIdx.x; i < n4; i += blockDim.x)
    reinterpret_cast<float4*>(out)[i] =
        reinterpret_cast<const float4*>(src)[i];
  for (int i = (n4 << 2) + threadIdx.x; i < n; i += blockDim.x)
    out[i] = src[i];
}

void gather_grads(torch::Tensor srcs, torch::Tensor table,
                  torch::Tensor dst) {
  CHK(dst); CHK(srcs); CHK(table);
  TORCH_CHECK(table.dim() == 2 && table.size(1) == 5, "table must be [B,5]");
  TORCH_CHECK(srcs.scalar_type() == torch::kLong, "srcs must be int64 ptrs");
  TORCH_CHECK(table.scalar_type() == torch::kInt, "table must be int32");
  const int B = (int)table.size(0);
  hipLaunchKernelGGL(gather_chunks_kernel, dim3(B), dim3(FT_BLOCK), 0,
                     STREAM,
                     reinterpret_cast<const unsigned long long*>(
                         srcs.data_ptr<long>()),
                     table.data_ptr<int>(), dst.data_ptr<float>());
}

// fp32 master arena -> bf16 compute arena (one pass; used after any direct
// mutation of the master outside the fused SGD kernel, e.g. aggregation)
__global__ void cast_half_kernel(const float* __restrict__ src,
                                 __hip_bfloat16* __restrict__ dst, long n8) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n8;
       i += stride) {
    float4 lo = reinterpret_cast<const float4*>(src)[2 * i];
    float4 hi = reinterpret_cast<const float4*>(src)[2 * i + 1];
    struct alignas(16) V8 { __hip_bfloat16 d[8]; };
    V8 v;
    const float* l = reinterpret_cast<const float*>(&lo);
    const float* h = reinterpret_cast<const float*>(&hi);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v.d[j] = __float2bfloat16(l[j]);
      v.d[4 + j] = __float2bfloat16(h[j]);
    }
    reinterpret_cast<V8*>(dst)[i] = v;
  }
}

void cast_to_half(torch::Tensor src, torch::Tensor dst) {
  CHK(src); CHK(dst);
  TORCH_CHECK(src.numel() == dst.numel() && src.numel() % 8 == 0,
              "cast_to_half: numel mismatch / not 8-aligned");
  long n8 = src.numel() / 8;
  hipLaunchKernelGGL(cast_half_kernel, dim3(ft_grid(n8)), dim3(FT_BLOCK), 0,
                     STREAM, src.data_ptr<float>(),
                     reinterpret_cast<__hip_bfloat16*>(dst.data_ptr()), n8);
}

// ==========================================================================
// MFMA wrw v2 (convwrw2.h): transposed LDS staging, alignbit tap shifts
// ==========================================================================
torch::Tensor conv3x3_wrw2(torch::Tensor dy, torch::Tensor x) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda() && dy.dim() == 4 && x.dim() == 4,
              "conv3x3_wrw2: 4D GPU tensors");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast)
                  && x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3_wrw2: channels_last only");
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16
                  && x.scalar_type() == torch::kBFloat16,
              "conv3x3_wrw2: bf16 only");
  const int N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = dy.size(1);
  const bool ok = (Co == Ci) && ((Co == 16 && W == 32) ||
                                 (Co == 32 && W == 16) ||
                                 (Co == 64 && W == 8));
  TORCH_CHECK(ok, "conv3x3_wrw2: unsupported (Co,Ci,W)=", Co, ",", Ci, ",",
              W);
  auto f32 = x.options().dtype(torch::kFloat);
  auto dw = torch::empty(
      {Co, Ci, 3, 3},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const __hip_bfloat16* dyp =
      reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr());
  const __hip_bfloat16* xp =
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  static const int cap = ft_env_int("FT_WRW2_NBLK", 1024);
  int nblk, RQ, Q;
  if (Co == 16) {
    const long tiles = (long)N * (H / 4);        // SR=4
    nblk = (int)(tiles < cap ? tiles : cap);
    Q = 1;
    RQ = nblk;                                   // waves merged in-block
  } else if (Co == 32) {
    const long tiles = (long)N * (H / 8);        // SR=8
    nblk = (int)(tiles < cap ? tiles : cap);
    Q = 4;
    RQ = nblk;
  } else {
    const long tiles = (long)N;                  // SPOS = whole image
    int streams = (int)(tiles < cap / 8 ? tiles : cap / 8);
    nblk = streams * 4;
    Q = 16;
    RQ = streams;
  }
  auto part = torch::empty({(long)Q * RQ, 9L * 256}, f32);
  if (Co == 16)
    hipLaunchKernelGGL((conv3x3_wrw2_k<16, 16, 32>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H, RQ);
  else if (Co == 32)
    hipLaunchKernelGGL((conv3x3_wrw2_k<32, 32, 16>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H, RQ);
  else
    hipLaunchKernelGGL((conv3x3_wrw2_k<64, 64, 8>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H, RQ);
  // stripes so the reduce grid covers the chip (~2048 blocks) without
  // dropping below ~64 rows per block
  int stripes = (int)(2048 / (Q * 36));
  const int by_rows = RQ / 64;
  if (stripes > by_rows) stripes = by_rows;
  stripes = stripes < 1 ? 1 : (stripes > 16 ? 16 : stripes);
  auto red = torch::empty({(long)stripes * Q * 9 * 256}, f32);
  const int red_grid = Q * 36 * stripes;
  hipLaunchKernelGGL(conv3x3_wrw2_reduce_k, dim3(red_grid), dim3(FT_BLOCK),
                     0, STREAM, part.data_ptr<float>(), RQ, stripes,
                     red.data_ptr<float>(), Q);
  hipLaunchKernelGGL(conv3x3_wrw2_cast_k,
                     dim3((9 * Co * Ci + FT_BLOCK - 1) / FT_BLOCK),
                     dim3(FT_BLOCK), 0, STREAM, red.data_ptr<float>(),
                     stripes,
                     reinterpret_cast<__hip_bfloat16*>(dw.data_ptr()),
                     Co, Ci);
  return dw;
}

// ==========================================================================
// MFMA direct NHWC 3x3/s1/p1 conv FORWARD with fused BN prologue/epilogue
// (convfwd.h)
// ==========================================================================
torch::Tensor conv3x3_bn_fwd(torch::Tensor x, torch::Tensor w,
                             torch::Tensor ysum, torch::Tensor in_a,
                             torch::Tensor in_b, torch::Tensor res,
                             bool relu_in) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4,
              "conv3x3_bn_fwd: 4D GPU tensors");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3_bn_fwd: channels_last only");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
                  w.scalar_type() == torch::kBFloat16,
              "conv3x3_bn_fwd: bf16 only");
  const int N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = w.size(0);
  const bool ok = (Co == Ci) && ((Co == 16 && W == 32) ||
                                 (Co == 32 && W == 16) ||
                                 (Co == 64 && W == 8));
  TORCH_CHECK(ok, "conv3x3_bn_fwd: unsupported (Co,Ci,W)=",
              Co, ",", Ci, ",", W);
  TORCH_CHECK(H % 8 == 0, "conv3x3_bn_fwd: H % 8 != 0");
  auto y = torch::empty(
      {N, Co, H, W},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  float* ysp = ysum.defined() && ysum.numel() > 0
                   ? ysum.data_ptr<float>() : nullptr;
  if (ysp) {
    const long grid_total =
        (long)N * (H / 8) * (Co == 64 ? 2 : 1);
    TORCH_CHECK(ysum.numel() == grid_total * Co * 2,
                "conv3x3_bn_fwd: ysum must be [grid, Co, 2] fp32, grid=",
                grid_total);
  }
  const float* ap = in_a.defined() && in_a.numel() > 0
                        ? in_a.data_ptr<float>() : nullptr;
  const float* bp = in_b.defined() && in_b.numel() > 0
                        ? in_b.data_ptr<float>() : nullptr;
  const __hip_bfloat16* rp =
      res.defined() && res.numel() > 0
          ? reinterpret_cast<const __hip_bfloat16*>(res.data_ptr())
          : nullptr;
  TORCH_CHECK((ap == nullptr) == (bp == nullptr),
              "conv3x3_bn_fwd: in_a and in_b go together");
  const __hip_bfloat16* xp =
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  const __hip_bfloat16* wp =
      reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  __hip_bfloat16* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr());
  const bool fuse = ap != nullptr;
  const bool hres = rp != nullptr;
  TORCH_CHECK(!hres || fuse, "conv3x3_bn_fwd: residual needs in_a/in_b");
#define FT_CONV_LAUNCH(CI_, CO_, COT_, W_, XN_)                            \
  {                                                                        \
    const int grid = N * (H / 8) * XN_;                                    \
    if (fuse && hres)                                                      \
      hipLaunchKernelGGL((conv3x3_bn_fwd_k<CI_, CO_, COT_, W_, true,       \
                                           true>),                        \
                         dim3(grid), dim3(FT_BLOCK), 0, STREAM, xp, wp,    \
                         yp, ysp, ap, bp, rp, N, H, relu_in ? 1 : 0);      \
    else if (fuse)                                                         \
      hipLaunchKernelGGL((conv3x3_bn_fwd_k<CI_, CO_, COT_, W_, true,       \
                                           false>),                       \
                         dim3(grid), dim3(FT_BLOCK), 0, STREAM, xp, wp,    \
                         yp, ysp, ap, bp, rp, N, H, relu_in ? 1 : 0);      \
    else                                                                   \
      hipLaunchKernelGGL((conv3x3_bn_fwd_k<CI_, CO_, COT_, W_, false,      \
                                           false>),                       \
                         dim3(grid), dim3(FT_BLOCK), 0, STREAM, xp, wp,    \
                         yp, ysp, ap, bp, rp, N, H, relu_in ? 1 : 0);      \
  }
  if (Co == 16) {
    FT_CONV_LAUNCH(16, 16, 16, 32, 1)
  } else if (Co == 32) {
    FT_CONV_LAUNCH(32, 32, 32, 16, 1)
  } else {
    FT_CONV_LAUNCH(64, 64, 32, 8, 2)
  }
#undef FT_CONV_LAUNCH
  return y;
}

// ==========================================================================
// MFMA fused conv fwd, STRIDE 2 (the ResNet transition convs)
// ==========================================================================
torch::Tensor conv3x3s2_bn_fwd(torch::Tensor x, torch::Tensor w,
                               torch::Tensor ysum) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.dim() == 4 &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3s2_bn_fwd: channels_last GPU");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
                  w.scalar_type() == torch::kBFloat16,
              "conv3x3s2_bn_fwd: bf16 only");
  const int N = x.size(0), Ci = x.size(1), HI = x.size(2), WI = x.size(3);
  const int Co = w.size(0);
  const int H = HI / 2, W = WI / 2;
  const bool ok = (Co == 2 * Ci) && ((Ci == 16 && WI == 32) ||
                                     (Ci == 32 && WI == 16));
  TORCH_CHECK(ok, "conv3x3s2_bn_fwd: unsupported (Ci,Co,WI)=", Ci, ",",
              Co, ",", WI);
  TORCH_CHECK(H % 8 == 0, "conv3x3s2_bn_fwd: H_out % 8 != 0");
  auto y = torch::empty(
      {N, Co, H, W},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  float* ysp = ysum.defined() && ysum.numel() > 0
                   ? ysum.data_ptr<float>() : nullptr;
  const int grid = N * (H / 8) * (Co / 32);
  if (ysp)
    TORCH_CHECK(ysum.numel() == (long)grid * Co * 2,
                "conv3x3s2_bn_fwd: ysum must be [grid, Co, 2], grid=",
                grid);
  const __hip_bfloat16* xp =
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  const __hip_bfloat16* wp =
      reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  __hip_bfloat16* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr());
  if (Ci == 16)
    hipLaunchKernelGGL((conv3x3_bn_fwd_k<16, 32, 32, 16, false, false, 2>),
                       dim3(grid), dim3(FT_BLOCK), 0, STREAM, xp, wp, yp,
                       ysp, nullptr, nullptr, nullptr, N, H, 0);
  else
    hipLaunchKernelGGL((conv3x3_bn_fwd_k<32, 64, 32, 8, false, false, 2>),
                       dim3(grid), dim3(FT_BLOCK), 0, STREAM, xp, wp, yp,
                       ysp, nullptr, nullptr, nullptr, N, H, 0);
  return y;
}

// ==========================================================================
// MFMA direct conv backward-data (convfwd.h conv3x3_dgrad_k)
// ==========================================================================
torch::Tensor conv3x3_dgrad(torch::Tensor dy, torch::Tensor w) {
  TORCH_CHECK(dy.is_cuda() && w.is_cuda() && dy.dim() == 4,
              "conv3x3_dgrad: 4D GPU tensors");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3_dgrad: channels_last only");
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16 &&
                  w.scalar_type() == torch::kBFloat16,
              "conv3x3_dgrad: bf16 only");
  const int N = dy.size(0), Co = dy.size(1), H = dy.size(2),
            W = dy.size(3);
  const int Ci = w.size(1);
  const bool ok = (Co == Ci) && ((Co == 16 && W == 32) ||
                                 (Co == 32 && W == 16) ||
                                 (Co == 64 && W == 8));
  TORCH_CHECK(ok, "conv3x3_dgrad: unsupported (Co,Ci,W)=", Co, ",", Ci,
              ",", W);
  TORCH_CHECK(H % 8 == 0, "conv3x3_dgrad: H % 8 != 0");
  auto dx = torch::empty(
      {N, Ci, H, W},
      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  const __hip_bfloat16* dyp =
      reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr());
  const __hip_bfloat16* wp =
      reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  __hip_bfloat16* dxp = reinterpret_cast<__hip_bfloat16*>(dx.data_ptr());
  if (Co == 16)
    hipLaunchKernelGGL((conv3x3_dgrad_k<16, 16, 16, 32, false>),
                       dim3(N * (H / 8)), dim3(FT_BLOCK), 0, STREAM, dyp,
                       wp, dxp, nullptr, nullptr, nullptr, nullptr,
                       nullptr, N, H, 0);
  else if (Co == 32)
    hipLaunchKernelGGL((conv3x3_dgrad_k<32, 32, 32, 16, false>),
                       dim3(N * (H / 8)), dim3(FT_BLOCK), 0, STREAM, dyp,
                       wp, dxp, nullptr, nullptr, nullptr, nullptr,
                       nullptr, N, H, 0);
  else
    hipLaunchKernelGGL((conv3x3_dgrad_k<64, 64, 32, 8, false>),
                       dim3(N * (H / 8) * 2), dim3(FT_BLOCK), 0, STREAM,
                       dyp, wp, dxp, nullptr, nullptr, nullptr, nullptr,
                       nullptr, N, H, 0);
  return dx;
}

// deferred-BN-backward dgrad: dy here is dz (the BN output gradient);
// the kernel applies dy_conv = A*mask(dz) + B + D*x_bn while staging and
// ALSO writes dy_conv out (the wrw consumer reads it) — the separate
// bnh_bwd_dx pass disappears.
std::vector<torch::Tensor> conv3x3_dgrad_bn(
    torch::Tensor dz, torch::Tensor w, torch::Tensor xbn, torch::Tensor z,
    torch::Tensor coefs, bool relu, torch::Tensor dres) {
  TORCH_CHECK(dz.is_cuda() && dz.dim() == 4 &&
                  dz.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3_dgrad_bn: channels_last GPU dz");
  TORCH_CHECK(dz.scalar_type() == torch::kBFloat16 &&
                  w.scalar_type() == torch::kBFloat16,
              "conv3x3_dgrad_bn: bf16 only");
  const int N = dz.size(0), Co = dz.size(1), H = dz.size(2),
            W = dz.size(3);
  const int Ci = w.size(1);
  const bool ok = (Co == Ci) && ((Co == 16 && W == 32) ||
                                 (Co == 32 && W == 16) ||
                                 (Co == 64 && W == 8));
  TORCH_CHECK(ok && H % 8 == 0, "conv3x3_dgrad_bn: unsupported shape");
  TORCH_CHECK(coefs.numel() == 3 * Co && coefs.is_cuda() &&
                  coefs.scalar_type() == torch::kFloat,
              "conv3x3_dgrad_bn: coefs must be fp32 [3, Co]");
  auto dx = torch::empty(
      {N, Ci, H, W},
      dz.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto dyc = torch::empty_like(dz);
  const bool want_dres = dres.defined() && dres.numel() > 0;
  if (want_dres)
    TORCH_CHECK(dres.sizes() == dz.sizes() &&
                    dres.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv3x3_dgrad_bn: dres must match dz, channels_last");
  const __hip_bfloat16* dzp =
      reinterpret_cast<const __hip_bfloat16*>(dz.data_ptr());
  const __hip_bfloat16* wp =
      reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  const __hip_bfloat16* xp =
      reinterpret_cast<const __hip_bfloat16*>(xbn.data_ptr());
  const __hip_bfloat16* zp =
      reinterpret_cast<const __hip_bfloat16*>(z.data_ptr());
  __hip_bfloat16* dxp = reinterpret_cast<__hip_bfloat16*>(dx.data_ptr());
  __hip_bfloat16* dycp = reinterpret_cast<__hip_bfloat16*>(dyc.data_ptr());
  __hip_bfloat16* drp = want_dres
      ? reinterpret_cast<__hip_bfloat16*>(dres.data_ptr()) : nullptr;
  const float* cp = coefs.data_ptr<float>();
  const int rl = relu ? 1 : 0;
#define FT_DG_LAUNCH(CI_, CO_, CT_, W_, XN_)                              \
  {                                                                       \
    if (want_dres)                                                        \
      hipLaunchKernelGGL((conv3x3_dgrad_k<CI_, CO_, CT_, W_, true,        \
                                          true>),                        \
                         dim3(N * (H / 8) * XN_), dim3(FT_BLOCK), 0,      \
                         STREAM, dzp, wp, dxp, xp, zp, cp, dycp, drp, N,  \
                         H, rl);                                          \
    else                                                                  \
      hipLaunchKernelGGL((conv3x3_dgrad_k<CI_, CO_, CT_, W_, true,        \
                                          false>),                       \
                         dim3(N * (H / 8) * XN_), dim3(FT_BLOCK), 0,      \
                         STREAM, dzp, wp, dxp, xp, zp, cp, dycp, drp, N,  \
                         H, rl);                                          \
  }
  if (Co == 16) {
    FT_DG_LAUNCH(16, 16, 16, 32, 1)
  } else if (Co == 32) {
    FT_DG_LAUNCH(32, 32, 32, 16, 1)
  } else {
    FT_DG_LAUNCH(64, 64, 32, 8, 2)
  }
#undef FT_DG_LAUNCH
  return {dx, dyc};
}



// ==========================================================================
// MFMA 3x3/s1/p1 NHWC bf16 conv weight gradient (convwrw.h)
// ==========================================================================
torch::Tensor conv3x3_wrw(torch::Tensor dy, torch::Tensor x) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda() && dy.dim() == 4 && x.dim() == 4,
              "conv3x3_wrw: 4D GPU tensors");
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast)
                  && x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3_wrw: channels_last only");
  TORCH_CHECK(dy.scalar_type() == torch::kBFloat16
                  && x.scalar_type() == torch::kBFloat16,
              "conv3x3_wrw: bf16 only");
  const int N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = dy.size(1);
  TORCH_CHECK(dy.size(0) == N && dy.size(2) == H && dy.size(3) == W,
              "conv3x3_wrw: stride-1 same-shape only");
  const bool ok = (Co == Ci) && ((Co == 16 && W == 32) ||
                                 (Co == 32 && W == 16) ||
                                 (Co == 64 && W == 8));
  TORCH_CHECK(ok, "conv3x3_wrw: unsupported (Co,Ci,W)=", Co, ",", Ci, ",", W);
  const int R = 32 / W;
  TORCH_CHECK(H % R == 0, "conv3x3_wrw: H % (32/W) != 0");
  const long tiles = (long)N * (H / R);
  auto f32 = x.options().dtype(torch::kFloat);
  static const int cap1 = ft_env_int("FT_WRW_NBLK1", 256);
  static const int capq = ft_env_int("FT_WRW_NBLKQ", 128);
  int nblk, groups;
  if (Co == 16) {  // Q==1: 4 independent tile streams per block
    groups = 4;
    long b = (tiles + 4 * 4 - 1) / (4 * 4);  // >=4 tiles per stream
    nblk = (int)(b < 1 ? 1 : (b > cap1 ? cap1 : b));
  } else {
    groups = 1;
    long b = tiles > capq ? capq : tiles;
    nblk = (int)(b < 1 ? 1 : b);
  }
  const long rows = (long)nblk * groups;
  auto part = torch::empty({rows, (long)9 * Co * Ci}, f32);
  auto dw = torch::empty(
      {Co, Ci, 3, 3},
      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const __hip_bfloat16* dyp =
      reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr());
  const __hip_bfloat16* xp =
      reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  if (Co == 16)
    hipLaunchKernelGGL((conv3x3_wrw_k<16, 16, 32>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H);
  else if (Co == 32)
    hipLaunchKernelGGL((conv3x3_wrw_k<32, 32, 16>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H);
  else
    hipLaunchKernelGGL((conv3x3_wrw_k<64, 64, 8>), dim3(nblk),
                       dim3(FT_BLOCK), 0, STREAM, dyp, xp,
                       part.data_ptr<float>(), N, H);
  const int wn = Co * 9 * Ci;
  hipLaunchKernelGGL(conv3x3_wrw_final_k, dim3((wn + 63) / 64),
                     dim3(FT_BLOCK), 0, STREAM, part.data_ptr<float>(),
                     rows, wn,
                     reinterpret_cast<__hip_bfloat16*>(dw.data_ptr()), Co,
                     Ci);
  return dw;
}

// ==========================================================================
// MFMA layout probe (convwrw.h) — used by the GPU layout test
// ==========================================================================
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  CHK(A); CHK(B);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 && A.numel() == 16 * 32
                  && B.numel() == 32 * 16, "probe expects A[16,32] B[32,16] bf16");
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat));
  hipLaunchKernelGGL(mfma_probe_gemm, dim3(1), dim3(64), 0, STREAM,
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                     D.data_ptr<float>());
  return D;
}

// ==========================================================================
// NHWC stem convolution (kernels in stemconv.h)
// ==========================================================================
static inline bool stem_cl(const torch::Tensor& t) {
  return t.is_contiguous(at::MemoryFormat::ChannelsLast);
}

torch::Tensor stem_conv_fwd(torch::Tensor x, torch::Tensor w, bool out_bf16) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && stem_cl(x),
              "stem_conv_fwd: x must be 4D channels_last on GPU");
  TORCH_CHECK(w.is_cuda() && w.dim() == 4 && stem_cl(w)
                  && w.scalar_type() == torch::kFloat,
              "stem_conv_fwd: w must be fp32 channels_last");
  const int N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci && w.size(2) == 3 && w.size(3) == 3
                  && Ci == 3 && (Co == 16 || Co == 32),
              "stem_conv_fwd: supported shapes are Ci=3, Co in {16,32}");
  auto y = torch::empty({N, Co, H, W},
                        x.options()
                            .dtype(out_bf16 ? torch::kBFloat16
                                            : torch::kFloat)
                            .memory_format(at::MemoryFormat::ChannelsLast));
  const long total = (long)N * H * W;
  const int grid = ft_grid(total);
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "stem_fwd", [&] {
    using TX = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                  __hip_bfloat16, scalar_t>;
    if (out_bf16) {
      auto* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr());
      if (Co == 16)
        hipLaunchKernelGGL((stem_fwd_k<TX, __hip_bfloat16, 3, 16>),
                           dim3(grid), dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           w.data_ptr<float>(), yp, N, H, W);
      else
        hipLaunchKernelGGL((stem_fwd_k<TX, __hip_bfloat16, 3, 32>),
                           dim3(grid), dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           w.data_ptr<float>(), yp, N, H, W);
    } else {
      auto* yp = reinterpret_cast<float*>(y.data_ptr());
      if (Co == 16)
        hipLaunchKernelGGL((stem_fwd_k<TX, float, 3, 16>), dim3(grid),
                           dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           w.data_ptr<float>(), yp, N, H, W);
      else
        hipLaunchKernelGGL((stem_fwd_k<TX, float, 3, 32>), dim3(grid),
                           dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           w.data_ptr<float>(), yp, N, H, W);
    }
  });
  return y;
}

torch::Tensor stem_conv_wrw(torch::Tensor dy, torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && stem_cl(x), "stem_wrw: x");
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4 && stem_cl(dy), "stem_wrw: dy");
  const int N = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int Co = dy.size(1);
  TORCH_CHECK(Ci == 3 && (Co == 16 || Co == 32),
              "stem_wrw: supported shapes are Ci=3, Co in {16,32}");
  const int wn = Co * 9 * Ci;
  const int B = 1024;
  auto f32 = x.options().dtype(torch::kFloat);
  auto part = torch::empty({B, wn}, f32);
  auto dw = torch::empty(
      {Co, Ci, 3, 3}, f32.memory_format(at::MemoryFormat::ChannelsLast));
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "stem_wrw_x", [&] {
    using TX = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                  __hip_bfloat16, scalar_t>;
    AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16,
                                   dy.scalar_type(), "stem_wrw_dy", [&] {
      using TY = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                    __hip_bfloat16, scalar_t>;
      if (Co == 16)
        hipLaunchKernelGGL((stem_wrw_k<TY, TX, 3, 16>), dim3(B),
                           dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TY*>(dy.data_ptr()),
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           part.data_ptr<float>(), N, H, W);
      else
        hipLaunchKernelGGL((stem_wrw_k<TY, TX, 3, 32>), dim3(B),
                           dim3(FT_BLOCK), 0, STREAM,
                           reinterpret_cast<const TY*>(dy.data_ptr()),
                           reinterpret_cast<const TX*>(x.data_ptr()),
                           part.data_ptr<float>(), N, H, W);
    });
  });
  hipLaunchKernelGGL(stem_wrw_final_k,
                     dim3((wn + FT_BLOCK / WAVE - 1) / (FT_BLOCK / WAVE)),
                     dim3(FT_BLOCK), 0, STREAM, part.data_ptr<float>(), B,
                     wn, dw.data_ptr<float>());
  return dw;
}

// ==========================================================================
// fused spatial BatchNorm (training fwd/bwd) — kernels in batchnorm.h
// ==========================================================================
// number of per-channel partial blocks for the reduction kernels: target
// ~512 blocks total (2/CU for a ~2-5 us kernel), ≤64 partials per channel
// so the consumer's finalize loop stays trivial.
static inline int bn_nparts(long per_v, long C) {
  long target = 512 / (C > 0 ? C : 1);
  long by_work = (per_v + FT_BLOCK - 1) / FT_BLOCK;
  long B = target < by_work ? target : by_work;
  if (B > 64) B = 64;
  if (B < 1) B = 1;
  return (int)B;
}

// grid.x for the elementwise kernels (pure streaming, no partial buffer):
// ~1024 blocks total, 2+ vector-iterations per thread.
static inline int bn_ew_chunks(long per_v, long C) {
  long cap = 1024 / (C > 0 ? C : 1);
  long by_work = (per_v + FT_BLOCK - 1) / FT_BLOCK;
  long c = cap < by_work ? cap : by_work;
  return (int)(c < 1 ? 1 : c);
}

// NHWC eligibility: C a multiple of the 16 B vector width with a
// power-of-two group count ≤ 64 lanes and ≤ 256 (blockDim) channels.
static inline bool bnh_ok(long C, int VN) {
  if (C <= 0 || C > 256 || C % VN) return false;
  long cgc = C / VN;
  return cgc <= 64 && (cgc & (cgc - 1)) == 0;
}
static inline int bnh_lgc(long C, int VN) {
  int l = 0;
  for (long c = C / VN; c > 1; c >>= 1) ++l;
  return l;
}
static inline int bnh_red_grid(long tasks) {
  static const int cap = ft_env_int("FT_BNH_RED_CAP", 256);
  static const int iters = ft_env_int("FT_BNH_RED_ITERS", 8);
  long b = (tasks + FT_BLOCK * iters - 1) / (FT_BLOCK * iters);
  if (b > cap) b = cap;
  return (int)(b < 1 ? 1 : b);
}
static inline int bnh_ew_grid(long tasks) {
  static const int cap = ft_env_int("FT_BNH_EW_CAP", 1024);
  static const int iters = ft_env_int("FT_BNH_EW_ITERS", 2);
  long b = (tasks + FT_BLOCK * iters - 1) / (FT_BLOCK * iters);
  if (b > cap) b = cap;
  return (int)(b < 1 ? 1 : b);
}

// collapse [G, C, 2] stats partials to [B2, C, 2]: bnh_norm_k's
// per-block finalize re-reads ALL partial rows in EVERY normalize block
// (B=1024 conv-epilogue rows measured 21 us/norm vs 7.9 at B~64).
__global__ void __launch_bounds__(FT_BLOCK) part_reduce_k(
    const float* __restrict__ part, long G, int cols /* C*2 */,
    float* __restrict__ out, int B2) {
  // block b: rows [b*G/B2, (b+1)*G/B2), 2D thread split (row-parallel x
  // col) + LDS tree — the first serial-per-column version measured
  // 7.9 us/call (105 us/step over 19 layers)
  __shared__ float lds[FT_BLOCK];
  const long r0 = (long)blockIdx.x * G / B2;
  const long r1 = (long)(blockIdx.x + 1) * G / B2;
  const int rp = FT_BLOCK / cols;            // row-parallel factor
  const int c = threadIdx.x % cols, p = threadIdx.x / cols;
  float s = 0.f;
  if (p < rp)
    for (long r = r0 + p; r < r1; r += rp) s += part[r * cols + c];
  lds[threadIdx.x] = s;
  __syncthreads();
  if (p == 0) {
    for (int i = 1; i < rp; ++i) s += lds[i * cols + c];
    out[(long)blockIdx.x * cols + c] = s;
  }
}

// bn_fwd_train with PRECOMPUTED stats partials (conv3x3_bn_fwd epilogue
// rows, [B, C, 2] fp32): the bnh_stats pass is skipped entirely — one
// bnh_norm_k launch finalizes the partials and normalizes.
std::vector<torch::Tensor> bn_fwd_train_part(
    torch::Tensor x, torch::Tensor part, torch::Tensor weight,
    torch::Tensor bias, torch::Tensor running_mean, torch::Tensor running_var,
    double eps, double momentum, bool relu, torch::Tensor res) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn_fwd_train_part: 4D GPU");
  const bool nhwc =
      x.is_contiguous(at::MemoryFormat::ChannelsLast) && !x.is_contiguous();
  TORCH_CHECK(nhwc, "bn_fwd_train_part: channels_last only");
  long N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(part.is_cuda() && part.dim() == 3 && part.size(1) == C &&
                  part.size(2) == 2 &&
                  part.scalar_type() == torch::kFloat,
              "bn_fwd_train_part: part must be [B, C, 2] fp32");
  auto f32 = x.options().dtype(torch::kFloat);
  auto save_mean = torch::empty({C}, f32);
  auto save_ivar = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto mask = torch::empty({0}, x.options().dtype(torch::kByte));
  bool track = running_mean.numel() > 0;
  if (part.size(0) > 64) {
    const int B2 = 32;
    auto small = torch::empty({B2, C, 2}, f32);
    hipLaunchKernelGGL(part_reduce_k, dim3(B2), dim3(FT_BLOCK), 0, STREAM,
                       part.data_ptr<float>(), part.size(0), (int)(C * 2),
                       small.data_ptr<float>(), B2);
    part = small;
  }
  const int B = (int)part.size(0);
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "bnh_fwd_part", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, scalar_t>;
    constexpr int VN = BnVec<T>::N;
    TORCH_CHECK(bnh_ok(C, VN), "bn_fwd_train_part: unsupported C=", C);
    const int lgc = bnh_lgc(C, VN);
    const long NI = N * HW, tasks = NI << lgc;
    hipLaunchKernelGGL((bnh_norm_k<T, typename BnVec<T>::V, VN>),
                       dim3(bnh_ew_grid(tasks)), dim3(FT_BLOCK), 0, STREAM,
                       reinterpret_cast<const T*>(x.data_ptr()),
                       reinterpret_cast<T*>(y.data_ptr()),
                       part.data_ptr<float>(), B,
                       weight.numel() ? weight.data_ptr<float>() : nullptr,
                       bias.numel() ? bias.data_ptr<float>() : nullptr,
                       save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(),
                       track ? running_mean.data_ptr<float>() : nullptr,
                       track ? running_var.data_ptr<float>() : nullptr,
                       res.numel()
                           ? reinterpret_cast<const T*>(res.data_ptr())
                           : nullptr,
                       nullptr, NI, C, lgc, (float)eps, (float)momentum,
                       relu ? 1 : 0);
  });
  return {y, save_mean, save_ivar, mask};
}

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, torch::Tensor weight,
                                        torch::Tensor bias,
                                        torch::Tensor running_mean,
                                        torch::Tensor running_var,
                                        double eps, double momentum,
                                        bool relu, torch::Tensor res) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn_fwd_train expects 4D GPU");
  const bool nhwc =
      x.is_contiguous(at::MemoryFormat::ChannelsLast) && !x.is_contiguous();
  TORCH_CHECK(nhwc || x.is_contiguous(), "bn_fwd_train: x not contiguous");
  long N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(N * HW < (1L << 31), "bn_fwd_train: N*HW must fit in int32");
  auto f32 = x.options().dtype(torch::kFloat);
  auto save_mean = torch::empty({C}, f32);
  auto save_ivar = torch::empty({C}, f32);
  auto y = torch::empty_like(x);
  auto mask = torch::empty({0}, x.options().dtype(torch::kByte));
  bool track = running_mean.numel() > 0;
  if (nhwc) {
    AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                   "bnh_fwd", [&] {
      using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                   __hip_bfloat16, scalar_t>;
      constexpr int VN = BnVec<T>::N;
      TORCH_CHECK(bnh_ok(C, VN), "bn_fwd_train NHWC: unsupported C=", C);
      const int lgc = bnh_lgc(C, VN);
      const long NI = N * HW, tasks = NI << lgc;
      const int B = bnh_red_grid(tasks);
      auto part = torch::empty({B, C, 2}, f32);
      // measured ~1-2% SLOWER at ResNet-20/b256 (y re-reads are
      // LLC-hot; byte traffic adds overhead) — default off, kept for
      // larger feature maps (profiles/r01_bench_notes.md)
      static const bool use_mask = ft_env_int("FT_BNH_MASK", 0) != 0;
      if (relu && VN == 8 && use_mask)  // bwd reads 1 bit/elem instead of y
        mask = torch::empty({(NI * C) >> 3},
                            x.options().dtype(torch::kByte));
      hipLaunchKernelGGL((bnh_stats_k<T, typename BnVec<T>::V, VN>),
                         dim3(B), dim3(FT_BLOCK), 0, STREAM,
                         reinterpret_cast<const T*>(x.data_ptr()), NI, C,
                         lgc, part.data_ptr<float>());
      hipLaunchKernelGGL((bnh_norm_k<T, typename BnVec<T>::V, VN>),
                         dim3(bnh_ew_grid(tasks)), dim3(FT_BLOCK), 0, STREAM,
                         reinterpret_cast<const T*>(x.data_ptr()),
                         reinterpret_cast<T*>(y.data_ptr()),
                         part.data_ptr<float>(), B,
                         weight.numel() ? weight.data_ptr<float>() : nullptr,
                         bias.numel() ? bias.data_ptr<float>() : nullptr,
                         save_mean.data_ptr<float>(),
                         save_ivar.data_ptr<float>(),
                         track ? running_mean.data_ptr<float>() : nullptr,
                         track ? running_var.data_ptr<float>() : nullptr,
                         res.numel()
                             ? reinterpret_cast<const T*>(res.data_ptr())
                             : nullptr,
                         mask.numel() ? mask.data_ptr<unsigned char>()
                                      : nullptr,
                         NI, C, lgc, (float)eps, (float)momentum,
                         relu ? 1 : 0);
    });
    return {y, save_mean, save_ivar, mask};
  }
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "bn_fwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, scalar_t>;
    const bool vec = (HW % BnVec<T>::N) == 0;
    const int VN = vec ? BnVec<T>::N : 1;
    const long per_v = (N * HW) / VN;
    const int B = bn_nparts(per_v, C);
    auto part = torch::empty({C, B, 2}, f32);
    dim3 rgrid(B, C), egrid(bn_ew_chunks(per_v, C), C);
    auto stats_fn = vec ? bn_stats_k<T, typename BnVec<T>::V, BnVec<T>::N>
                        : bn_stats_k<T, typename Bn1<T>::V, 1>;
    auto norm_fn = vec ? bn_norm_k<T, typename BnVec<T>::V, BnVec<T>::N>
                       : bn_norm_k<T, typename Bn1<T>::V, 1>;
    hipLaunchKernelGGL(stats_fn, rgrid, dim3(FT_BLOCK), 0, STREAM,
                       reinterpret_cast<const T*>(x.data_ptr()), N, C, HW,
                       part.data_ptr<float>());
    hipLaunchKernelGGL(norm_fn, egrid, dim3(FT_BLOCK), 0, STREAM,
                       reinterpret_cast<const T*>(x.data_ptr()),
                       reinterpret_cast<T*>(y.data_ptr()),
                       part.data_ptr<float>(), B,
                       weight.numel() ? weight.data_ptr<float>() : nullptr,
                       bias.numel() ? bias.data_ptr<float>() : nullptr,
                       save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(),
                       track ? running_mean.data_ptr<float>() : nullptr,
                       track ? running_var.data_ptr<float>() : nullptr,
                       res.numel() ? reinterpret_cast<const T*>(res.data_ptr())
                                   : nullptr,
                       N, C, HW, (float)eps, (float)momentum, relu ? 1 : 0);
  });
  return {y, save_mean, save_ivar, mask};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor y, torch::Tensor mask,
                                  torch::Tensor save_mean,
                                  torch::Tensor save_ivar,
                                  torch::Tensor weight, bool relu,
                                  bool has_res) {
  TORCH_CHECK(x.is_cuda() && dy.is_cuda(), "bn_bwd expects GPU tensors");
  const bool nhwc =
      x.is_contiguous(at::MemoryFormat::ChannelsLast) && !x.is_contiguous();
  long N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  TORCH_CHECK(N * HW < (1L << 31), "bn_bwd: N*HW must fit in int32");
  auto f32 = x.options().dtype(torch::kFloat);
  auto dx = torch::empty_like(x);
  auto dres = has_res ? torch::empty_like(x)
                      : torch::empty({0}, x.options());
  auto dweight = torch::empty({C}, f32);
  auto dbias = torch::empty({C}, f32);
  if (nhwc) {
    TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast),
                "bn_bwd NHWC: dy must be channels_last");
    AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                   "bnh_bwd", [&] {
      using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                   __hip_bfloat16, scalar_t>;
      constexpr int VN = BnVec<T>::N;
      TORCH_CHECK(bnh_ok(C, VN), "bn_bwd NHWC: unsupported C=", C);
      const int lgc = bnh_lgc(C, VN);
      const long NI = N * HW, tasks = NI << lgc;
      const int B = bnh_red_grid(tasks);
      auto part = torch::empty({B, C, 2}, f32);
      const unsigned char* mp =
          mask.numel() ? mask.data_ptr<unsigned char>() : nullptr;
      const T* yp = (relu && !mp)
                        ? reinterpret_cast<const T*>(y.data_ptr())
                        : nullptr;
      TORCH_CHECK(!relu || mp || y.numel(), "bn_bwd NHWC: relu needs y/mask");
      hipLaunchKernelGGL((bnh_bwd_stats_k<T, typename BnVec<T>::V, VN>),
                         dim3(B), dim3(FT_BLOCK), 0, STREAM,
                         reinterpret_cast<const T*>(dy.data_ptr()),
                         reinterpret_cast<const T*>(x.data_ptr()), yp, mp,
                         save_mean.data_ptr<float>(),
                         save_ivar.data_ptr<float>(), NI, C, lgc,
                         part.data_ptr<float>(), relu ? 1 : 0);
      hipLaunchKernelGGL((bnh_bwd_dx_k<T, typename BnVec<T>::V, VN>),
                         dim3(bnh_ew_grid(tasks)), dim3(FT_BLOCK), 0, STREAM,
                         reinterpret_cast<const T*>(dy.data_ptr()),
                         reinterpret_cast<const T*>(x.data_ptr()), yp, mp,
                         part.data_ptr<float>(), B,
                         save_mean.data_ptr<float>(),
                         save_ivar.data_ptr<float>(),
                         weight.numel() ? weight.data_ptr<float>() : nullptr,
                         reinterpret_cast<T*>(dx.data_ptr()),
                         has_res ? reinterpret_cast<T*>(dres.data_ptr())
                                 : nullptr,
                         dweight.data_ptr<float>(), dbias.data_ptr<float>(),
                         NI, C, lgc, relu ? 1 : 0);
    });
    return {dx, dweight, dbias, dres};
  }
  TORCH_CHECK(x.is_contiguous() && dy.is_contiguous(),
              "bn_bwd: x/dy must be contiguous");
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "bn_bwd", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, scalar_t>;
    const T* yp = relu ? reinterpret_cast<const T*>(y.data_ptr()) : nullptr;
    const bool vec = (HW % BnVec<T>::N) == 0;
    const int VN = vec ? BnVec<T>::N : 1;
    const long per_v = (N * HW) / VN;
    const int B = bn_nparts(per_v, C);
    auto part = torch::empty({C, B, 2}, f32);
    dim3 rgrid(B, C), egrid(bn_ew_chunks(per_v, C), C);
    auto stats_fn = vec ? bn_bwd_stats_k<T, typename BnVec<T>::V, BnVec<T>::N>
                        : bn_bwd_stats_k<T, typename Bn1<T>::V, 1>;
    auto dx_fn = vec ? bn_bwd_dx_k<T, typename BnVec<T>::V, BnVec<T>::N>
                     : bn_bwd_dx_k<T, typename Bn1<T>::V, 1>;
    hipLaunchKernelGGL(stats_fn, rgrid, dim3(FT_BLOCK), 0,
                       STREAM, reinterpret_cast<const T*>(dy.data_ptr()),
                       reinterpret_cast<const T*>(x.data_ptr()), yp,
                       save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(), N, C, HW,
                       part.data_ptr<float>(), relu ? 1 : 0);
    hipLaunchKernelGGL(dx_fn, egrid, dim3(FT_BLOCK), 0, STREAM,
                       reinterpret_cast<const T*>(dy.data_ptr()),
                       reinterpret_cast<const T*>(x.data_ptr()), yp,
                       part.data_ptr<float>(), B, save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(),
                       weight.numel() ? weight.data_ptr<float>() : nullptr,
                       reinterpret_cast<T*>(dx.data_ptr()),
                       has_res ? reinterpret_cast<T*>(dres.data_ptr())
                               : nullptr,
                       dweight.data_ptr<float>(), dbias.data_ptr<float>(),
                       N, C, HW, relu ? 1 : 0);
  });
  return {dx, dweight, dbias, dres};
}

// deferred BN backward: stats + per-channel affine coefs (+ optional
// dres); the elementwise dx pass moves into the consuming conv's staging
std::vector<torch::Tensor> bn_bwd_defer(
    torch::Tensor dz, torch::Tensor x, torch::Tensor z,
    torch::Tensor save_mean, torch::Tensor save_ivar, torch::Tensor weight,
    bool relu, bool has_res) {
  TORCH_CHECK(x.is_cuda() && dz.is_cuda() &&
                  x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn_bwd_defer: channels_last GPU");
  long N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto f32 = x.options().dtype(torch::kFloat);
  auto coefs = torch::empty({3, C}, f32);
  auto dweight = torch::empty({C}, f32);
  auto dbias = torch::empty({C}, f32);
  auto dres = has_res ? torch::empty_like(x)
                      : torch::empty({0}, x.options());
  AT_DISPATCH_FLOATING_TYPES_AND(at::ScalarType::BFloat16, x.scalar_type(),
                                 "bnh_bwd_defer", [&] {
    using T = std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>,
                                 __hip_bfloat16, scalar_t>;
    constexpr int VN = BnVec<T>::N;
    TORCH_CHECK(bnh_ok(C, VN), "bn_bwd_defer: unsupported C=", C);
    const int lgc = bnh_lgc(C, VN);
    const long NI = N * HW, tasks = NI << lgc;
    const int B = bnh_red_grid(tasks);
    auto part = torch::empty({B, C, 2}, f32);
    const T* yp = relu ? reinterpret_cast<const T*>(z.data_ptr()) : nullptr;
    hipLaunchKernelGGL((bnh_bwd_stats_k<T, typename BnVec<T>::V, VN>),
                       dim3(B), dim3(FT_BLOCK), 0, STREAM,
                       reinterpret_cast<const T*>(dz.data_ptr()),
                       reinterpret_cast<const T*>(x.data_ptr()), yp,
                       nullptr, save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(), NI, C, lgc,
                       part.data_ptr<float>(), relu ? 1 : 0);
    torch::Tensor red = part;
    int Bred = B;
    if (B > 64) {
      // collapse partials first: the 1-block coef kernel over ~1000 rows
      // is a serial latency chain (measured 5.9 us/call, ~98 us/step)
      const int B2 = 32;
      red = torch::empty({B2, C, 2}, f32);
      hipLaunchKernelGGL(part_reduce_k, dim3(B2), dim3(FT_BLOCK), 0,
                         STREAM, part.data_ptr<float>(), B, (int)(C * 2),
                         red.data_ptr<float>(), B2);
      Bred = B2;
    }
    hipLaunchKernelGGL(bnh_bwd_coef_k, dim3(1), dim3(FT_BLOCK), 0, STREAM,
                       red.data_ptr<float>(), Bred, NI,
                       save_mean.data_ptr<float>(),
                       save_ivar.data_ptr<float>(),
                       weight.numel() ? weight.data_ptr<float>() : nullptr,
                       C, coefs.data_ptr<float>(),
                       dweight.data_ptr<float>(), dbias.data_ptr<float>());
    if (has_res) {
      const long total_v = NI * C / VN;
      hipLaunchKernelGGL((bnh_mask_dres_k<T, typename BnVec<T>::V, VN>),
                         dim3(ft_grid(total_v)), dim3(FT_BLOCK), 0, STREAM,
                         reinterpret_cast<const T*>(dz.data_ptr()),
                         reinterpret_cast<const T*>(z.data_ptr()),
                         reinterpret_cast<T*>(dres.data_ptr()), total_v);
    }
  });
  return {coefs, dweight, dbias, dres};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bn_fwd_train", &bn_fwd_train);
  m.def("bn_fwd_train_part", &bn_fwd_train_part);
  m.def("bn_bwd", &bn_bwd);
  m.def("fused_sgd_step", &fused_sgd_step, "fused dual-mode SGD step");
  m.def("weighted_diff_restore", &weighted_diff_restore);
  m.def("scaled_diff", &scaled_diff);
  m.def("axpby", &axpby);
  m.def("error_feedback_update", &error_feedback_update);
  m.def("delta_update", &delta_update);
  m.def("scaffold_control_update", &scaffold_control_update);
  m.def("blend", &blend);
  m.def("alpha_grad", &alpha_grad);
  m.def("quantize_adaptive", &quantize_adaptive);
  m.def("dequantize", &dequantize);
  m.def("dequant_accumulate", &dequant_accumulate);
  m.def("topk_compress", &topk_compress);
  m.def("scatter_accumulate", &scatter_accumulate);
  m.def("multi_diff_accumulate", &multi_diff_accumulate);
  m.def("fedadam_normalize", &fedadam_normalize);
  m.def("gather_grads", &gather_grads);
  m.def("stem_conv_fwd", &stem_conv_fwd);
  m.def("stem_conv_wrw", &stem_conv_wrw);
  m.def("cast_to_half", &cast_to_half);
  m.def("mfma_probe", &mfma_probe);
  m.def("conv3x3_wrw", &conv3x3_wrw);
  m.def("conv3x3_wrw2", &conv3x3_wrw2);
  m.def("conv3x3_dgrad", &conv3x3_dgrad);
  m.def("conv3x3_dgrad_bn", &conv3x3_dgrad_bn);
  m.def("bn_bwd_defer", &bn_bwd_defer);
  m.def("conv3x3_bn_fwd", &conv3x3_bn_fwd);
  m.def("conv3x3s2_bn_fwd", &conv3x3s2_bn_fwd);
}
