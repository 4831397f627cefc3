// Fused bf16 NHWC BatchNorm (+ReLU, +residual) for the ResNet headline
// config.  Round-2 finding (profiles/resnet_kstats_r2.txt): the r1 step
// spent 38% in MIOpen's fp32 BatchNorm kernels and 19% more in the
// bf16<->fp32 cast copies its fp32-only interface forces, vs ~30% in
// the convs themselves.  These kernels read/write the bf16 activation
// DIRECTLY (fp32 arithmetic inside), and fold the adjacent elementwise
// ops into the same pass:
//   fwd:  y = relu?(gamma*xhat + beta [+ res])     xhat=(x-mean)*rstd
//   bwd:  dy_eff = dy * (pre-relu > 0)  [mask recomputed from x — no
//         saved activation]; dbeta = sum(dy_eff); dgamma = sum(dy_eff*xhat);
//         dx = gamma*rstd*(dy_eff - dbeta/R - xhat*dgamma/R); dres = dy_eff
//
// Layout: channels_last (N,C,H,W) = physical [R][C] bf16 with C
// contiguous, R = N*H*W; per-channel params/stats fp32.  Stats and
// backward reductions are two-stage column reduces (the colsum
// pattern) with bf16x8-vectorized row traffic (G13).

#include "../common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));

struct F8 { float v[8]; };

DEVINL F8 up8(bf16x8_t x) {
  F8 r;
  #pragma unroll
  for (int j = 0; j < 8; ++j) r.v[j] = bf2f(x[j]);
  return r;
}

// thread -> (vector-column, row-group) map: CV = C/8 vector columns.
// CV <= 256: one block in x, G = 256/CV row-threads per column.
// CV > 256: gridDim.x column blocks, G = 1.
struct ColMap {
  int vc;     // global vector column (-1 = idle)
  int g;      // row-group index within block
  int G;      // row-groups per block
};

DEVINL ColMap colmap(int C) {
  const int CV = C / 8;
  ColMap m;
  if (CV >= 256) {
    m.vc = blockIdx.x * 256 + (int)threadIdx.x;
    m.g = 0;
    m.G = 1;
    if (m.vc >= CV) m.vc = -1;
  } else {
    m.G = 256 / CV;
    m.vc = (int)threadIdx.x % CV;
    m.g = (int)threadIdx.x / CV;
    // tail threads when CV does not divide 256 would re-walk other
    // groups' rows — idle them (all ResNet C are powers of two, but
    // the kernel must not silently double-count for other C)
    if (m.g >= m.G) m.vc = -1;
  }
  return m;
}

// ---- fwd stats: partial sum / sumsq per channel -> mid[ns][2][C]
__global__ void __launch_bounds__(256)
bn_stats_kernel(const short* __restrict__ x, float* __restrict__ mid,
                long R, int C) {
  // ONE shared object (guide trap 4a): [2][C<=2048] fp32 fold buffer
  __shared__ float acc[2][2048];
  const ColMap m = colmap(C);
  float s[8] = {}, q[8] = {};
  if (m.vc >= 0) {
    const int ns = gridDim.y;
    for (long r = (long)blockIdx.y * m.G + m.g; r < R; r += (long)ns * m.G) {
      F8 xv = up8(*(const bf16x8_t*)(x + r * C + m.vc * 8));
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        s[j] += xv.v[j];
        q[j] += xv.v[j] * xv.v[j];
      }
    }
  }
  float* out = mid + ((long)blockIdx.y * 2) * C;
  if (m.G == 1) {
    if (m.vc >= 0)
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        out[m.vc * 8 + j] = s[j];
        out[C + m.vc * 8 + j] = q[j];
      }
    return;
  }
  // G>1 (C < 2048): DETERMINISTIC fold — per-group partials in
  // [g][C] slots (G*C = 2048 floats per tensor), group 0 sums in
  // fixed g order (atomicAdd would vary fp32 order run to run and
  // break the bitwise-determinism race screen)
  if (m.vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[0][m.g * C + m.vc * 8 + j] = s[j];
      acc[1][m.g * C + m.vc * 8 + j] = q[j];
    }
  }
  __syncthreads();
  if (m.g == 0 && m.vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float ts = 0.f, tq = 0.f;
      for (int gg = 0; gg < m.G; ++gg) {
        ts += acc[0][gg * C + m.vc * 8 + j];
        tq += acc[1][gg * C + m.vc * 8 + j];
      }
      out[m.vc * 8 + j] = ts;
      out[C + m.vc * 8 + j] = tq;
    }
  }
}

// ---- finalize stats: mean/rstd + running-stat update.
// Block = 64 channels x 4 slice-groups (the r2 first cut used one
// thread per channel over ns=1024 slices: 256 us of pure L2 latency
// per call = 46% of the step; this form is a few us).
__global__ void __launch_bounds__(256)
bn_stats_final_kernel(const float* __restrict__ mid, float* __restrict__ mean,
                      float* __restrict__ rstd, float* __restrict__ rmean,
                      float* __restrict__ rvar, int ns, long R, int C,
                      float momentum, float eps) {
  __shared__ float acc[2][16][16];
  const int cl = (int)threadIdx.x & 15;
  const int g = (int)threadIdx.x >> 4;
  const int c = blockIdx.x * 16 + cl;
  float s = 0.f, q = 0.f;
  if (c < C)
    for (int i = g; i < ns; i += 16) {
      s += mid[((long)i * 2) * C + c];
      q += mid[((long)i * 2 + 1) * C + c];
    }
  acc[0][g][cl] = s;
  acc[1][g][cl] = q;
  __syncthreads();
  if (g || c >= C) return;
  s = 0.f; q = 0.f;
  #pragma unroll
  for (int j = 0; j < 16; ++j) {
    s += acc[0][j][cl];
    q += acc[1][j][cl];
  }
  const float mu = s / (float)R;
  float var = q / (float)R - mu * mu;
  var = fmaxf(var, 0.f);
  mean[c] = mu;
  rstd[c] = rsqrtf(var + eps);
  if (rmean) {
    // torch semantics: running update uses the UNBIASED batch variance
    const float ub = (R > 1) ? var * (float)R / (float)(R - 1) : var;
    rmean[c] = (1.f - momentum) * rmean[c] + momentum * mu;
    rvar[c] = (1.f - momentum) * rvar[c] + momentum * ub;
  }
}

// ---- apply: y = relu?(gamma*xhat + beta [+ res])
template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
bn_apply_kernel(const short* __restrict__ x, const short* __restrict__ res,
                const float* __restrict__ mean, const float* __restrict__ rstd,
                const float* __restrict__ w, const float* __restrict__ b,
                short* __restrict__ y, long R, int C) {
  const ColMap m = colmap(C);
  if (m.vc < 0) return;
  float mu[8], rs[8], ga[8], be[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[m.vc * 8 + j];
    rs[j] = rstd[m.vc * 8 + j];
    ga[j] = w[m.vc * 8 + j];
    be[j] = b[m.vc * 8 + j];
  }
  const long stride = (long)gridDim.y * m.G;
  for (long r = (long)blockIdx.y * m.G + m.g; r < R; r += stride) {
    const long off = r * C + m.vc * 8;
    F8 xv = up8(*(const bf16x8_t*)(x + off));
    bf16x8_t out;
    F8 rv;
    if (RES) rv = up8(*(const bf16x8_t*)(res + off));
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = ga[j] * (xv.v[j] - mu[j]) * rs[j] + be[j];
      if (RES) v += rv.v[j];
      if (RELU) v = fmaxf(v, 0.f);
      out[j] = f2bf(v);
    }
    *(bf16x8_t*)(y + off) = out;
  }
}

// ---- bwd reduce: partial sum(dy_eff), sum(dy_eff * xhat) -> mid
template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
bn_bwd_reduce_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                     const short* __restrict__ res,
                     const float* __restrict__ mean,
                     const float* __restrict__ rstd,
                     const float* __restrict__ w, const float* __restrict__ b,
                     float* __restrict__ mid, long R, int C) {
  __shared__ float acc[2][2048];
  const ColMap m = colmap(C);
  float sd[8] = {}, sx[8] = {};
  if (m.vc >= 0) {
    float mu[8], rs[8], ga[8], be[8];
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[m.vc * 8 + j];
      rs[j] = rstd[m.vc * 8 + j];
      ga[j] = w[m.vc * 8 + j];
      be[j] = b[m.vc * 8 + j];
    }
    const int ns = gridDim.y;
    for (long r = (long)blockIdx.y * m.G + m.g; r < R; r += (long)ns * m.G) {
      const long off = r * C + m.vc * 8;
      F8 dv = up8(*(const bf16x8_t*)(dy + off));
      F8 xv = up8(*(const bf16x8_t*)(x + off));
      F8 rv;
      if (RES) rv = up8(*(const bf16x8_t*)(res + off));
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float xh = (xv.v[j] - mu[j]) * rs[j];
        float d = dv.v[j];
        if (RELU) {
          float pre = ga[j] * xh + be[j];
          if (RES) pre += rv.v[j];
          d = pre > 0.f ? d : 0.f;
        }
        sd[j] += d;
        sx[j] += d * xh;
      }
    }
  }
  float* out = mid + ((long)blockIdx.y * 2) * C;
  if (m.G == 1) {
    if (m.vc >= 0)
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        out[m.vc * 8 + j] = sd[j];
        out[C + m.vc * 8 + j] = sx[j];
      }
    return;
  }
  if (m.vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      acc[0][m.g * C + m.vc * 8 + j] = sd[j];
      acc[1][m.g * C + m.vc * 8 + j] = sx[j];
    }
  }
  __syncthreads();
  if (m.g == 0 && m.vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float ts = 0.f, tx = 0.f;
      for (int gg = 0; gg < m.G; ++gg) {
        ts += acc[0][gg * C + m.vc * 8 + j];
        tx += acc[1][gg * C + m.vc * 8 + j];
      }
      out[m.vc * 8 + j] = ts;
      out[C + m.vc * 8 + j] = tx;
    }
  }
}

__global__ void __launch_bounds__(256)
bn_bwd_final_kernel(const float* __restrict__ mid, float* __restrict__ dw,
                    float* __restrict__ db, int ns, int C) {
  __shared__ float acc[2][16][16];
  const int cl = (int)threadIdx.x & 15;
  const int g = (int)threadIdx.x >> 4;
  const int c = blockIdx.x * 16 + cl;
  float sd = 0.f, sx = 0.f;
  if (c < C)
    for (int i = g; i < ns; i += 16) {
      sd += mid[((long)i * 2) * C + c];
      sx += mid[((long)i * 2 + 1) * C + c];
    }
  acc[0][g][cl] = sd;
  acc[1][g][cl] = sx;
  __syncthreads();
  if (g || c >= C) return;
  float td = 0.f, tx = 0.f;
  #pragma unroll
  for (int j = 0; j < 16; ++j) {
    td += acc[0][j][cl];
    tx += acc[1][j][cl];
  }
  db[c] = td;
  dw[c] = tx;
}

// ---- bwd dx (+ dres): dx = gamma*rstd*(dy_eff - db/R - xhat*dw/R)
template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
bn_bwd_dx_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                 const short* __restrict__ res,
                 const float* __restrict__ mean,
                 const float* __restrict__ rstd, const float* __restrict__ w,
                 const float* __restrict__ b, const float* __restrict__ dw,
                 const float* __restrict__ db, short* __restrict__ dx,
                 short* __restrict__ dres, long R, int C) {
  const ColMap m = colmap(C);
  if (m.vc < 0) return;
  const float invR = 1.f / (float)R;
  float mu[8], rs[8], ga[8], be[8], mdb[8], mdw[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    mu[j] = mean[m.vc * 8 + j];
    rs[j] = rstd[m.vc * 8 + j];
    ga[j] = w[m.vc * 8 + j];
    be[j] = b[m.vc * 8 + j];
    mdb[j] = db[m.vc * 8 + j] * invR;
    mdw[j] = dw[m.vc * 8 + j] * invR;
  }
  const long stride = (long)gridDim.y * m.G;
  for (long r = (long)blockIdx.y * m.G + m.g; r < R; r += stride) {
    const long off = r * C + m.vc * 8;
    F8 dv = up8(*(const bf16x8_t*)(dy + off));
    F8 xv = up8(*(const bf16x8_t*)(x + off));
    F8 rv;
    if (RES) rv = up8(*(const bf16x8_t*)(res + off));
    bf16x8_t odx, odr;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float xh = (xv.v[j] - mu[j]) * rs[j];
      float d = dv.v[j];
      if (RELU) {
        float pre = ga[j] * xh + be[j];
        if (RES) pre += rv.v[j];
        d = pre > 0.f ? d : 0.f;
      }
      if (RES) odr[j] = f2bf(d);
      odx[j] = f2bf(ga[j] * rs[j] * (d - mdb[j] - xh * mdw[j]));
    }
    *(bf16x8_t*)(dx + off) = odx;
    if (RES) *(bf16x8_t*)(dres + off) = odr;
  }
}

void check_nhwc(const torch::Tensor& t, const char* nm) {
  TORCH_CHECK(t.dim() == 4 && t.stride(1) == 1, nm,
              " must be 4-D channels_last (stride(1)==1)");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, nm, " must be bf16");
}

dim3 row_grid(long R, int C, int ns) {
  const int CV = C / 8;
  return dim3(CV >= 256 ? (CV + 255) / 256 : 1, ns);
}

int slices(long R, int C) {
  // ~8192 concurrent row streams fills the chip for these reduces;
  // more slices only grow the finalize kernels' work
  const int CV = C / 8;
  const int cvb = CV >= 256 ? (CV + 255) / 256 : 1;
  const int G = CV >= 256 ? 1 : 256 / CV;
  long want = 4096 / ((long)G * cvb);
  if (want > 512) want = 512;
  const long max_useful = (R + G - 1) / G;
  if (want > max_useful) want = max_useful;
  return (int)std::max<long>(1, want);
}

}  // namespace

// y = relu?(BN(x) [+ res]); returns {y, mean, rstd} (training mode).
std::vector<torch::Tensor> bn_fwd(torch::Tensor x,
                                  c10::optional<torch::Tensor> res,
                                  torch::Tensor weight, torch::Tensor bias,
                                  c10::optional<torch::Tensor> running_mean,
                                  c10::optional<torch::Tensor> running_var,
                                  double momentum, double eps, bool relu) {
  CHECK_CUDA(x);
  check_nhwc(x, "x");
  const int C = (int)x.size(1);
  TORCH_CHECK(C % 8 == 0, "bn_fwd: C % 8 != 0");
  const long R = x.numel() / C;
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto mean = torch::empty({C}, fopt);
  auto rstd = torch::empty({C}, fopt);
  auto y = torch::empty_like(x);
  auto stream = scaffold_stream();
  const int ns = slices(R, C);
  auto mid = torch::empty({ns, 2, (long)C}, fopt);
  hipLaunchKernelGGL(bn_stats_kernel, row_grid(R, C, ns), dim3(256), 0, stream,
                     (const short*)x.data_ptr(), mid.data_ptr<float>(), R, C);
  hipLaunchKernelGGL(bn_stats_final_kernel, dim3((C + 15) / 16), dim3(256),
                     0, stream, mid.data_ptr<float>(), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(),
                     running_mean ? running_mean->data_ptr<float>() : nullptr,
                     running_var ? running_var->data_ptr<float>() : nullptr,
                     ns, R, C, (float)momentum, (float)eps);
  const short* rp = nullptr;
  if (res.has_value()) {
    check_nhwc(res.value(), "res");
    rp = (const short*)res->data_ptr();
  }
  #define APPLY(RL, RS)                                                        \
    hipLaunchKernelGGL((bn_apply_kernel<RL, RS>), row_grid(R, C, ns),          \
                       dim3(256), 0, stream, (const short*)x.data_ptr(), rp,   \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),         \
                       weight.data_ptr<float>(), bias.data_ptr<float>(),       \
                       (short*)y.data_ptr(), R, C)
  if (relu && rp) APPLY(true, true);
  else if (relu) APPLY(true, false);
  else if (rp) APPLY(false, true);
  else APPLY(false, false);
  #undef APPLY
  HIP_CHECK(hipGetLastError());
  return {y, mean, rstd};
}

// returns {dx, dw, db} or {dx, dres, dw, db} when res participates.
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  c10::optional<torch::Tensor> res,
                                  torch::Tensor weight, torch::Tensor bias,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  bool relu) {
  CHECK_CUDA(dy);
  check_nhwc(x, "x");
  const int C = (int)x.size(1);
  const long R = x.numel() / C;
  auto dyc = dy.stride(1) == 1 ? dy : dy.contiguous(at::MemoryFormat::ChannelsLast);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto dw = torch::empty({C}, fopt);
  auto db = torch::empty({C}, fopt);
  auto dx = torch::empty_like(x);
  torch::Tensor dres;
  auto stream = scaffold_stream();
  const int ns = slices(R, C);
  auto mid = torch::empty({ns, 2, (long)C}, fopt);
  const short* rp = nullptr;
  short* drp = nullptr;
  if (res.has_value()) {
    rp = (const short*)res->data_ptr();
    dres = torch::empty_like(x);
    drp = (short*)dres.data_ptr();
  }
  #define RED(RL, RS)                                                          \
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<RL, RS>), row_grid(R, C, ns),     \
                       dim3(256), 0, stream, (const short*)dyc.data_ptr(),     \
                       (const short*)x.data_ptr(), rp,                         \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),         \
                       weight.data_ptr<float>(), bias.data_ptr<float>(),       \
                       mid.data_ptr<float>(), R, C)
  if (relu && rp) RED(true, true);
  else if (relu) RED(true, false);
  else if (rp) RED(false, true);
  else RED(false, false);
  #undef RED
  hipLaunchKernelGGL(bn_bwd_final_kernel, dim3((C + 15) / 16), dim3(256), 0,
                     stream, mid.data_ptr<float>(), dw.data_ptr<float>(),
                     db.data_ptr<float>(), ns, C);
  #define DXK(RL, RS)                                                          \
    hipLaunchKernelGGL((bn_bwd_dx_kernel<RL, RS>), row_grid(R, C, ns),         \
                       dim3(256), 0, stream, (const short*)dyc.data_ptr(),     \
                       (const short*)x.data_ptr(), rp,                         \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),         \
                       weight.data_ptr<float>(), bias.data_ptr<float>(),       \
                       dw.data_ptr<float>(), db.data_ptr<float>(),             \
                       (short*)dx.data_ptr(), drp, R, C)
  if (relu && rp) DXK(true, true);
  else if (relu) DXK(true, false);
  else if (rp) DXK(false, true);
  else DXK(false, false);
  #undef DXK
  HIP_CHECK(hipGetLastError());
  if (res.has_value()) return {dx, dres, dw, db};
  return {dx, dw, db};
}
