#include "hip/hip_runtime.h"
// K7 — RMSNorm fwd/bwd + RoPE apply (Llama path).
// RMSNorm mirrors the LayerNorm structure (one workgroup per row,
// wave64 reductions); RoPE is a vectorized elementwise rotate with
// HOST-precomputed cos/sin tables (Appendix B: no on-device trig).

#include "../common.h"

namespace {

template <typename T>
DEVINL float load_f(const T* p, long i);
template <> DEVINL float load_f<short>(const short* p, long i) { return bf2f(p[i]); }
template <> DEVINL float load_f<float>(const float* p, long i) { return p[i]; }

template <typename T>
DEVINL void store_f(T* p, long i, float v);
template <> DEVINL void store_f<short>(short* p, long i, float v) { p[i] = f2bf(v); }
template <> DEVINL void store_f<float>(float* p, long i, float v) { p[i] = v; }

template <typename T>
__global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                   T* __restrict__ y, float* __restrict__ rstd_out,
                   int D, float eps) {
  __shared__ float red[8];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)D;
  T* yr = y + row * (long)D;
  float ss = 0.f;
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float v = load_f(xr, i);
    ss += v * v;
  }
  ss = block_reduce_sum(ss, red);
  const float rstd = rsqrtf(ss / D + eps);
  if (threadIdx.x == 0) rstd_out[row] = rstd;
  for (int i = threadIdx.x; i < D; i += blockDim.x)
    store_f(yr, i, load_f(xr, i) * rstd * load_f(w, i));
}

// dx = rstd*dyw - x * rstd^3/D * sum(dyw * x);  dyw = dy*w
template <typename T>
__global__ void __launch_bounds__(256)
rmsnorm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                   const T* __restrict__ w, const float* __restrict__ rstd,
                   T* __restrict__ dx, float* __restrict__ dw_part,
                   int D, int rows_per_block, long R) {
  __shared__ float red[8];
  extern __shared__ float cols[];  // D dgamma partials
  for (int i = threadIdx.x; i < D; i += blockDim.x) cols[i] = 0.f;
  __syncthreads();

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, R);
  for (long row = row0; row < row1; ++row) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float rs = rstd[row];
    float s = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float xv = load_f(xr, i);
      float dyv = load_f(dyr, i);
      s += dyv * load_f(w, i) * xv;
      cols[i] += dyv * xv * rs;  // dgamma partial
    }
    s = block_reduce_sum(s, red);
    const float c = rs * rs * rs * s / D;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float dyw = load_f(dyr, i) * load_f(w, i);
      store_f(dxr, i, rs * dyw - load_f(xr, i) * c);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x)
    dw_part[(long)blockIdx.x * D + i] = cols[i];
}

template <typename T>
__global__ void __launch_bounds__(256)
col_reduce1_kernel(const float* __restrict__ part, T* __restrict__ out,
                   int nblocks, int D) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= D) return;
  float s = 0.f;
  for (int b = 0; b < nblocks; ++b) s += part[(long)b * D + i];
  store_f(out, i, s);
}

// x: (B,H,S,D) contiguous; cos/sin: (S, D/2) fp32.  Rotates interleaved
// pairs (x[2i], x[2i+1]); inverse applies the transposed rotation (bwd).
template <typename T>
__global__ void __launch_bounds__(256)
rope_kernel(const T* __restrict__ x, const float* __restrict__ cos_t,
            const float* __restrict__ sin_t, T* __restrict__ y,
            long BH, int S, int D, int inverse) {
  const int half = D / 2;
  const long total = BH * (long)S * half;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += stride) {
    const int hpos = (int)(idx % half);
    const long srow = idx / half;          // (bh, s) flattened
    const int spos = (int)(srow % S);
    const long base = srow * (long)D + 2 * hpos;
    float c = cos_t[(long)spos * half + hpos];
    float s = sin_t[(long)spos * half + hpos];
    if (inverse) s = -s;
    const float x1 = load_f(x, base);
    const float x2 = load_f(x, base + 1);
    store_f(y, base, x1 * c - x2 * s);
    store_f(y, base + 1, x1 * s + x2 * c);
  }
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto rstd = torch::empty({R}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)R), block(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<short>, grid, block, 0, stream,
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       (short*)y.data_ptr(), rstd.data_ptr<float>(), D, (float)eps);
  } else {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       y.data_ptr<float>(), rstd.data_ptr<float>(), D, (float)eps);
  }
  HIP_CHECK(hipGetLastError());
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  CHECK_CUDA(dy); CHECK_CONTIG(dy); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  const int target_blocks = 1024;
  const int rows_per_block = (int)std::max<long>(1, (R + target_blocks - 1) / target_blocks);
  const int nblocks = (int)((R + rows_per_block - 1) / rows_per_block);
  auto dw_part = torch::empty({nblocks, D}, fopt);
  const size_t lds = (size_t)D * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024 - 1024, "rmsnorm_bwd: D too large for LDS path");
  auto stream = scaffold_stream();
  dim3 block(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<short>, dim3(nblocks), block, lds, stream,
                       (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                       (const short*)w.data_ptr(), rstd.data_ptr<float>(),
                       (short*)dx.data_ptr(), dw_part.data_ptr<float>(),
                       D, rows_per_block, R);
    hipLaunchKernelGGL(col_reduce1_kernel<short>, dim3((D + 255) / 256), block, 0,
                       stream, dw_part.data_ptr<float>(), (short*)dw.data_ptr(),
                       nblocks, D);
  } else {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<float>, dim3(nblocks), block, lds, stream,
                       dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), rstd.data_ptr<float>(),
                       dx.data_ptr<float>(), dw_part.data_ptr<float>(),
                       D, rows_per_block, R);
    hipLaunchKernelGGL(col_reduce1_kernel<float>, dim3((D + 255) / 256), block, 0,
                       stream, dw_part.data_ptr<float>(), dw.data_ptr<float>(),
                       nblocks, D);
  }
  HIP_CHECK(hipGetLastError());
  return {dx, dw};
}

torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t,
                         bool inverse) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  TORCH_CHECK(x.dim() == 4, "rope expects (B,H,S,D)");
  const long BH = x.size(0) * x.size(1);
  const int S = (int)x.size(2);
  const int D = (int)x.size(3);
  TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "rope table shape");
  auto y = torch::empty_like(x);
  auto stream = scaffold_stream();
  const long total = BH * (long)S * (D / 2);
  dim3 grid(capped_blocks(total, 256, 2048)), block(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rope_kernel<short>, grid, block, 0, stream,
                       (const short*)x.data_ptr(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), (short*)y.data_ptr(),
                       BH, S, D, inverse ? 1 : 0);
  } else {
    hipLaunchKernelGGL(rope_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), y.data_ptr<float>(),
                       BH, S, D, inverse ? 1 : 0);
  }
  HIP_CHECK(hipGetLastError());
  return y;
}
