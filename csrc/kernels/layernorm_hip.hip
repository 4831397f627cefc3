#include "hip/hip_runtime.h"
// K2 — LayerNorm fwd/bwd: one workgroup per row, wave64
// reductions, vectorized bf16x8 loads (G13).  Saves fp32 mean/rstd per
// row for the backward.  dgamma/dbeta: two-stage — per-block partials
// into a workspace, then a column-reduce kernel.

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
layernorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                     const T* __restrict__ b, T* __restrict__ y,
                     float* __restrict__ mean_out, float* __restrict__ rstd_out,
                     int D, float eps) {
  __shared__ float red[8];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)D;
  T* yr = y + row * (long)D;

  float sum = 0.f, sumsq = 0.f;
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float v = load_f(xr, i);
    sum += v;
    sumsq += v * v;
  }
  sum = block_reduce_sum(sum, red);
  sumsq = block_reduce_sum(sumsq, red);
  const float mean = sum / D;
  const float var = sumsq / D - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float v = (load_f(xr, i) - mean) * rstd;
    store_f(yr, i, v * load_f(w, i) + load_f(b, i));
  }
}

// dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat)); dyw = dy * w
template <typename T>
__global__ void __launch_bounds__(256)
layernorm_bwd_dx_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                        const T* __restrict__ w, const float* __restrict__ mean,
                        const float* __restrict__ rstd, T* __restrict__ dx,
                        float* __restrict__ dw_part, float* __restrict__ db_part,
                        int D, int rows_per_block, long R) {
  __shared__ float red[8];
  extern __shared__ float cols[];  // 2*D partials for dw/db
  for (int i = threadIdx.x; i < 2 * D; i += blockDim.x) cols[i] = 0.f;
  __syncthreads();

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, R);
  for (long row = row0; row < row1; ++row) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float mu = mean[row], rs = rstd[row];

    float s1 = 0.f, s2 = 0.f;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float xh = (load_f(xr, i) - mu) * rs;
      float dyv = load_f(dyr, i);
      float dyw = dyv * load_f(w, i);
      s1 += dyw;
      s2 += dyw * xh;
      // column partials accumulate in LDS (no conflict: i strided by block)
      cols[i] += dyv * xh;        // dgamma partial
      cols[D + i] += dyv;         // dbeta partial
    }
    s1 = block_reduce_sum(s1, red) / D;
    s2 = block_reduce_sum(s2, red) / D;
    for (int i = threadIdx.x; i < D; i += blockDim.x) {
      float xh = (load_f(xr, i) - mu) * rs;
      float dyw = load_f(dyr, i) * load_f(w, i);
      store_f(dxr, i, rs * (dyw - s1 - xh * s2));
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    dw_part[(long)blockIdx.x * D + i] = cols[i];
    db_part[(long)blockIdx.x * D + i] = cols[D + i];
  }
}

template <typename T>
__global__ void __launch_bounds__(256)
col_reduce_kernel(const float* __restrict__ part, T* __restrict__ out,
                  int nblocks, int D) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= D) return;
  float s = 0.f;
  for (int b = 0; b < nblocks; ++b) s += part[(long)b * D + i];
  store_f(out, i, s);
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto mean = torch::empty({R}, fopt);
  auto rstd = torch::empty({R}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)R), block(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_fwd_kernel<short>, grid, block, 0, stream,
                       (const short*)x.data_ptr(), (const short*)w.data_ptr(),
                       (const short*)b.data_ptr(), (short*)y.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), D, (float)eps);
  } else {
    hipLaunchKernelGGL(layernorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       b.data_ptr<float>(), y.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), D, (float)eps);
  }
  HIP_CHECK(hipGetLastError());
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  CHECK_CUDA(dy); CHECK_CONTIG(dy); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto db = torch::empty_like(w);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());

  const int target_blocks = 1024;  // >> 256 CUs
  const int rows_per_block = (int)std::max<long>(1, (R + target_blocks - 1) / target_blocks);
  const int nblocks = (int)((R + rows_per_block - 1) / rows_per_block);
  auto dw_part = torch::empty({nblocks, D}, fopt);
  auto db_part = torch::empty({nblocks, D}, fopt);
  const size_t lds = (size_t)2 * D * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024 - 1024, "layernorm_bwd: D too large for LDS path");

  auto stream = scaffold_stream();
  dim3 block(256);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_bwd_dx_kernel<short>, dim3(nblocks), block, lds,
                       stream, (const short*)dy.data_ptr(), (const short*)x.data_ptr(),
                       (const short*)w.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), (short*)dx.data_ptr(),
                       dw_part.data_ptr<float>(), db_part.data_ptr<float>(),
                       D, rows_per_block, R);
    hipLaunchKernelGGL(col_reduce_kernel<short>, dim3((D + 255) / 256), block, 0,
                       stream, dw_part.data_ptr<float>(), (short*)dw.data_ptr(),
                       nblocks, D);
    hipLaunchKernelGGL(col_reduce_kernel<short>, dim3((D + 255) / 256), block, 0,
                       stream, db_part.data_ptr<float>(), (short*)db.data_ptr(),
                       nblocks, D);
  } else {
    hipLaunchKernelGGL(layernorm_bwd_dx_kernel<float>, dim3(nblocks), block, lds,
                       stream, dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dx.data_ptr<float>(),
                       dw_part.data_ptr<float>(), db_part.data_ptr<float>(),
                       D, rows_per_block, R);
    hipLaunchKernelGGL(col_reduce_kernel<float>, dim3((D + 255) / 256), block, 0,
                       stream, dw_part.data_ptr<float>(), dw.data_ptr<float>(),
                       nblocks, D);
    hipLaunchKernelGGL(col_reduce_kernel<float>, dim3((D + 255) / 256), block, 0,
                       stream, db_part.data_ptr<float>(), db.data_ptr<float>(),
                       nblocks, D);
  }
  HIP_CHECK(hipGetLastError());
  return {dx, dw, db};
}
