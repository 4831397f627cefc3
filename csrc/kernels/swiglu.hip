// SwiGLU elementwise fusion for the Llama MLP:
//   fwd: h = silu(g) * u            (one pass: read g,u -> write h)
//   bwd: dg = dh * u * silu'(g)     (one pass: read dh,g,u -> write dg,du)
//        du = dh * silu(g)
// vs eager (separate silu + mul + their backwards) this halves the HBM
// passes over the (M, d_ff) activations.  Pure-bandwidth kernel:
// bf16x8 vector loads (G13), grid-stride, fp32 math.

#include "../common.h"

namespace {

DEVINL float sigm(float x) { return 1.0f / (1.0f + __expf(-x)); }

__global__ void __launch_bounds__(256)
swiglu_fwd_kernel(const short* __restrict__ g, const short* __restrict__ u,
                  short* __restrict__ h, long n8) {
  const long stride = grid_stride();
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8; i += stride) {
    bf16x8 gv = *(const bf16x8*)(g + i * 8);
    bf16x8 uv = *(const bf16x8*)(u + i * 8);
    bf16x8 out;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(gv[j]);
      out[j] = f2bf(x * sigm(x) * bf2f(uv[j]));
    }
    *(bf16x8*)(h + i * 8) = out;
  }
}

__global__ void __launch_bounds__(256)
swiglu_bwd_kernel(const short* __restrict__ dh, const short* __restrict__ g,
                  const short* __restrict__ u, short* __restrict__ dg,
                  short* __restrict__ du, long n8) {
  const long stride = grid_stride();
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n8; i += stride) {
    bf16x8 dhv = *(const bf16x8*)(dh + i * 8);
    bf16x8 gv = *(const bf16x8*)(g + i * 8);
    bf16x8 uv = *(const bf16x8*)(u + i * 8);
    bf16x8 dgv, duv;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(gv[j]);
      const float s = sigm(x);
      const float si = x * s;                  // silu(g)
      const float dsi = s + si * (1.0f - s);   // silu'(g)
      const float d = bf2f(dhv[j]);
      dgv[j] = f2bf(d * bf2f(uv[j]) * dsi);
      duv[j] = f2bf(d * si);
    }
    *(bf16x8*)(dg + i * 8) = dgv;
    *(bf16x8*)(du + i * 8) = duv;
  }
}

// packed variant: gate|up live in ONE [M][2F] buffer (the fused
// gate+up GEMM's output); h = silu(gu[:, :F]) * gu[:, F:].
__global__ void __launch_bounds__(256)
swiglu_packed_fwd_kernel(const short* __restrict__ gu, short* __restrict__ h,
                         long rows, int F8) {
  const long stride = grid_stride();
  const long total = rows * F8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const long row = i / F8;
    const int c8 = (int)(i % F8);
    const short* gp = gu + row * (2L * 8 * F8) + c8 * 8;
    bf16x8 gv = *(const bf16x8*)gp;
    bf16x8 uv = *(const bf16x8*)(gp + 8L * F8);
    bf16x8 out;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(gv[j]);
      out[j] = f2bf(x * sigm(x) * bf2f(uv[j]));
    }
    *(bf16x8*)(h + row * (8L * F8) + c8 * 8) = out;
  }
}

__global__ void __launch_bounds__(256)
swiglu_packed_bwd_kernel(const short* __restrict__ dh,
                         const short* __restrict__ gu,
                         short* __restrict__ dgu, long rows, int F8) {
  const long stride = grid_stride();
  const long total = rows * F8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const long row = i / F8;
    const int c8 = (int)(i % F8);
    const short* gp = gu + row * (2L * 8 * F8) + c8 * 8;
    bf16x8 gv = *(const bf16x8*)gp;
    bf16x8 uv = *(const bf16x8*)(gp + 8L * F8);
    bf16x8 dhv = *(const bf16x8*)(dh + row * (8L * F8) + c8 * 8);
    bf16x8 dgv, duv;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(gv[j]);
      const float s = sigm(x);
      const float si = x * s;
      const float dsi = s + si * (1.0f - s);
      const float d = bf2f(dhv[j]);
      dgv[j] = f2bf(d * bf2f(uv[j]) * dsi);
      duv[j] = f2bf(d * si);
    }
    short* op = dgu + row * (2L * 8 * F8) + c8 * 8;
    *(bf16x8*)op = dgv;
    *(bf16x8*)(op + 8L * F8) = duv;
  }
}

}  // namespace

torch::Tensor swiglu_packed_fwd(torch::Tensor gu) {
  CHECK_CUDA(gu); CHECK_CONTIG(gu);
  TORCH_CHECK(gu.scalar_type() == torch::kBFloat16);
  const long F = gu.size(-1) / 2;
  TORCH_CHECK(F % 8 == 0, "swiglu_packed: d_ff % 8");
  const long rows = gu.numel() / (2 * F);
  std::vector<long> shape(gu.sizes().begin(), gu.sizes().end());
  shape.back() = F;
  auto h = torch::empty(shape, gu.options());
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(swiglu_packed_fwd_kernel, dim3(capped_blocks(rows * (F / 8))),
                     dim3(256), 0, stream, (const short*)gu.data_ptr(),
                     (short*)h.data_ptr(), rows, (int)(F / 8));
  HIP_CHECK(hipGetLastError());
  return h;
}

torch::Tensor swiglu_packed_bwd(torch::Tensor dh, torch::Tensor gu) {
  CHECK_CUDA(dh); CHECK_CONTIG(dh); CHECK_CONTIG(gu);
  const long F = gu.size(-1) / 2;
  const long rows = gu.numel() / (2 * F);
  auto dgu = torch::empty_like(gu);
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(swiglu_packed_bwd_kernel, dim3(capped_blocks(rows * (F / 8))),
                     dim3(256), 0, stream, (const short*)dh.data_ptr(),
                     (const short*)gu.data_ptr(), (short*)dgu.data_ptr(),
                     rows, (int)(F / 8));
  HIP_CHECK(hipGetLastError());
  return dgu;
}

torch::Tensor swiglu_fwd(torch::Tensor g, torch::Tensor u) {
  CHECK_CUDA(g); CHECK_CONTIG(g); CHECK_CONTIG(u);
  TORCH_CHECK(g.scalar_type() == torch::kBFloat16 && g.numel() % 8 == 0);
  auto h = torch::empty_like(g);
  const long n8 = g.numel() / 8;
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(capped_blocks(n8)), dim3(256), 0,
                     stream, (const short*)g.data_ptr(),
                     (const short*)u.data_ptr(), (short*)h.data_ptr(), n8);
  HIP_CHECK(hipGetLastError());
  return h;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dh, torch::Tensor g,
                                      torch::Tensor u) {
  CHECK_CUDA(dh); CHECK_CONTIG(dh); CHECK_CONTIG(g); CHECK_CONTIG(u);
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  const long n8 = g.numel() / 8;
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(capped_blocks(n8)), dim3(256), 0,
                     stream, (const short*)dh.data_ptr(),
                     (const short*)g.data_ptr(), (const short*)u.data_ptr(),
                     (short*)dg.data_ptr(), (short*)du.data_ptr(), n8);
  HIP_CHECK(hipGetLastError());
  return {dg, du};
}
