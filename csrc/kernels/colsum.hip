// Column sum of a [R][C] bf16 matrix -> [C] (bias gradients in the
// linear backward).  torch's generic reduce ran these at ~1 TB/s; this
// is a plain coalesced two-stage reduce (lane = one column within a
// 64-col group, rows sliced over gridDim.y to fill the chip, fp32
// accumulation, same shape as the layernorm dgamma reduce).

#include "../common.h"

namespace {

__global__ void __launch_bounds__(256)
colsum_slice_kernel(const short* __restrict__ x, float* __restrict__ mid,
                    long R, int C) {
  __shared__ float acc[4][64];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int col = blockIdx.x * 64 + lane;
  const int ns = gridDim.y;
  float s = 0.f;
  if (col < C)
    for (long r = (long)blockIdx.y * 4 + wv; r < R; r += (long)ns * 4)
      s += bf2f(x[r * C + col]);
  acc[wv][lane] = s;
  __syncthreads();
  if (wv == 0 && col < C)
    mid[(long)blockIdx.y * C + col] =
        acc[0][lane] + acc[1][lane] + acc[2][lane] + acc[3][lane];
}

__global__ void __launch_bounds__(256)
colsum_final_kernel(const float* __restrict__ mid, short* __restrict__ out,
                    int nrows, int C) {
  __shared__ float acc[4][64];
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int col = blockIdx.x * 64 + lane;
  float s = 0.f;
  if (col < C)
    for (int r = wv; r < nrows; r += 4) s += mid[(long)r * C + col];
  acc[wv][lane] = s;
  __syncthreads();
  if (wv == 0 && col < C)
    out[col] = f2bf(acc[0][lane] + acc[1][lane] + acc[2][lane] + acc[3][lane]);
}

}  // namespace

torch::Tensor colsum(torch::Tensor x) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  TORCH_CHECK(x.dim() == 2 && x.scalar_type() == torch::kBFloat16);
  const long R = x.size(0);
  const int C = (int)x.size(1);
  auto out = torch::empty({C}, x.options());
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  const int NS = R > 512 ? 16 : 1;
  auto mid = torch::empty({NS, (long)C}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((C + 63) / 64, NS), block(256);
  hipLaunchKernelGGL(colsum_slice_kernel, grid, block, 0, stream,
                     (const short*)x.data_ptr(), mid.data_ptr<float>(), R, C);
  hipLaunchKernelGGL(colsum_final_kernel, dim3((C + 63) / 64), block, 0,
                     stream, mid.data_ptr<float>(), (short*)out.data_ptr(),
                     NS, C);
  HIP_CHECK(hipGetLastError());
  return out;
}
