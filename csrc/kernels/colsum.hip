// Column sum of a [R][C] bf16 matrix -> [C] (bias gradients in the
// linear backward).  torch's generic reduce ran these at ~1 TB/s; the
// r1 version used scalar 2-B loads and 16 slices and measured ~23% of
// HBM roofline (7% of the GPT-2 step, kstats_gpt2_r2).  This version
// is bf16x8-vectorized (G13) with a chip-filling slice count and a
// parallel 16x16 finalize, the same shape as the fused-BN reduces.

#include "../common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));

// thread -> (vector column, row group); CV = C/8
__global__ void __launch_bounds__(256)
colsum_slice_kernel(const short* __restrict__ x, float* __restrict__ mid,
                    long R, int C) {
  __shared__ float acc[2048];
  const int CV = C / 8;
  int vc, g, G;
  if (CV >= 256) {
    vc = blockIdx.x * 256 + (int)threadIdx.x;
    g = 0;
    G = 1;
    if (vc >= CV) vc = -1;
  } else {
    G = 256 / CV;
    vc = (int)threadIdx.x % CV;
    g = (int)threadIdx.x / CV;
    // when CV does not divide 256 the tail threads would re-walk other
    // groups' rows (double count — caught by test_colsum_vs_fp32 at
    // C=768): idle them instead
    if (g >= G) vc = -1;
  }
  float s[8] = {};
  if (vc >= 0) {
    const int ns = gridDim.y;
    for (long r = (long)blockIdx.y * G + g; r < R; r += (long)ns * G) {
      bf16x8_t xv = *(const bf16x8_t*)(x + r * C + vc * 8);
      #pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf2f(xv[j]);
    }
  }
  float* out = mid + (long)blockIdx.y * C;
  if (G == 1) {
    if (vc >= 0)
      #pragma unroll
      for (int j = 0; j < 8; ++j) out[vc * 8 + j] = s[j];
    return;
  }
  // deterministic fold: per-group partials land in [g][C] slots (G*C
  // is always 2048 floats) and group 0 sums them in FIXED g order —
  // an LDS atomicAdd fold would vary fp32 addition order run to run
  // and break the bitwise-determinism race screen
  if (vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) acc[g * C + vc * 8 + j] = s[j];
  }
  __syncthreads();
  if (g == 0 && vc >= 0) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      float t = 0.f;
      for (int gg = 0; gg < G; ++gg) t += acc[gg * C + vc * 8 + j];
      out[vc * 8 + j] = t;
    }
  }
}

// scalar tail path for C % 8 != 0 (rare; the r1 structure)
__global__ void __launch_bounds__(256)
colsum_slice_scalar_kernel(const short* __restrict__ x,
                           float* __restrict__ mid, long R, int C) {
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
  __shared__ float acc[16][16];
  const int cl = (int)threadIdx.x & 15;
  const int g = (int)threadIdx.x >> 4;
  const int c = blockIdx.x * 16 + cl;
  float s = 0.f;
  if (c < C)
    for (int r = g; r < nrows; r += 16) s += mid[(long)r * C + c];
  acc[g][cl] = s;
  __syncthreads();
  if (g || c >= C) return;
  s = 0.f;
  #pragma unroll
  for (int j = 0; j < 16; ++j) s += acc[j][cl];
  out[c] = f2bf(s);
}

}  // namespace

torch::Tensor colsum(torch::Tensor x) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  TORCH_CHECK(x.dim() == 2 && x.scalar_type() == torch::kBFloat16);
  const long R = x.size(0);
  const int C = (int)x.size(1);
  auto out = torch::empty({C}, x.options());
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto stream = scaffold_stream();
  if (C % 8 == 0) {
    const int CV = C / 8;
    const int cvb = CV >= 256 ? (CV + 255) / 256 : 1;
    const int G = CV >= 256 ? 1 : 256 / CV;
    long ns = 4096 / ((long)G * cvb);
    if (ns > 512) ns = 512;
    const long max_useful = (R + G - 1) / G;
    if (ns > max_useful) ns = max_useful;
    if (ns < 1) ns = 1;
    auto mid = torch::empty({ns, (long)C}, fopt);
    hipLaunchKernelGGL(colsum_slice_kernel, dim3(cvb, (unsigned)ns), dim3(256),
                       0, stream, (const short*)x.data_ptr(),
                       mid.data_ptr<float>(), R, C);
    hipLaunchKernelGGL(colsum_final_kernel, dim3((C + 15) / 16), dim3(256), 0,
                       stream, mid.data_ptr<float>(), (short*)out.data_ptr(),
                       (int)ns, C);
  } else {
    const int NS = R > 512 ? 16 : 1;
    auto mid = torch::empty({NS, (long)C}, fopt);
    hipLaunchKernelGGL(colsum_slice_scalar_kernel, dim3((C + 63) / 64, NS),
                       dim3(256), 0, stream, (const short*)x.data_ptr(),
                       mid.data_ptr<float>(), R, C);
    hipLaunchKernelGGL(colsum_final_kernel, dim3((C + 15) / 16), dim3(256), 0,
                       stream, mid.data_ptr<float>(), (short*)out.data_ptr(),
                       NS, C);
  }
  HIP_CHECK(hipGetLastError());
  return out;
}
