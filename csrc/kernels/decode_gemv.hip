// Decode-regime GEMV: y[B][N] = x[B][K] @ W[N][K]^T (+bias), B <= 8.
//
// The serving decode step is weight-READ-bound (guide Appendix B
// "GEMV / M <= 16 decode weights: operand streamed once ... load
// straight to VGPRs, deep unroll, late vmcnt; the LDS round trip is
// pure overhead").  hipBLASLt on these skinny shapes measured
// ~1.7 TB/s inside the Llama decode step (9.7 ms vs the 2.5 ms
// 16-GB-weight floor); this kernel streams each W row once, coalesced
// bf16x8 per lane, with x pre-upcast to fp32 (L1-resident, re-read per
// row) and per-batch fp32 accumulation.
//
// One wave per output row slice: wave w handles rows n = w, w+NW, ...;
// per row the 64 lanes split K into 512-element chunks, accumulate
// fp32 partials per batch, and wave-reduce at the row end.

#include "../common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

template <int B>
__global__ void __launch_bounds__(256)
decode_gemv_kernel(const float* __restrict__ x,  // [B][K] fp32 (pre-upcast)
                   const short* __restrict__ w,  // [N][K] bf16
                   const short* __restrict__ bias,  // [N] bf16 or null
                   short* __restrict__ y,        // [B][N] bf16
                   int N, int K) {
  const int l = threadIdx.x & 63;
  const int wave = (blockIdx.x * 256 + (int)threadIdx.x) >> 6;
  const int nwaves = (gridDim.x * 256) >> 6;

  // TWO rows per wave iteration with a 4-deep unrolled k-loop: 8
  // independent W loads in flight per lane (the v1 single-row
  // sequential loop was latency-bound — each 512-chunk load waited the
  // previous one's HBM round trip)
  for (int n = wave * 2; n < N; n += nwaves * 2) {
    const int n1 = min(n + 1, N - 1);
    const short* wr0 = w + (long)n * K;
    const short* wr1 = w + (long)n1 * K;
    float acc0[B] = {}, acc1[B] = {};
    #pragma unroll 4
    for (int k0 = l * 8; k0 < K; k0 += 64 * 8) {
      bf16x8_t wv0 = *(const bf16x8_t*)(wr0 + k0);
      bf16x8_t wv1 = *(const bf16x8_t*)(wr1 + k0);
      float wf0[8], wf1[8];
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        wf0[j] = bf2f(wv0[j]);
        wf1[j] = bf2f(wv1[j]);
      }
      #pragma unroll
      for (int b = 0; b < B; ++b) {
        const f32x4_t xlo = *(const f32x4_t*)(x + (long)b * K + k0);
        const f32x4_t xhi = *(const f32x4_t*)(x + (long)b * K + k0 + 4);
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc0[b] = fmaf(wf0[j], xlo[j], acc0[b]);
          acc1[b] = fmaf(wf1[j], xlo[j], acc1[b]);
        }
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          acc0[b] = fmaf(wf0[4 + j], xhi[j], acc0[b]);
          acc1[b] = fmaf(wf1[4 + j], xhi[j], acc1[b]);
        }
      }
    }
    const float bv0 = bias ? bf2f(bias[n]) : 0.f;
    const float bv1 = bias ? bf2f(bias[n1]) : 0.f;
    #pragma unroll
    for (int b = 0; b < B; ++b) {
      const float s0 = wave_reduce_sum(acc0[b]);
      const float s1 = wave_reduce_sum(acc1[b]);
      if (l == 0) {
        y[(long)b * N + n] = f2bf(s0 + bv0);
        if (n1 != n) y[(long)b * N + n1] = f2bf(s1 + bv1);
      }
    }
  }
}

}  // namespace

// x (B, K) bf16 -> y (B, N) bf16; B <= 8, K % 8 == 0.
torch::Tensor decode_gemv(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias) {
  CHECK_CUDA(x); CHECK_CONTIG(x); CHECK_CUDA(w); CHECK_CONTIG(w);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16, "decode_gemv: bf16 only");
  const int B = (int)x.size(0);
  const int K = (int)x.size(1);
  const int N = (int)w.size(0);
  TORCH_CHECK(w.size(1) == K && B >= 1 && B <= 8 && K % 8 == 0,
              "decode_gemv: B<=8, K%8==0 (got B=", B, " K=", K, ")");
  auto xf = x.to(torch::kFloat32);  // tiny; L1-resident during the sweep
  auto y = torch::empty({B, N}, x.options());
  const short* bp = nullptr;
  if (bias.has_value()) {
    CHECK_CONTIG(bias.value());
    bp = (const short*)bias->data_ptr();
  }
  auto stream = scaffold_stream();
  // 4 waves/block, 2 rows per wave iteration
  const int blocks = std::min(2048, (N + 7) / 8);
  #define DG(BB)                                                               \
    hipLaunchKernelGGL((decode_gemv_kernel<BB>), dim3(blocks), dim3(256), 0,   \
                       stream, xf.data_ptr<float>(),                           \
                       (const short*)w.data_ptr(), bp, (short*)y.data_ptr(),   \
                       N, K)
  switch (B) {
    case 1: DG(1); break;
    case 2: DG(2); break;
    case 3: DG(3); break;
    case 4: DG(4); break;
    case 5: DG(5); break;
    case 6: DG(6); break;
    case 7: DG(7); break;
    case 8: DG(8); break;
  }
  #undef DG
  HIP_CHECK(hipGetLastError());
  return y;
}
