// K5 — hand-written MFMA GEMM for the benched transformer shapes
// (capability C5 "attention/matmul path").
//
// Computes the nn.Linear forward  Y[M,N] = X[M,K] @ W[N,K]^T (+ bias)
// with an optionally fused epilogue (tanh-GELU for the GPT-2 MLP,
// residual-add for projection layers).  bf16 in/out, fp32 accumulate.
//
// MI355X-first design (cdna_hip_programming.md §5 "step-3" structure):
//  * TN data flow: BOTH operands are K-major (X rows and W rows are
//    contiguous along K) so A- and B-tiles stage identically — no
//    transpose anywhere.
//  * 128x128 block tile, BK=64, 4 waves (256 thr) in 2x2; each wave owns
//    a 64x64 sub-tile = 4x4 fragments of v_mfma_f32_16x16x32_bf16.
//  * global->LDS staging via __builtin_amdgcn_global_load_lds width=16
//    (the +67% lever — guide §5 common-mistake 1), double-buffered,
//    ONE __syncthreads per K-tile (2-phase minimum recipe, T3).
//  * LDS XOR-swizzle (T2/rule 21): glds writes lane-linear, so the
//    swizzle is applied to the per-lane GLOBAL source address and the
//    same XOR on the ds_read side; the XOR permutes 16B slots WITHIN a
//    128 B row = one cacheline, so coalescing is preserved (the "glds
//    K-tile trap" does not apply at BK=64: row == cacheline).
//    Conflict: <=2-way instead of 8-way on ds_read_b128.
//  * XCD-aware bijective blockIdx swizzle (T1) for L2 tile reuse.
//
// General/odd-shaped GEMMs (e.g. the vocab-sized lm_head) stay on
// hipBLASLt via torch.matmul per SURVEY §2.2 — this kernel targets the
// benched transformer-block shapes, which are all multiples of the tile.
//
// Expected class per the guide's measured ladder: ~870 TF (step-3
// structure at 4096^3); epilogue fusion removes the separate bias+GELU
// elementwise passes and one full activation read (HBM-bound win).

#include "../common.h"

namespace {

typedef short bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

enum Epilogue { EPI_NONE = 0, EPI_GELU = 1, EPI_RESID = 2 };

DEVINL float gelu_tanh(float v) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  return 0.5f * v * (1.0f + tanhf(c * (v + 0.044715f * v * v * v)));
}

// Stage one 128x64 bf16 operand tile (16 KiB = 1024 16-B slots) into LDS
// starting at lds_base, XOR-pre-swizzling the global source (rule 21).
// Each of the 4 waves issues 4 glds instructions (1 KiB each).
DEVINL void stage_tile(const short* __restrict__ gsrc, long ld,  // row stride
                       short* lds_base, int w, int l) {
  #pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int s = (rr * 4 + w) * 64 + l;        // slot this lane fills
    const int row = s >> 3;                     // 8 slots per 128-B row
    const int kb = (s & 7) ^ (row & 7);         // inverse-swizzled source
    glds16(gsrc + (long)row * ld + kb * 8, lds_base + (rr * 4 + w) * 512);
  }
}

template <int EPI, bool HAS_BIAS>
__global__ __launch_bounds__(256) void linear_fwd_kernel(
    const short* __restrict__ X,    // [M][K] bf16
    const short* __restrict__ W,    // [N][K] bf16
    const short* __restrict__ bias, // [N] bf16 or null
    const short* __restrict__ res,  // [M][N] bf16 or null (EPI_RESID)
    short* __restrict__ Y,          // [M][N] bf16
    short* __restrict__ Pre,        // [M][N] bf16 pre-activation (EPI_GELU)
    int M, int N, int K) {
  // one LDS object (guide §5 trap 4a): [buf][A|B][128][64] bf16 = 64 KiB
  __shared__ __attribute__((aligned(16))) short smem[2 * 2 * 128 * 64];

  const int w = threadIdx.x >> 6;   // wave 0..3
  const int l = threadIdx.x & 63;   // lane
  const int wm = w >> 1, wn = w & 1;  // 2x2 wave grid

  // T1: bijective XCD-contiguous remap of the block id
  const unsigned nwg = gridDim.x;
  const unsigned q8 = nwg / 8, r8 = nwg % 8;
  const unsigned xcd = blockIdx.x % 8, idx = blockIdx.x / 8;
  const unsigned wg =
      (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
  const int gn = N / 128;
  const int m0 = (int)(wg / gn) * 128;
  const int n0 = (int)(wg % gn) * 128;

  const int nt = K / 64;
  f32x4_t acc[4][4] = {};

  // prologue: stage tile 0 into buffer 0
  stage_tile(X + (long)m0 * K, K, smem, w, l);
  stage_tile(W + (long)n0 * K, K, smem + 128 * 64, w, l);
  __syncthreads();  // drains glds (vmcnt 0) + barrier

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    short* sA = smem + cur * (2 * 128 * 64);
    short* sB = sA + 128 * 64;
    if (t + 1 < nt) {
      short* nA = smem + (cur ^ 1) * (2 * 128 * 64);
      stage_tile(X + (long)m0 * K + (t + 1) * 64, K, nA, w, l);
      stage_tile(W + (long)n0 * K + (t + 1) * 64, K, nA + 128 * 64, w, l);
    }
    // compute on buf[cur] while next tile's glds are in flight
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8_t af[4], bfr[4];
      const int kb = kk * 4 + (l >> 4);
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int row = wm * 64 + mi * 16 + (l & 15);
        af[mi] = *(const bf16x8_t*)&sA[(row * 8 + (kb ^ (row & 7))) * 8];
      }
      #pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int col = wn * 64 + ni * 16 + (l & 15);
        bfr[ni] = *(const bf16x8_t*)&sB[(col * 8 + (kb ^ (col & 7))) * 8];
      }
      #pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // waves done reading buf[cur]; next glds drained
    cur ^= 1;
  }

  // epilogue: C/D map col=lane&15, row=(lane>>4)*4+r (probe-verified)
  #pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    const int colg = n0 + wn * 64 + ni * 16 + (l & 15);
    const float bv = HAS_BIAS ? bf2f(bias[colg]) : 0.0f;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int rowg = m0 + wm * 64 + mi * 16 + (l >> 4) * 4;
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mi][ni][r] + bv;
        const long off = (long)(rowg + r) * N + colg;
        if (EPI == EPI_GELU) {
          Pre[off] = f2bf(v);
          v = gelu_tanh(v);
        } else if (EPI == EPI_RESID) {
          v += bf2f(res[off]);
        }
        Y[off] = f2bf(v);
      }
    }
  }
}

template <int EPI>
void launch_epi(bool has_bias, dim3 grid, hipStream_t stream,
                const short* X, const short* W, const short* bias,
                const short* res, short* Y, short* Pre, int M, int N, int K) {
  if (has_bias)
    hipLaunchKernelGGL((linear_fwd_kernel<EPI, true>), grid, dim3(256), 0,
                       stream, X, W, bias, res, Y, Pre, M, N, K);
  else
    hipLaunchKernelGGL((linear_fwd_kernel<EPI, false>), grid, dim3(256), 0,
                       stream, X, W, bias, res, Y, Pre, M, N, K);
}

}  // namespace

// Returns {Y} for EPI_NONE/EPI_RESID, {Y, Pre} for EPI_GELU.
std::vector<torch::Tensor> linear_fwd(torch::Tensor x, torch::Tensor w,
                                      c10::optional<torch::Tensor> bias,
                                      long epilogue,
                                      c10::optional<torch::Tensor> residual) {
  CHECK_CUDA(x); CHECK_CONTIG(x); CHECK_CUDA(w); CHECK_CONTIG(w);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 &&
              w.scalar_type() == torch::kBFloat16, "linear_fwd: bf16 only");
  const long K = x.size(-1);
  const long M = x.numel() / K;
  const long N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "weight shape mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
              "linear_fwd tile constraint: M%128,N%128,K%64 == 0 (got ",
              M, "x", N, "x", K, ") — route odd shapes to torch.matmul");

  auto opt = x.options();
  std::vector<long> yshape(x.sizes().begin(), x.sizes().end());
  yshape.back() = N;
  auto y = torch::empty(yshape, opt);
  torch::Tensor pre;
  const short* bptr = nullptr;
  if (bias.has_value()) {
    CHECK_CONTIG(bias.value());
    TORCH_CHECK(bias->scalar_type() == torch::kBFloat16);
    bptr = (const short*)bias->data_ptr();
  }
  const short* rptr = nullptr;
  if (epilogue == 2) {
    TORCH_CHECK(residual.has_value(), "EPI_RESID needs residual tensor");
    CHECK_CONTIG(residual.value());
    rptr = (const short*)residual->data_ptr();
  }
  short* preptr = nullptr;
  if (epilogue == 1) {
    pre = torch::empty(yshape, opt);
    preptr = (short*)pre.data_ptr();
  }

  dim3 grid((M / 128) * (N / 128));
  auto stream = scaffold_stream();
  const short* X = (const short*)x.data_ptr();
  const short* W = (const short*)w.data_ptr();
  short* Y = (short*)y.data_ptr();
  switch (epilogue) {
    case 0: launch_epi<0>(bptr != nullptr, grid, stream, X, W, bptr, rptr, Y, preptr, M, N, K); break;
    case 1: launch_epi<1>(bptr != nullptr, grid, stream, X, W, bptr, rptr, Y, preptr, M, N, K); break;
    case 2: launch_epi<2>(bptr != nullptr, grid, stream, X, W, bptr, rptr, Y, preptr, M, N, K); break;
    default: TORCH_CHECK(false, "bad epilogue ", epilogue);
  }
  HIP_CHECK(hipGetLastError());
  if (epilogue == 1) return {y, pre};
  return {y};
}
