// K4 — flash-style causal attention forward (correctness-first tile
// version; the MFMA 8-wave structure of cdna_hip_programming.md §B
// replaces the inner loops in a later milestone — same launch contract).
//
// Layout: q (B,H,S,D), k/v (B,Hkv,S,D) row-major bf16/fp32; GQA via
// Hkv | H.  One 256-thread workgroup per (b, h, q-tile of QT rows);
// K/V tiles staged in LDS; online softmax (m, l) per q row; saves
// per-row LSE for the backward (externalized state — SURVEY §5.7c).

#include "../common.h"

namespace {

constexpr int QT = 16;   // q rows per block
constexpr int KC = 64;   // kv chunk

template <typename T>
DEVINL float load_f(const T* p, long i);
template <> DEVINL float load_f<short>(const short* p, long i) { return bf2f(p[i]); }
template <> DEVINL float load_f<float>(const float* p, long i) { return p[i]; }

template <typename T>
DEVINL void store_f(T* p, long i, float v);
template <> DEVINL void store_f<short>(short* p, long i, float v) { p[i] = f2bf(v); }
template <> DEVINL void store_f<float>(float* p, long i, float v) { p[i] = v; }

template <typename T, int D>
__global__ void __launch_bounds__(256)
attn_fwd_kernel(const T* __restrict__ q, const T* __restrict__ k,
                const T* __restrict__ v, T* __restrict__ o,
                float* __restrict__ lse, int B, int H, int Hkv, int S,
                float scale, int causal) {
  __shared__ float s_q[QT][D];
  __shared__ float s_k[KC][D];
  __shared__ float s_v[KC][D];
  __shared__ float s_p[QT][KC];
  __shared__ float s_o[QT][D];
  __shared__ float s_m[QT], s_l[QT], s_alpha[QT];

  const int bh = blockIdx.y;            // b*H + h
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = blockIdx.x * QT;
  if (q0 >= S) return;

  const long qbase = (((long)b * H + h) * S + q0) * D;
  const long kbase = (((long)b * Hkv + hk) * S) * D;
  const int tid = threadIdx.x;

  for (int i = tid; i < QT * D; i += 256) {
    const int qi = i / D, d = i % D;
    s_q[qi][d] = (q0 + qi < S) ? load_f(q, qbase + (long)qi * D + d) : 0.f;
    s_o[qi][d] = 0.f;
  }
  if (tid < QT) { s_m[tid] = -INFINITY; s_l[tid] = 0.f; }
  __syncthreads();

  const int kv_end = causal ? min(S, q0 + QT) : S;
  for (int c0 = 0; c0 < kv_end; c0 += KC) {
    const int cn = min(KC, kv_end - c0);
    for (int i = tid; i < KC * D; i += 256) {
      const int j = i / D, d = i % D;
      const long src = kbase + (long)(c0 + j) * D + d;
      const bool ok = j < cn;
      s_k[j][d] = ok ? load_f(k, src) : 0.f;
      s_v[j][d] = ok ? load_f(v, src) : 0.f;
    }
    __syncthreads();

    for (int i = tid; i < QT * KC; i += 256) {
      const int qi = i / KC, j = i % KC;
      float acc = 0.f;
      #pragma unroll 8
      for (int d = 0; d < D; ++d) acc += s_q[qi][d] * s_k[j][d];
      acc *= scale;
      const bool masked =
          j >= cn || (causal && (c0 + j) > (q0 + qi)) || (q0 + qi >= S);
      s_p[qi][j] = masked ? -INFINITY : acc;
    }
    __syncthreads();

    // per-row max + rescale factor (thread qi owns row qi)
    if (tid < QT) {
      float mrow = s_m[tid];
      #pragma unroll 4
      for (int j = 0; j < KC; ++j) mrow = fmaxf(mrow, s_p[tid][j]);
      const float alpha =
          (s_m[tid] == -INFINITY || mrow == -INFINITY) ? 0.f : __expf(s_m[tid] - mrow);
      s_alpha[tid] = alpha;
      s_m[tid] = mrow;
      s_l[tid] *= alpha;
    }
    __syncthreads();

    for (int i = tid; i < QT * KC; i += 256) {
      const int qi = i / KC, j = i % KC;
      const float m = s_m[qi];
      const float sv = s_p[qi][j];
      s_p[qi][j] = (sv == -INFINITY || m == -INFINITY) ? 0.f : __expf(sv - m);
    }
    for (int i = tid; i < QT * D; i += 256) {
      const int qi = i / D, d = i % D;
      s_o[qi][d] *= s_alpha[qi];
    }
    __syncthreads();

    if (tid < QT) {
      float srow = 0.f;
      #pragma unroll 4
      for (int j = 0; j < KC; ++j) srow += s_p[tid][j];
      s_l[tid] += srow;
    }
    for (int i = tid; i < QT * D; i += 256) {
      const int qi = i / D, d = i % D;
      float acc = 0.f;
      #pragma unroll 8
      for (int j = 0; j < KC; ++j) acc += s_p[qi][j] * s_v[j][d];
      s_o[qi][d] += acc;
    }
    __syncthreads();
  }

  for (int i = tid; i < QT * D; i += 256) {
    const int qi = i / D, d = i % D;
    if (q0 + qi < S) {
      const float l = s_l[qi];
      store_f(o, qbase + (long)qi * D + d, l > 0.f ? s_o[qi][d] / l : 0.f);
    }
  }
  if (tid < QT && q0 + tid < S) {
    const long r = ((long)b * H + h) * S + q0 + tid;
    lse[r] = s_m[tid] + __logf(fmaxf(s_l[tid], 1e-30f));
  }
}

template <typename T>
void launch_fwd(const torch::Tensor& q, const torch::Tensor& k,
                const torch::Tensor& v, torch::Tensor& o, torch::Tensor& lse,
                int B, int H, int Hkv, int S, int D, float scale, bool causal,
                hipStream_t stream) {
  dim3 grid((S + QT - 1) / QT, B * H), block(256);
  #define LAUNCH_D(DD)                                                         \
    hipLaunchKernelGGL((attn_fwd_kernel<T, DD>), grid, block, 0, stream,       \
                       (const T*)q.data_ptr(), (const T*)k.data_ptr(),         \
                       (const T*)v.data_ptr(), (T*)o.data_ptr(),               \
                       lse.data_ptr<float>(), B, H, Hkv, S, scale,             \
                       causal ? 1 : 0)
  if (D == 64) LAUNCH_D(64);
  else if (D == 128) LAUNCH_D(128);
  else if (D == 32) LAUNCH_D(32);
  else TORCH_CHECK(false, "attn_fwd: unsupported head dim ", D);
  #undef LAUNCH_D
}

}  // namespace

// MFMA fast paths: 4-wave/128-row (attn_fwd_mfma.hip) and 8-wave/256-row
// (attn_fwd_8w.hip); bf16, D in {64,128}
void attn_fwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                   torch::Tensor o, torch::Tensor lse, int B, int H, int Hkv,
                   int S, int D, double scale, bool causal);
void attn_fwd_8w(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor o, torch::Tensor lse, int B, int H, int Hkv,
                 int S, int D, double scale, bool causal);

namespace {
// SCAFFOLD_ATTN_FWD = 4w | 8w | auto (default).  auto: the 8-wave
// kernel for q blocks it can fill (S >= 256), 4-wave otherwise.
// Read per call (cheap vs the kernel) so tests can exercise both
// kernels in one process.
int attn_fwd_mode() {
  const char* e = getenv("SCAFFOLD_ATTN_FWD");
  if (e && strcmp(e, "4w") == 0) return 4;
  if (e && strcmp(e, "8w") == 0) return 8;
  return 0;
}
}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal) {
  CHECK_CUDA(q);
  const int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(1);
  TORCH_CHECK(H % Hkv == 0, "GQA: H must be a multiple of Hkv");
  auto lse = torch::empty({B, H, S},
      torch::TensorOptions().dtype(torch::kFloat32).device(q.device()));
  const float scale = 1.0f / std::sqrt((float)D);
  auto stream = scaffold_stream();
  if (q.scalar_type() == torch::kBFloat16 && (D == 64 || D == 128)) {
    // strided MFMA path: consume permuted views directly; O written in
    // (B,S,H,D) physical layout so the caller's transpose+reshape is free
    auto qc = q.stride(3) == 1 ? q : q.contiguous();
    auto kc = k.stride(3) == 1 ? k : k.contiguous();
    auto vc = v.stride(3) == 1 ? v : v.contiguous();
    auto o = torch::empty({B, S, H, D}, q.options()).permute({0, 2, 1, 3});
    const int mode = attn_fwd_mode();
    const bool use8w = (mode == 8) || (mode == 0 && S >= 256);
    if (use8w)
      attn_fwd_8w(qc, kc, vc, o, lse, B, H, Hkv, S, D, scale, causal);
    else
      attn_fwd_mfma(qc, kc, vc, o, lse, B, H, Hkv, S, D, scale, causal);
    HIP_CHECK(hipGetLastError());
    return {o, lse};
  }
  auto qc = q.contiguous();
  auto kc = k.contiguous();
  auto vc = v.contiguous();
  auto o = torch::empty_like(qc);
  if (q.scalar_type() == torch::kBFloat16) {
    launch_fwd<short>(qc, kc, vc, o, lse, B, H, Hkv, S, D, scale, causal, stream);
  } else {
    launch_fwd<float>(qc, kc, vc, o, lse, B, H, Hkv, S, D, scale, causal, stream);
  }
  HIP_CHECK(hipGetLastError());
  return {o, lse};
}
