// K4 — flash attention backward (correctness-first tile version).
// Standard flash bwd: delta = rowsum(dO*O); recompute P from Q,K + saved
// LSE per tile; dS = P * (dP - delta) * scale; three kernels:
//   1) delta          2) dQ (per q-tile over kv)   3) dK/dV (per kv-tile over q)

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

constexpr int QT = 16;
constexpr int KC = 64;
constexpr int KT = 16;
constexpr int QC = 64;

template <typename T, int D>
__global__ void __launch_bounds__(256)
attn_delta_kernel(const T* __restrict__ dout, const T* __restrict__ o,
                  const float* __restrict__ dlse,
                  float* __restrict__ delta, long rows) {
  // one wave per row: rowsum(do * o) over D
  const long row = (long)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const long base = row * D;
  float acc = 0.f;
  for (int d = lane; d < D; d += 64)
    acc += load_f(dout, base + d) * load_f(o, base + d);
  acc = wave_reduce_sum(acc);
  if (lane == 0) delta[row] = acc - (dlse ? dlse[row] : 0.f);
}

template <typename T, int D>
__global__ void __launch_bounds__(256)
attn_bwd_dq_kernel(const T* __restrict__ dout, const T* __restrict__ q,
                   const T* __restrict__ k, const T* __restrict__ v,
                   const float* __restrict__ lse, const float* __restrict__ delta,
                   T* __restrict__ dq, int B, int H, int Hkv, int S,
                   float scale, int causal) {
  __shared__ float s_q[QT][D], s_do[QT][D], s_dq[QT][D];
  __shared__ float s_k[KC][D], s_v[KC][D];
  __shared__ float s_p[QT][KC], s_ds[QT][KC];
  __shared__ float s_lse[QT], s_delta[QT];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = blockIdx.x * QT;
  if (q0 >= S) return;
  const long qbase = (((long)b * H + h) * S + q0) * D;
  const long kbase = (((long)b * Hkv + hk) * S) * D;
  const long rbase = ((long)b * H + h) * S + q0;
  const int tid = threadIdx.x;

  for (int i = tid; i < QT * D; i += 256) {
    const int qi = i / D, d = i % D;
    const bool ok = q0 + qi < S;
    s_q[qi][d] = ok ? load_f(q, qbase + (long)qi * D + d) : 0.f;
    s_do[qi][d] = ok ? load_f(dout, qbase + (long)qi * D + d) : 0.f;
    s_dq[qi][d] = 0.f;
  }
  if (tid < QT) {
    const bool ok = q0 + tid < S;
    s_lse[tid] = ok ? lse[rbase + tid] : 0.f;
    s_delta[tid] = ok ? delta[rbase + tid] : 0.f;
  }
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
      const bool masked =
          j >= cn || (causal && (c0 + j) > (q0 + qi)) || (q0 + qi >= S);
      float s = 0.f, dp = 0.f;
      if (!masked) {
        #pragma unroll 8
        for (int d = 0; d < D; ++d) {
          s += s_q[qi][d] * s_k[j][d];
          dp += s_do[qi][d] * s_v[j][d];
        }
      }
      const float p = masked ? 0.f : __expf(s * scale - s_lse[qi]);
      s_p[qi][j] = p;
      s_ds[qi][j] = p * (dp - s_delta[qi]) * scale;
    }
    __syncthreads();

    for (int i = tid; i < QT * D; i += 256) {
      const int qi = i / D, d = i % D;
      float acc = 0.f;
      #pragma unroll 8
      for (int j = 0; j < KC; ++j) acc += s_ds[qi][j] * s_k[j][d];
      s_dq[qi][d] += acc;
    }
    __syncthreads();
  }

  for (int i = tid; i < QT * D; i += 256) {
    const int qi = i / D, d = i % D;
    if (q0 + qi < S) store_f(dq, qbase + (long)qi * D + d, s_dq[qi][d]);
  }
}

template <typename T, int D>
__global__ void __launch_bounds__(256)
attn_bwd_dkv_kernel(const T* __restrict__ dout, const T* __restrict__ q,
                    const T* __restrict__ k, const T* __restrict__ v,
                    const float* __restrict__ lse, const float* __restrict__ delta,
                    T* __restrict__ dk, T* __restrict__ dv, int B, int H,
                    int Hkv, int S, float scale, int causal) {
  __shared__ float s_k[KT][D], s_v[KT][D], s_dk[KT][D], s_dv[KT][D];
  __shared__ float s_q[QC][D], s_do[QC][D];
  __shared__ float s_p[QC][KT], s_ds[QC][KT];
  __shared__ float s_lse[QC], s_delta[QC];

  const int bhk = blockIdx.y;           // b*Hkv + hk
  const int b = bhk / Hkv, hk = bhk % Hkv;
  const int group = H / Hkv;
  const int k0 = blockIdx.x * KT;
  if (k0 >= S) return;
  const long kvbase = (((long)b * Hkv + hk) * S + k0) * D;
  const int tid = threadIdx.x;

  for (int i = tid; i < KT * D; i += 256) {
    const int j = i / D, d = i % D;
    const bool ok = k0 + j < S;
    s_k[j][d] = ok ? load_f(k, kvbase + (long)j * D + d) : 0.f;
    s_v[j][d] = ok ? load_f(v, kvbase + (long)j * D + d) : 0.f;
    s_dk[j][d] = 0.f;
    s_dv[j][d] = 0.f;
  }
  __syncthreads();

  for (int g = 0; g < group; ++g) {
    const int h = hk * group + g;
    const long qhead = (((long)b * H + h) * S) * D;
    const long rhead = ((long)b * H + h) * S;
    const int qstart = causal ? (k0 / QC) * QC : 0;
    for (int c0 = qstart; c0 < S; c0 += QC) {
      const int cn = min(QC, S - c0);
      for (int i = tid; i < QC * D; i += 256) {
        const int qi = i / D, d = i % D;
        const bool ok = qi < cn;
        s_q[qi][d] = ok ? load_f(q, qhead + (long)(c0 + qi) * D + d) : 0.f;
        s_do[qi][d] = ok ? load_f(dout, qhead + (long)(c0 + qi) * D + d) : 0.f;
      }
      if (tid < QC) {
        const bool ok = tid < cn;
        s_lse[tid] = ok ? lse[rhead + c0 + tid] : 0.f;
        s_delta[tid] = ok ? delta[rhead + c0 + tid] : 0.f;
      }
      __syncthreads();

      for (int i = tid; i < QC * KT; i += 256) {
        const int qi = i / KT, j = i % KT;
        const bool masked =
            qi >= cn || k0 + j >= S || (causal && (k0 + j) > (c0 + qi));
        float s = 0.f, dp = 0.f;
        if (!masked) {
          #pragma unroll 8
          for (int d = 0; d < D; ++d) {
            s += s_q[qi][d] * s_k[j][d];
            dp += s_do[qi][d] * s_v[j][d];
          }
        }
        const float p = masked ? 0.f : __expf(s * scale - s_lse[qi]);
        s_p[qi][j] = p;
        s_ds[qi][j] = p * (dp - s_delta[qi]) * scale;
      }
      __syncthreads();

      for (int i = tid; i < KT * D; i += 256) {
        const int j = i / D, d = i % D;
        float adv = 0.f, adk = 0.f;
        #pragma unroll 8
        for (int qi = 0; qi < QC; ++qi) {
          adv += s_p[qi][j] * s_do[qi][d];
          adk += s_ds[qi][j] * s_q[qi][d];
        }
        s_dv[j][d] += adv;
        s_dk[j][d] += adk;
      }
      __syncthreads();
    }
  }

  for (int i = tid; i < KT * D; i += 256) {
    const int j = i / D, d = i % D;
    if (k0 + j < S) {
      store_f(dk, kvbase + (long)j * D + d, s_dk[j][d]);
      store_f(dv, kvbase + (long)j * D + d, s_dv[j][d]);
    }
  }
}

template <typename T>
void launch_bwd(const torch::Tensor& dout, const torch::Tensor& q,
                const torch::Tensor& k, const torch::Tensor& v,
                const torch::Tensor& o, const torch::Tensor& lse,
                torch::Tensor& dq, torch::Tensor& dk, torch::Tensor& dv,
                torch::Tensor& delta, int B, int H, int Hkv, int S, int D,
                float scale, bool causal, hipStream_t stream,
                const float* dlse_ptr) {
  const long rows = (long)B * H * S;
  dim3 dblock(256), dgrid((unsigned)((rows + 3) / 4));
  dim3 block(256);
  dim3 grid_q((S + QT - 1) / QT, B * H);
  dim3 grid_kv((S + KT - 1) / KT, B * Hkv);
  #define LAUNCH_D(DD)                                                          \
    do {                                                                        \
      hipLaunchKernelGGL((attn_delta_kernel<T, DD>), dgrid, dblock, 0, stream,  \
                         (const T*)dout.data_ptr(), (const T*)o.data_ptr(),     \
                         dlse_ptr, delta.data_ptr<float>(), rows);                        \
      hipLaunchKernelGGL((attn_bwd_dq_kernel<T, DD>), grid_q, block, 0, stream, \
                         (const T*)dout.data_ptr(), (const T*)q.data_ptr(),     \
                         (const T*)k.data_ptr(), (const T*)v.data_ptr(),        \
                         lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                         (T*)dq.data_ptr(), B, H, Hkv, S, scale, causal);       \
      hipLaunchKernelGGL((attn_bwd_dkv_kernel<T, DD>), grid_kv, block, 0,       \
                         stream, (const T*)dout.data_ptr(),                     \
                         (const T*)q.data_ptr(), (const T*)k.data_ptr(),        \
                         (const T*)v.data_ptr(), lse.data_ptr<float>(),         \
                         delta.data_ptr<float>(), (T*)dk.data_ptr(),            \
                         (T*)dv.data_ptr(), B, H, Hkv, S, scale, causal);       \
    } while (0)
  if (D == 64) LAUNCH_D(64);
  else if (D == 128) LAUNCH_D(128);
  else if (D == 32) LAUNCH_D(32);
  else TORCH_CHECK(false, "attn_bwd: unsupported head dim ", D);
  #undef LAUNCH_D
}

}  // namespace

// MFMA fast path (attn_bwd_mfma.hip): bf16, D in {64,128}
void attn_bwd_mfma(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                   torch::Tensor delta, torch::Tensor dq, torch::Tensor dk,
                   torch::Tensor dv, int B, int H, int Hkv, int S, int D,
                   double scale, bool causal, const float* dlse_ptr);

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal,
                                    c10::optional<torch::Tensor> dlse) {
  const float* dlse_ptr = nullptr;
  torch::Tensor dlse_c;
  if (dlse.has_value()) {
    dlse_c = dlse->contiguous();
    TORCH_CHECK(dlse_c.scalar_type() == torch::kFloat32 &&
                dlse_c.numel() == lse.numel(), "attn_bwd: bad dlse");
    dlse_ptr = dlse_c.data_ptr<float>();
  }
  CHECK_CUDA(dout);
  const int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2),
            D = (int)q.size(3);
  const int Hkv = (int)k.size(1);
  auto delta = torch::empty({(long)B * H * S},
      torch::TensorOptions().dtype(torch::kFloat32).device(q.device()));
  const float scale = 1.0f / std::sqrt((float)D);
  auto stream = scaffold_stream();
  if (q.scalar_type() == torch::kBFloat16 && (D == 64 || D == 128)) {
    // strided MFMA path; grads produced in (B,S,H,D) physical layout so
    // the upstream view backward (QKV un-transpose) is coalesced
    auto doc = dout.stride(3) == 1 ? dout : dout.contiguous();
    auto qc = q.stride(3) == 1 ? q : q.contiguous();
    auto kc = k.stride(3) == 1 ? k : k.contiguous();
    auto vc = v.stride(3) == 1 ? v : v.contiguous();
    auto oc = o.stride(3) == 1 ? o : o.contiguous();
    auto dq = torch::empty({B, S, H, D}, q.options()).permute({0, 2, 1, 3});
    auto dk = torch::empty({B, S, Hkv, D}, q.options()).permute({0, 2, 1, 3});
    auto dv = torch::empty({B, S, Hkv, D}, q.options()).permute({0, 2, 1, 3});
    attn_bwd_mfma(doc, qc, kc, vc, oc, lse, delta, dq, dk, dv, B, H, Hkv, S, D,
                  scale, causal, dlse_ptr);
    HIP_CHECK(hipGetLastError());
    return {dq, dk, dv};
  }
  auto doc = dout.contiguous();
  auto qc = q.contiguous();
  auto kc = k.contiguous();
  auto vc = v.contiguous();
  auto oc = o.contiguous();
  auto dq = torch::empty_like(qc);
  auto dk = torch::empty_like(kc);
  auto dv = torch::empty_like(vc);
  if (q.scalar_type() == torch::kBFloat16)
    launch_bwd<short>(doc, qc, kc, vc, oc, lse, dq, dk, dv, delta, B, H, Hkv, S, D,
                      scale, causal, stream, dlse_ptr);
  else
    launch_bwd<float>(doc, qc, kc, vc, oc, lse, dq, dk, dv, delta, B, H, Hkv, S, D,
                      scale, causal, stream, dlse_ptr);
  HIP_CHECK(hipGetLastError());
  return {dq, dk, dv};
}
