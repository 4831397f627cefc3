#include "hip/hip_runtime.h"
// K4 — MFMA flash attention forward (bf16, CDNA4 32x32x16 matrix cores).
//
// Structure (cdna_hip_programming.md §B "fused attention prefill", adapted):
//  * 4 waves / workgroup, each wave owns 32 q rows INDEPENDENTLY (no LDS,
//    no barriers — K/V tiles are L2-resident at these S, and LDS staging
//    of L2-fit data measured as pure overhead: guide common-mistake #7).
//  * swapped QK^T: S^T = mfma(A=K_tile, B=Q^T) so each lane holds a full
//    q-row slice of scores -> softmax is lane-local + one shfl_xor(32)
//    to combine the two half-wave reg sets.  (guide §B "swapped QK^T")
//  * P -> bf16 B-fragments via pack + permlane32_swap (T12: one swap
//    fills two output words).
//  * O^T accumulated via mfma(A=V^T, B=P^T): C cols = q = lane&31, so
//    the online-softmax rescale is a pure lane-local multiply.
//
// Fragment maps verified on hardware by tests/ops/test_gpu_ops.py::
// test_mfma_layout_32x32x16 (probe.hip):
//   A(32x16): lane l -> row=l&31, k=8*(l>>5)+j
//   B(16x32): lane l -> col=l&31, k=8*(l>>5)+j
//   C(32x32): lane l, reg r -> col=l&31, row=(r&3)+8*(r>>2)+4*(l>>5)

#include "../common.h"

namespace {

typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

DEVINL unsigned pack_bf16x2(float lo, float hi) {
  return (unsigned)(unsigned short)f2bf(lo) |
         ((unsigned)(unsigned short)f2bf(hi) << 16);
}

template <int D>
__global__ void __launch_bounds__(256)
attn_fwd_mfma_kernel(const short* __restrict__ q, const short* __restrict__ k,
                     const short* __restrict__ v, short* __restrict__ o,
                     float* __restrict__ lse, int B, int H, int Hkv, int S,
                     float scale, int causal, Strides4 qs, Strides4 ks,
                     Strides4 vs, Strides4 os) {
  constexpr int NSL = D / 16;   // QK^T k-slices (over d)
  constexpr int NDT = D / 32;   // O^T d-tiles
  const int wid = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;        // this lane's q row (within wave tile)
  const int half = l >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = blockIdx.x * 128 + wid * 32;   // wave's q-tile origin
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long kb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vb = (long)b * vs.sb + (long)hk * vs.sh;

  // ---- load Q B-fragments once (row clamped at S-1 for pad rows)
  const int qrow = min(q0 + lq, S - 1);
  bf16x8_t qf[NSL];
  #pragma unroll
  for (int s = 0; s < NSL; ++s)
    qf[s] = *(const bf16x8_t*)(q + qb + (long)qrow * qs.ss + 16 * s + 8 * half);

  f32x16_t oacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) oacc[t] = {};
  float m_run = -1e30f, l_run = 0.f;

  const int kv_end = causal ? min(S, q0 + 32) : S;
  for (int c0 = 0; c0 < kv_end; c0 += 32) {
    // ---- S^T = K_tile . Q^T  (32k x 32q)
    f32x16_t sacc = {};
    const int krow = min(c0 + lq, S - 1);
    #pragma unroll
    for (int s = 0; s < NSL; ++s) {
      bf16x8_t kf = *(const bf16x8_t*)(k + kb + (long)krow * ks.ss + 16 * s + 8 * half);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc, 0, 0, 0);
    }

    // ---- scale + mask (lane: col=q=lq; reg r: row=k idx)
    const bool need_mask = (c0 + 31 > q0 && causal) || (c0 + 31 >= S);
    float p[16];
    float mtile = -1e30f;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      float sv = sacc[r] * scale;
      if (need_mask) {
        const int kg = c0 + krel;
        if (kg >= S || (causal && kg > q0 + lq)) sv = -1e30f;
      }
      p[r] = sv;
      mtile = fmaxf(mtile, sv);
    }
    mtile = fmaxf(mtile, __shfl_xor(mtile, 32, 64));  // combine half-wave regs

    // ---- online softmax update (lane-local; both halves identical)
    const float m_new = fmaxf(m_run, mtile);
    const float alpha = __expf(m_run - m_new);
    m_run = m_new;
    float psum = 0.f;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = __expf(p[r] - m_new);
      psum += p[r];
    }
    l_run = l_run * alpha + psum + __shfl_xor(psum, 32, 64);
    #pragma unroll
    for (int t = 0; t < NDT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) oacc[t][r] *= alpha;

    // ---- P^T B-fragments: pack quads, permlane32_swap halves (T12)
    // slice s (k=16s..16s+15): frag words = [a.x', a.y', b.x', b.y']
    unsigned pfrag[2][4];
    #pragma unroll
    for (int s = 0; s < 2; ++s) {
      unsigned ax = pack_bf16x2(p[8 * s + 0], p[8 * s + 1]);
      unsigned ay = pack_bf16x2(p[8 * s + 2], p[8 * s + 3]);
      unsigned bx = pack_bf16x2(p[8 * s + 4], p[8 * s + 5]);
      unsigned by = pack_bf16x2(p[8 * s + 6], p[8 * s + 7]);
      auto rx = __builtin_amdgcn_permlane32_swap(ax, bx, false, false);
      auto ry = __builtin_amdgcn_permlane32_swap(ay, by, false, false);
      pfrag[s][0] = rx[0];
      pfrag[s][1] = ry[0];
      pfrag[s][2] = rx[1];
      pfrag[s][3] = ry[1];
    }

    // ---- O^T += V^T . P^T  per d-tile, 2 k-slices of 16
    #pragma unroll
    for (int t = 0; t < NDT; ++t) {
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        // A = V^T: lane -> row=d=32t+lq, k=16s+8*half+j  (column read of V)
        short vf[8];
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int kg = min(c0 + 16 * s + 8 * half + j, S - 1);
          vf[j] = v[vb + (long)kg * vs.ss + 32 * t + lq];
        }
        oacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *(const bf16x8_t*)vf, *(const bf16x8_t*)pfrag[s], oacc[t], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O = O^T / l, scatter store (lane's q row fixed)
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (q0 + lq < S) {
    const long obase = (long)b * os.sb + (long)h * os.sh + (long)(q0 + lq) * os.ss;
    #pragma unroll
    for (int t = 0; t < NDT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = 32 * t + (r & 3) + 8 * (r >> 2) + 4 * half;
        o[obase + d] = f2bf(oacc[t][r] * inv_l);
      }
    if (half == 0)
      lse[((long)b * H + h) * S + q0 + lq] = m_run + __logf(fmaxf(l_run, 1e-30f));
  }
}

}  // namespace

void attn_fwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                   torch::Tensor o, torch::Tensor lse, int B, int H, int Hkv,
                   int S, int D, double scale, bool causal) {
  auto stream = scaffold_stream();
  dim3 grid((S + 127) / 128, B * H), block(256);
  const Strides4 qs = strides_of(q), ks = strides_of(k), vs = strides_of(v),
                 os = strides_of(o);
  #define LAUNCH_F(DD)                                                         \
    hipLaunchKernelGGL((attn_fwd_mfma_kernel<DD>), grid, block, 0, stream,     \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                       lse.data_ptr<float>(), B, H, Hkv, S, (float)scale,      \
                       causal ? 1 : 0, qs, ks, vs, os)
  if (D == 64) LAUNCH_F(64);
  else if (D == 128) LAUNCH_F(128);
  else TORCH_CHECK(false, "attn_fwd_mfma: unsupported D ", D);
  #undef LAUNCH_F
  HIP_CHECK(hipGetLastError());
}
