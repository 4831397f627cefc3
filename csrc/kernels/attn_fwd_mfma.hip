// K4 — MFMA flash attention forward (bf16, CDNA4 32x32x16 matrix cores).
//
// Structure (cdna_hip_programming.md §B "fused attention prefill", adapted):
//  * 4 waves / workgroup over a 128-row q block; wave w owns the
//    INTERLEAVED rows q0 + 4*i + w (i=0..31) so every wave sees the same
//    causal kv range — no idle tail waves at the block's barriers.
//  * swapped QK^T: S^T = mfma(A=K_tile, B=Q^T) so each lane holds a full
//    q-row slice of scores -> softmax is lane-local + one shfl_xor(32)
//    to combine the two half-wave reg sets.  (guide §B "swapped QK^T")
//  * P -> bf16 B-fragments via pack + permlane32_swap (T12: one swap
//    fills two output words).
//  * V tile cooperatively staged in LDS in the tr-image layout
//    (common.h) by global_load_lds, then consumed as V^T A-fragments via
//    ds_read_b64_tr_b16 (T10) — replaces the 32/64 scalar 2-B global
//    column loads per tile of the v1 kernel, and makes the kernel
//    insensitive to the (B,S,H,D)-transposed stride of the models' QKV.
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
  constexpr int NSLOT = 32 * D / 8 / 256;  // glds16 per thread per tile
  __shared__ __attribute__((aligned(16))) short vtile[32 * D];  // tr image

  const int wid = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int half = l >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = blockIdx.x * 128;
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long kb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vb = (long)b * vs.sb + (long)hk * vs.sh;

  // this lane's q row: interleaved assignment q0 + 4*lq + wid
  const int qrow_g = q0 + 4 * lq + wid;
  const int qrow = min(qrow_g, S - 1);
  bf16x8_t qf[NSL];
  #pragma unroll
  for (int s = 0; s < NSL; ++s)
    qf[s] = *(const bf16x8_t*)(q + qb + (long)qrow * qs.ss + 16 * s + 8 * half);

  f32x16_t oacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) oacc[t] = {};
  float m_run = -1e30f, l_run = 0.f;

  // tr-read addresses for the PV step (per d-tile t, k-quad pair at s)
  // subtile(q4, g) at element (q4*(D/16)+g)*64; lane column = l&15
  // per-lane read slot p: lane receives column (l&15) of each 4x16
  // subtile when lane p reads subtile_base + (p>>2)*32 + (p&3)*8
  // (hardware-measured semantics — common.h)
  unsigned tr_addr[NDT][2][2];
  {
    const unsigned base = lds_byte_off(vtile);
    const int p = l & 15;
    #pragma unroll
    for (int t = 0; t < NDT; ++t)
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        const int g = 2 * t + (lq >> 4);
        const int q4 = 4 * s + 2 * half;
        tr_addr[t][s][0] = base + 128u * (unsigned)(q4 * (D / 16) + g)
                           + (p >> 2) * 32 + (p & 3) * 8;
        tr_addr[t][s][1] = tr_addr[t][s][0] + 128u * (unsigned)(D / 16);
      }
  }

  const int kv_end = causal ? min(S, q0 + 128) : S;
  for (int c0 = 0; c0 < kv_end; c0 += 32) {
    // ---- cooperative V-tile stage into the tr image (glds16)
    #pragma unroll
    for (int it = 0; it < NSLOT; ++it) {
      const int s8 = (it * 4 + wid) * 64 + l;       // 8-element chunk id
      const int sub = s8 >> 3;
      const int rem8 = s8 & 7;                      // (k&3)*2 + c8
      const int krow = min(c0 + 4 * (sub / (D / 16)) + (rem8 >> 1), S - 1);
      const int d0 = 16 * (sub % (D / 16)) + 8 * (rem8 & 1);
      glds16(v + vb + (long)krow * vs.ss + d0,
             vtile + (long)(it * 4 + wid) * 512);
    }
    __syncthreads();  // drains the glds (vmcnt 0) + barrier

    // ---- S^T = K_tile . Q^T  (32k x 32q)
    f32x16_t sacc = {};
    const int krow = min(c0 + lq, S - 1);
    __builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
    #pragma unroll
    for (int s = 0; s < NSL; ++s) {
      bf16x8_t kf = *(const bf16x8_t*)(k + kb + (long)krow * ks.ss + 16 * s + 8 * half);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc, 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- scale + mask (lane: col=q; reg r: row=k idx)
    const bool need_mask = (causal && c0 + 31 > qrow_g) || (c0 + 31 >= S);
    float p[16];
    float mtile = -1e30f;
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      float sv = sacc[r] * scale;
      if (need_mask) {
        const int kg = c0 + krel;
        if (kg >= S || (causal && kg > qrow_g)) sv = -1e30f;
      }
      p[r] = sv;
      mtile = fmaxf(mtile, sv);
    }
    mtile = fmaxf(mtile, __shfl_xor(mtile, 32, 64));  // combine half-wave regs

    // ---- online softmax update (lane-local; both halves identical).
    // T13 with THR=0 (exact numerics): once the running max stops
    // growing — the common case after the first tiles — the whole wave
    // skips the O-rescale pass.
    float psum = 0.f;
    if (!__any(mtile > m_run)) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = __expf(p[r] - m_run);
        psum += p[r];
      }
      l_run += psum + __shfl_xor(psum, 32, 64);
    } else {
      const float m_new = fmaxf(m_run, mtile);
      const float alpha = __expf(m_run - m_new);
      m_run = m_new;
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
    }

    // ---- P^T B-fragments: pack quads, permlane32_swap halves (T12)
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

    // ---- O^T += V^T . P^T per d-tile; V^T A-fragments by tr-read (T10)
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int t = 0; t < NDT; ++t) {
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        bf16x8_t vf = tr_read_frag(tr_addr[t][s][0], tr_addr[t][s][1]);
        oacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vf, *(const bf16x8_t*)pfrag[s], oacc[t], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // all waves done with vtile before restage
  }

  // ---- epilogue: O = O^T / l, scatter store (lane's q row fixed)
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (qrow_g < S) {
    const long obase = (long)b * os.sb + (long)h * os.sh + (long)qrow_g * os.ss;
    #pragma unroll
    for (int t = 0; t < NDT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = 32 * t + (r & 3) + 8 * (r >> 2) + 4 * half;
        o[obase + d] = f2bf(oacc[t][r] * inv_l);
      }
    if (half == 0)
      lse[((long)b * H + h) * S + qrow_g] = m_run + __logf(fmaxf(l_run, 1e-30f));
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
