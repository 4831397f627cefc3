// K4 — MFMA flash attention forward (bf16, CDNA4 32x32x16 matrix cores).
//
// Structure (cdna_hip_programming.md §B "fused attention prefill" ladder,
// adapted to 32x32x16 fragments):
//  * 4 waves / workgroup over a 128-row q block; wave w owns the
//    INTERLEAVED rows q0 + 4*i + w (i=0..31) so every wave sees the same
//    causal kv range — no idle tail waves at the block's barriers.
//  * KVBLK=64: K and V staged cooperatively in LDS per 64-row kv tile
//    (one barrier pair per 64 rows; softmax max/rescale amortized over
//    64 scores per lane).
//  * K image: row-major with the 16-B chunk XOR swizzle
//    (chunk ^= row&7, the T2 pattern — <=2-way on ds_read_b128) filled
//    by global_load_lds with the inverse swizzle on the SOURCE address
//    (rule 21; the XOR permutes chunks within one 128-B row = one
//    cacheline, so global coalescing is preserved).
//  * V image: tr-layout (common.h) consumed as V^T A-fragments via
//    ds_read_b64_tr_b16 (T10) — hardware-measured semantics.
//  * swapped QK^T: S^T = mfma(A=K, B=Q^T) so each lane holds full q-row
//    score slices; softmax is lane-local + one shfl_xor(32).
//  * P -> bf16 B-fragments via pack + permlane32_swap (T12).
//  * T13 defer-max with THR=0 (exact): skip the O-rescale pass when the
//    running max did not grow (wave-uniform branch).
//  * T5 setprio(1) around the MFMA clusters.
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

// second launch_bounds arg = waves per SIMD (guide §6 G1): force the
// allocator to fit 2 waves/SIMD (<=256 regs incl. AGPRs) — at D=128 the
// default allocation was 362 regs -> 1 wave/SIMD and every barrier
// stalled the whole SIMD (measured 11% slower than the 2-wave build).
template <int D, int PACK>
__global__ void __launch_bounds__(256, 2)
attn_fwd_mfma_kernel(const short* __restrict__ q, const short* __restrict__ k,
                     const short* __restrict__ v, short* __restrict__ o,
                     float* __restrict__ lse, int B, int H, int Hkv, int S,
                     float scale, int causal, int nbx, Strides4 qs,
                     Strides4 ks, Strides4 vs, Strides4 os) {
  constexpr int NSL = D / 16;   // QK^T k-slices (over d)
  constexpr int NDT = D / 32;   // O^T d-tiles
  constexpr int CH = D / 8;     // 16-B chunks per row
  constexpr int BUF = 2 * 64 * D;  // K row-image + V tr-image per buffer
  // double-buffered: tile c0+64 streams in while c0 computes, so the
  // end-of-tile __syncthreads drains glds that issued a full compute
  // phase earlier (HBM latency hidden, 2-phase minimum recipe)
  __shared__ __attribute__((aligned(16))) short kv_lds[2 * BUF];

  const int wid = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int half = l >> 5;

  int bx, bh;
  xcd_tile_map(nbx, &bx, &bh);   // co-locate one head's tile-blocks per XCD
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = bx * 128;
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long kb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vb = (long)b * vs.sb + (long)hk * vs.sh;

  // this lane's q row: interleaved assignment q0 + 4*lq + wid
  const int qrow_g = q0 + 4 * lq + wid;
  const int qrow = min(qrow_g, S - 1);
  // D=64: Q fragments pinned in registers.  D=128: reloaded from L1
  // per tile (32 VGPRs saved buys 2 waves/SIMD with zero spill; the
  // wave's 8-KiB q tile stays L1-resident across the kv loop).
  constexpr bool QREG = (D <= 64);
  const short* qrow_ptr = q + qb + (long)qrow * qs.ss + 8 * half;
  bf16x8_t qf[QREG ? NSL : 1];
  if constexpr (QREG) {
    #pragma unroll
    for (int s = 0; s < NSL; ++s)
      qf[s] = *(const bf16x8_t*)(qrow_ptr + 16 * s);
  }

  f32x16_t oacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) oacc[t] = {};
  float m_run = -1e30f, l_run = 0.f;

  // tr-read base for the PV step: per-(t,s,sub) offsets are constants
  // added at the read site (1 VGPR instead of a 16-entry address array)
  //   addr(t,s,q4half) = tr_base + 128*((4s+2*half)*(D/16) + 2t) [+128*(D/16) for q4+1]
  const unsigned tr_base0 = lds_byte_off(kv_lds + 64 * D) + 128u * (lq >> 4)
                            + ((l & 15) >> 2) * 32 + (l & 3) * 8
                            + (half ? 128u * 2 * (D / 16) : 0);

  // cooperative stage of the 64-row K and V tiles into buffer `buf`
  auto stage = [&](int buf, int c0) {
    constexpr int KSLOT = 64 * D / 8 / 256;  // glds per thread per image
    short* kdst = kv_lds + buf * BUF;
    short* vdst = kdst + 64 * D;
    #pragma unroll
    for (int it = 0; it < KSLOT; ++it) {
      // K: row-image, chunk-XOR swizzled via the SOURCE address
      const int s16 = (it * 4 + wid) * 64 + l;
      const int krow = min(c0 + s16 / CH, S - 1);
      const int kbb = (s16 % CH) ^ ((s16 / CH) & 7);
      glds16(k + kb + (long)krow * ks.ss + kbb * 8,
             kdst + (long)(it * 4 + wid) * 512);
      // V: tr-image (two 32-row halves)
      const int sub = s16 / (32 * D / 8);
      const int s8 = s16 % (32 * D / 8);
      const int subidx = s8 >> 3;
      const int rem8 = s8 & 7;
      const int vrow = min(c0 + sub * 32 + 4 * (subidx / (D / 16)) + (rem8 >> 1), S - 1);
      const int d0 = 16 * (subidx % (D / 16)) + 8 * (rem8 & 1);
      glds16(v + vb + (long)vrow * vs.ss + d0,
             vdst + (long)(it * 4 + wid) * 512);
    }
  };

  const int kv_end = causal ? min(S, q0 + 128) : S;
  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int c0 = 0; c0 < kv_end; c0 += 64) {
    const short* kimg = kv_lds + cur * BUF;
    const unsigned tr_base = tr_base0 + (unsigned)cur * (BUF * 2);

    // ---- S^T = K_tile . Q^T for both 32-row sub-tiles
    f32x16_t sacc[2];
    sacc[0] = {};
    sacc[1] = {};
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int krow = sub * 32 + lq;
      #pragma unroll
      for (int s = 0; s < NSL; ++s) {
        const int kbb = (2 * s + half) ^ (krow & 7);
        bf16x8_t kf = *(const bf16x8_t*)(kimg + krow * D + kbb * 8);
        const bf16x8_t qv =
            QREG ? qf[s] : *(const bf16x8_t*)(qrow_ptr + 16 * s);
        sacc[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qv, sacc[sub], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // issue the NEXT tile's glds AFTER the QK^T block: any ordinary
    // VGPR-destination load inside that block (the D=128 per-tile Q
    // reload) makes hipcc wait vmcnt(0) — which would drain a prefetch
    // issued before it with zero compute cover (guide §5 trap b).
    // Issued here, the drain at the end-of-tile barrier is covered by
    // softmax + pack + PV.
    if (c0 + 64 < kv_end) stage(cur ^ 1, c0 + 64);

    // ---- scale + mask IN PLACE in sacc (register reuse: no separate
    // p[] array), then exponentiate in place
    const bool need_mask = (causal && c0 + 63 > qrow_g) || (c0 + 63 >= S);
    float mtile = -1e30f;
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krel = sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = sacc[sub][r] * scale;
        if (need_mask) {
          const int kg = c0 + krel;
          if (kg >= S || (causal && kg > qrow_g)) sv = -1e30f;
        }
        sacc[sub][r] = sv;
        mtile = fmaxf(mtile, sv);
      }
    mtile = fmaxf(mtile, __shfl_xor(mtile, 32, 64));

    // ---- online softmax update; defer-max THR=0 (exact)
    float psum = 0.f;
    if (!__any(mtile > m_run)) {
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          sacc[sub][r] = __expf(sacc[sub][r] - m_run);
          psum += sacc[sub][r];
        }
      l_run += psum + __shfl_xor(psum, 32, 64);
    } else {
      const float m_new = fmaxf(m_run, mtile);
      const float alpha = __expf(m_run - m_new);
      m_run = m_new;
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          sacc[sub][r] = __expf(sacc[sub][r] - m_new);
          psum += sacc[sub][r];
        }
      l_run = l_run * alpha + psum + __shfl_xor(psum, 32, 64);
      #pragma unroll
      for (int t = 0; t < NDT; ++t)
        #pragma unroll
        for (int r = 0; r < 16; ++r) oacc[t][r] *= alpha;
    }

    // ---- P^T B-fragments (T12) then O^T += V^T . P^T per sub-tile
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      unsigned pfrag[2][4];
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        unsigned ax = pack_bf16x2(sacc[sub][8 * s + 0], sacc[sub][8 * s + 1]);
        unsigned ay = pack_bf16x2(sacc[sub][8 * s + 2], sacc[sub][8 * s + 3]);
        unsigned bx = pack_bf16x2(sacc[sub][8 * s + 4], sacc[sub][8 * s + 5]);
        unsigned by = pack_bf16x2(sacc[sub][8 * s + 6], sacc[sub][8 * s + 7]);
        auto rx = __builtin_amdgcn_permlane32_swap(ax, bx, false, false);
        auto ry = __builtin_amdgcn_permlane32_swap(ay, by, false, false);
        pfrag[s][0] = rx[0];
        pfrag[s][1] = ry[0];
        pfrag[s][2] = rx[1];
        pfrag[s][3] = ry[1];
      }
      const unsigned sub_off = (unsigned)sub * (32u * D * 2);
      // batch 2 d-tiles x 2 k-slices of V^T fragments per LDS wait
      // (q4 = 4s, +2*half folded into tr_base)
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int t2 = 0; t2 < NDT; t2 += 2) {
        auto addr = [&](int t, int s) {
          return tr_base + sub_off + 128u * (unsigned)((4 * s) * (D / 16) + 2 * t);
        };
        const unsigned qstep = 128u * (D / 16);
        bf16x8_quad vq = tr_read_quad(
            addr(t2, 0), addr(t2, 0) + qstep, addr(t2, 1), addr(t2, 1) + qstep,
            addr(t2 + 1, 0), addr(t2 + 1, 0) + qstep,
            addr(t2 + 1, 1), addr(t2 + 1, 1) + qstep);
        oacc[t2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vq.f[0], *(const bf16x8_t*)pfrag[0], oacc[t2], 0, 0, 0);
        oacc[t2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vq.f[1], *(const bf16x8_t*)pfrag[1], oacc[t2], 0, 0, 0);
        oacc[t2 + 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vq.f[2], *(const bf16x8_t*)pfrag[0], oacc[t2 + 1], 0, 0, 0);
        oacc[t2 + 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            vq.f[3], *(const bf16x8_t*)pfrag[1], oacc[t2 + 1], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // drains the NEXT tile's glds (issued before compute — latency
    // hidden under this tile's MFMAs) + read-write barrier
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O = O^T / l, scatter store (lane's q row fixed)
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (qrow_g < S) {
    const long obase = (long)b * os.sb + (long)h * os.sh + (long)qrow_g * os.ss;
    // packed 8-B stores per r-quad (see the 8-wave kernel's note)
    typedef short bf16x4_s __attribute__((ext_vector_type(4)));
    if (PACK) {
      #pragma unroll
      for (int t = 0; t < NDT; ++t)
        #pragma unroll
        for (int rq = 0; rq < 4; ++rq) {
          bf16x4_s pk;
          #pragma unroll
          for (int j = 0; j < 4; ++j)
            pk[j] = f2bf(oacc[t][rq * 4 + j] * inv_l);
          *(bf16x4_s*)(o + obase + 32 * t + 8 * rq + 4 * half) = pk;
        }
    } else {
      #pragma unroll
      for (int t = 0; t < NDT; ++t)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int d = 32 * t + (r & 3) + 8 * (r >> 2) + 4 * half;
          o[obase + d] = f2bf(oacc[t][r] * inv_l);
        }
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
  const int nbx = (S + 127) / 128;
  dim3 grid(nbx * B * H), block(256);
  const Strides4 qs = strides_of(q), ks = strides_of(k), vs = strides_of(v),
                 os = strides_of(o);
  const char* pk_env = getenv("SCAFFOLD_ATTN_PACK");
  const int pack = pk_env ? atoi(pk_env) : 1;
  #define LAUNCH_F(DD)                                                         \
    if (pack)                                                                  \
      hipLaunchKernelGGL((attn_fwd_mfma_kernel<DD, 1>), grid, block, 0,        \
                       stream,                                                 \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                       lse.data_ptr<float>(), B, H, Hkv, S, (float)scale,      \
                       causal ? 1 : 0, nbx, qs, ks, vs, os);                   \
    else                                                                       \
      hipLaunchKernelGGL((attn_fwd_mfma_kernel<DD, 0>), grid, block, 0,        \
                       stream,                                                 \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                       lse.data_ptr<float>(), B, H, Hkv, S, (float)scale,      \
                       causal ? 1 : 0, nbx, qs, ks, vs, os)
  if (D == 64) LAUNCH_F(64);
  else if (D == 128) LAUNCH_F(128);
  else TORCH_CHECK(false, "attn_fwd_mfma: unsupported D ", D);
  #undef LAUNCH_F
  HIP_CHECK(hipGetLastError());
}
