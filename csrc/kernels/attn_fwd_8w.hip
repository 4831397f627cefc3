// K4 — 8-wave MFMA flash attention forward (bf16, CDNA4 32x32x16).
//
// The round-2 restructure of attn_fwd_mfma.hip following the guide's
// 8-warp ladder (cdna_hip_programming.md §B "fused attention prefill")
// and the round-1 PMC diagnosis (4-wave kernel latency-bound: MfmaUtil
// 6-10%, dependent-MFMA chains with too few co-resident waves).
//
// Structure vs the 4-wave kernel:
//  * 8 waves / 512 threads over a 256-row q block (wave w owns the
//    interleaved rows q0 + 8*i + w): one block covers 2 waves/SIMD with
//    INDEPENDENT per-wave softmax phases (T5 role-split pays), and the
//    8 waves share each staged K/V tile — half the per-wave staging
//    traffic of the 4-wave kernel.
//  * T14 register staging (guide: glds grafts onto a register-staged
//    attention pipeline measure <= 0): next tile's global loads issue
//    BEFORE the compute phase; the ds_write pass lands after the
//    barrier.  All loads are ordinary (no glds), so hipcc's counted
//    vmcnt covers both the staging and the D=128 per-tile Q reloads
//    (no trap-b vmcnt(0) drain — only mixed load KINDS trigger it).
//  * exp2-fold: scale*log2e folded into the pre-softmax scale; softmax
//    runs in the base-2 domain (v_exp_f32 IS exp2 — saves one VALU
//    multiply per score), lse converted to natural log at the end.
//  * P->bf16 via v_cvt_pk_bf16_f32 (T12 recipe; 1 instruction per pair
//    vs ~7 VALU for the round-to-nearest-even bit sequence).
//  * D=128 runs with Q reloaded from L1 per tile (the 4-wave kernel's
//    measured lesson: pinning 32 VGPRs of Q tips the allocation into
//    spill at 2 waves/SIMD); D=64 pins Q (16 VGPRs, fits).
//
// Layouts identical to attn_fwd_mfma.hip: K row-image with the
// chunk^row XOR (T2), V tr-image consumed by ds_read_b64_tr_b16 with
// the hardware-measured exchange semantics (csrc/common.h), swapped
// QK^T so softmax is lane-local, defer-max THR=0 (exact).

#include "../common.h"

namespace {

typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

// v_cvt_pk_bf16_f32 has no builtin on gfx950 (guide T12): one VALU op
// packing two f32 into a dword of 2 bf16 (RTNE).  Inputs are VALU
// results (exp2), so no producer wait states are needed.
DEVINL unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

template <int D, int PACK>
__global__ void __launch_bounds__(512, 2)
attn_fwd_8w_kernel(const short* __restrict__ q, const short* __restrict__ k,
                   const short* __restrict__ v, short* __restrict__ o,
                   float* __restrict__ lse, int B, int H, int Hkv, int S,
                   float scale2, int causal, int nbx, Strides4 qs,
                   Strides4 ks, Strides4 vs, Strides4 os) {
  constexpr int NSL = D / 16;
  constexpr int NDT = D / 32;
  constexpr int CH = D / 8;          // 16-B chunks per row
  constexpr int BUF = 2 * 64 * D;    // K image + V tr-image (shorts)
  __shared__ __attribute__((aligned(16))) short kv_lds[2 * BUF];

  const int wid = threadIdx.x >> 6;  // 0..7
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int half = l >> 5;

  int bx, bh;
  xcd_tile_map(nbx, &bx, &bh);   // co-locate one head's tile-blocks per XCD
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = bx * 256;
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long kb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vb = (long)b * vs.sb + (long)hk * vs.sh;

  // wave w owns interleaved rows q0 + 8*i + w (i = 0..31)
  const int qrow_g = q0 + 8 * lq + wid;
  const int qrow = min(qrow_g, S - 1);
  // D=64: Q pinned in registers; D=128: reloaded from L1 per tile
  // (32 VGPRs saved keeps the 2-wave/SIMD allocation spill-free)
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
  float m_run = -1e30f, l_run = 0.f;  // base-2 domain

  const unsigned tr_base0 = lds_byte_off(kv_lds + 64 * D) + 128u * (lq >> 4)
                            + ((l & 15) >> 2) * 32 + (l & 3) * 8
                            + (half ? 128u * 2 * (D / 16) : 0);

  // ---- staging.  D=64 (Q pinned, registers to spare): T14 register
  // staging — issue the next tile's loads before compute, ds_write
  // after the barrier.  D=128 (Q reloaded per tile): glds staging like
  // the 4-wave kernel — register staging at D=128 needs 16 staging
  // VGPRs + addresses on top of the 160 live accumulator/score regs and
  // measured 72 B/lane of spill; glds writes LDS directly (the
  // destination is wave-uniform base + lane*16, so the image layouts
  // are glds-compatible: the V "transpose" comes from the per-lane
  // SOURCE scatter, rule 21).
  constexpr bool GLDS = (D == 128);
  constexpr int KSLOT = 64 * D / 8 / 512;  // D=128: 2, D=64: 1
  bf16x8_t kreg[GLDS ? 1 : KSLOT], vreg[GLDS ? 1 : KSLOT];

  auto src_k = [&](int c0, int s16) {
    // K row-image source (inverse chunk-XOR)
    const int krow = min(c0 + s16 / CH, S - 1);
    const int kbb = (s16 % CH) ^ ((s16 / CH) & 7);
    return k + kb + (long)krow * ks.ss + kbb * 8;
  };
  auto src_v = [&](int c0, int s16) {
    // V tr-image source
    const int sub = s16 / (32 * D / 8);
    const int s8 = s16 % (32 * D / 8);
    const int subidx = s8 >> 3;
    const int rem8 = s8 & 7;
    const int vrow = min(c0 + sub * 32 + 4 * (subidx / (D / 16)) + (rem8 >> 1), S - 1);
    const int d0 = 16 * (subidx % (D / 16)) + 8 * (rem8 & 1);
    return v + vb + (long)vrow * vs.ss + d0;
  };

  auto stage_load = [&](int c0) {
    #pragma unroll
    for (int it = 0; it < KSLOT; ++it) {
      const int s16 = (it * 8 + wid) * 64 + l;
      kreg[it] = *(const bf16x8_t*)src_k(c0, s16);
      vreg[it] = *(const bf16x8_t*)src_v(c0, s16);
    }
  };
  auto stage_write = [&](int buf) {
    short* kdst = kv_lds + buf * BUF;
    short* vdst = kdst + 64 * D;
    #pragma unroll
    for (int it = 0; it < KSLOT; ++it) {
      const int s16 = (it * 8 + wid) * 64 + l;
      *(bf16x8_t*)(kdst + (long)s16 * 8) = kreg[it];
      *(bf16x8_t*)(vdst + (long)s16 * 8) = vreg[it];
    }
  };
  auto stage_glds = [&](int buf, int c0) {
    short* kdst = kv_lds + buf * BUF;
    short* vdst = kdst + 64 * D;
    #pragma unroll
    for (int it = 0; it < KSLOT; ++it) {
      const int s16 = (it * 8 + wid) * 64 + l;
      glds16(src_k(c0, s16), kdst + (long)(it * 8 + wid) * 512);
      glds16(src_v(c0, s16), vdst + (long)(it * 8 + wid) * 512);
    }
  };

  if constexpr (GLDS) {
    stage_glds(0, 0);
  } else {
    stage_load(0);
    stage_write(0);
  }
  __syncthreads();

  const int kv_end = causal ? min(S, q0 + 256) : S;
  int cur = 0;
  for (int c0 = 0; c0 < kv_end; c0 += 64) {
    const short* kimg = kv_lds + cur * BUF;
    const unsigned tr_base = tr_base0 + (unsigned)cur * (BUF * 2);

    // T14 STAGE_LOAD (register path): issue tile t+1's global loads
    // before compute; hipcc tracks them with counted vmcnt and waits at
    // stage_write.  (glds path issues AFTER QK^T: the D=128 per-tile Q
    // reloads are ordinary loads, and an ordinary load beside an
    // in-flight glds makes hipcc drain vmcnt(0) — guide §5 trap b.)
    if constexpr (!GLDS) {
      if (c0 + 64 < kv_end) stage_load(c0 + 64);
    }

    // ---- S^T = K . Q^T for both 32-row sub-tiles
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

    // glds path: issue the next tile's LDS-DMA after QK^T (its Q
    // reloads were the last ordinary loads this tile) — the drain at
    // the end-of-tile barrier is covered by softmax + pack + PV
    if constexpr (GLDS) {
      if (c0 + 64 < kv_end) stage_glds(cur ^ 1, c0 + 64);
    }

    // ---- scale (exp2-folded) + mask in place (defer-max THR=0)
    const bool need_mask = (causal && c0 + 63 > qrow_g) || (c0 + 63 >= S);
    float mtile = -1e30f;
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krel = sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = sacc[sub][r] * scale2;
        if (need_mask) {
          const int kg = c0 + krel;
          if (kg >= S || (causal && kg > qrow_g)) sv = -1e30f;
        }
        sacc[sub][r] = sv;
        mtile = fmaxf(mtile, sv);
      }
    mtile = fmaxf(mtile, __shfl_xor(mtile, 32, 64));

    float psum = 0.f;
    if (!__any(mtile > m_run)) {
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          sacc[sub][r] = __builtin_amdgcn_exp2f(sacc[sub][r] - m_run);
          psum += sacc[sub][r];
        }
      l_run += psum + __shfl_xor(psum, 32, 64);
    } else {
      const float m_new = fmaxf(m_run, mtile);
      const float alpha = __builtin_amdgcn_exp2f(m_run - m_new);
      m_run = m_new;
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          sacc[sub][r] = __builtin_amdgcn_exp2f(sacc[sub][r] - m_new);
          psum += sacc[sub][r];
        }
      l_run = l_run * alpha + psum + __shfl_xor(psum, 32, 64);
      #pragma unroll
      for (int t = 0; t < NDT; ++t)
        #pragma unroll
        for (int r = 0; r < 16; ++r) oacc[t][r] *= alpha;
    }

    // ---- P pack (cvt_pk + permlane32_swap, T12) + PV per sub-tile
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      unsigned pfrag[2][4];
      #pragma unroll
      for (int s = 0; s < 2; ++s) {
        unsigned ax = cvt_pk_bf16(sacc[sub][8 * s + 0], sacc[sub][8 * s + 1]);
        unsigned ay = cvt_pk_bf16(sacc[sub][8 * s + 2], sacc[sub][8 * s + 3]);
        unsigned bx = cvt_pk_bf16(sacc[sub][8 * s + 4], sacc[sub][8 * s + 5]);
        unsigned by = cvt_pk_bf16(sacc[sub][8 * s + 6], sacc[sub][8 * s + 7]);
        auto rx = __builtin_amdgcn_permlane32_swap(ax, bx, false, false);
        auto ry = __builtin_amdgcn_permlane32_swap(ay, by, false, false);
        pfrag[s][0] = rx[0];
        pfrag[s][1] = ry[0];
        pfrag[s][2] = rx[1];
        pfrag[s][3] = ry[1];
      }
      const unsigned sub_off = (unsigned)sub * (32u * D * 2);
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

    if constexpr (GLDS) {
      // single barrier: drains the in-flight glds (hipcc emits the
      // vmcnt(0) inside __syncthreads) + read/write epoch separation
      __syncthreads();
    } else {
      // T14 write-after-barrier: first barrier = all waves done reading
      // buf cur^1; hipcc's counted vmcnt drains the staged loads at the
      // writes; second barrier = image complete before next tile reads
      __syncthreads();
      if (c0 + 64 < kv_end) {
        stage_write(cur ^ 1);
      }
      __syncthreads();
    }
    cur ^= 1;
  }

  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
  if (qrow_g < S) {
    const long obase = (long)b * os.sb + (long)h * os.sh + (long)qrow_g * os.ss;
    // packed store: each r-quad's four d-values are consecutive in
    // THIS lane (d = 32t + (r&3) + 8(r>>2) + 4h), so one 8-B store per
    // quad replaces 4 scalar stores (the store tail is issue-bound —
    // guide T21 diagnostic)
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
      // lse contract is NATURAL log (backward consumes it):
      // max and sum ran in base 2, so lse = ln2 * (m2 + log2(l))
      lse[((long)b * H + h) * S + qrow_g] =
          0.6931471805599453f * (m_run + __log2f(fmaxf(l_run, 1e-30f)));
  }
}

}  // namespace

void attn_fwd_8w(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor o, torch::Tensor lse, int B, int H, int Hkv,
                 int S, int D, double scale, bool causal) {
  auto stream = scaffold_stream();
  const int nbx = (S + 255) / 256;
  dim3 grid(nbx * B * H), block(512);
  const Strides4 qs = strides_of(q), ks = strides_of(k), vs = strides_of(v),
                 os = strides_of(o);
  // exp2-fold: scale * log2(e) — softmax runs in the base-2 domain
  const float scale2 = (float)scale * 1.4426950408889634f;
  const char* pk_env = getenv("SCAFFOLD_ATTN_PACK");
  const int pack = pk_env ? atoi(pk_env) : 1;
  #define LAUNCH_F8(DD)                                                        \
    if (pack)                                                                  \
      hipLaunchKernelGGL((attn_fwd_8w_kernel<DD, 1>), grid, block, 0, stream,  \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                       lse.data_ptr<float>(), B, H, Hkv, S, scale2,            \
                       causal ? 1 : 0, nbx, qs, ks, vs, os);                   \
    else                                                                       \
      hipLaunchKernelGGL((attn_fwd_8w_kernel<DD, 0>), grid, block, 0, stream,  \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), (short*)o.data_ptr(),       \
                       lse.data_ptr<float>(), B, H, Hkv, S, scale2,            \
                       causal ? 1 : 0, nbx, qs, ks, vs, os)
  if (D == 64) LAUNCH_F8(64);
  else if (D == 128) LAUNCH_F8(128);
  else TORCH_CHECK(false, "attn_fwd_8w: unsupported D ", D);
  #undef LAUNCH_F8
  HIP_CHECK(hipGetLastError());
}
