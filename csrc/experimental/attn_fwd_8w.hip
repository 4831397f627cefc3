// EXPERIMENTAL — round-2 candidate attention forward (NOT in the build;
// see NOTES.md "Attention state + next steps").  Compile-checked for
// gfx950 (256 VGPR, 2 waves/SIMD, 72 B/lane spill — trim the Q pin or
// staging depth in round 2); numerics/perf NOT yet hardware-validated.
//
// Differences vs the shipped attn_fwd_mfma.hip, following the guide's
// 8-warp ladder (§B) and the round-1 PMC diagnosis (latency-bound,
// MfmaUtil ~10%):
//  * 8 waves / 512 threads over a 256-row q block (wave w owns rows
//    q0 + 8*i + w): one block per CU, the 8 waves share each staged
//    K/V tile so global traffic per tile is 1/2 of the 4-wave kernel's,
//    and the block's waves give each SIMD 2 co-resident waves with
//    independent per-wave softmax phases (T5 role-split).
//  * REGISTER staging with the T14 split (guide: glds attention grafts
//    measure <= 0): the next tile's global loads are issued before the
//    compute phase, and the ds_write pass lands after the barrier —
//    plain loads are tracked by hipcc's counted waits, so the trap-(b)
//    vmcnt(0) drain problem does not exist by construction.
//  * Q fragments pinned in registers (the wave re-reads them S/64
//    times; 32 VGPRs at D=128).
//
// Layouts are identical to the shipped kernel (K row-image with the
// chunk^row XOR; V tr-image consumed by ds_read_b64_tr_b16 with the
// hardware-measured exchange semantics — csrc/common.h).

#include "../common.h"

namespace {

typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

DEVINL unsigned pack_bf16x2(float lo, float hi) {
  return (unsigned)(unsigned short)f2bf(lo) |
         ((unsigned)(unsigned short)f2bf(hi) << 16);
}

template <int D>  // D == 128 intended
__global__ void __launch_bounds__(512, 2)
attn_fwd_8w_kernel(const short* __restrict__ q, const short* __restrict__ k,
                   const short* __restrict__ v, short* __restrict__ o,
                   float* __restrict__ lse, int B, int H, int Hkv, int S,
                   float scale, int causal, Strides4 qs, Strides4 ks,
                   Strides4 vs, Strides4 os) {
  constexpr int NSL = D / 16;
  constexpr int NDT = D / 32;
  constexpr int CH = D / 8;          // 16-B chunks per row
  constexpr int BUF = 2 * 64 * D;    // K image + V tr-image (shorts)
  __shared__ __attribute__((aligned(16))) short kv_lds[2 * BUF];

  const int wid = threadIdx.x >> 6;  // 0..7
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int half = l >> 5;

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = blockIdx.x * 256;
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long kb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vb = (long)b * vs.sb + (long)hk * vs.sh;

  // wave w owns interleaved rows q0 + 8*i + w (i = 0..31)
  const int qrow_g = q0 + 8 * lq + wid;
  const int qrow = min(qrow_g, S - 1);
  bf16x8_t qf[NSL];
  #pragma unroll
  for (int s = 0; s < NSL; ++s)
    qf[s] = *(const bf16x8_t*)(q + qb + (long)qrow * qs.ss + 16 * s + 8 * half);

  f32x16_t oacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) oacc[t] = {};
  float m_run = -1e30f, l_run = 0.f;

  const unsigned tr_base0 = lds_byte_off(kv_lds + 64 * D) + 128u * (lq >> 4)
                            + ((l & 15) >> 2) * 32 + (l & 3) * 8
                            + (half ? 128u * 2 * (D / 16) : 0);

  // ---- T14 register staging: each thread owns KSLOT K-slots and
  // VSLOT V-slots of the 16-KiB-per-operand tile image
  constexpr int KSLOT = 64 * D / 8 / 512;  // D=128: 2
  bf16x8_t kreg[KSLOT], vreg[KSLOT];

  auto stage_load = [&](int c0) {
    #pragma unroll
    for (int it = 0; it < KSLOT; ++it) {
      const int s16 = (it * 8 + wid) * 64 + l;
      // K row-image source (inverse chunk-XOR)
      const int krow = min(c0 + s16 / CH, S - 1);
      const int kbb = (s16 % CH) ^ ((s16 / CH) & 7);
      kreg[it] = *(const bf16x8_t*)(k + kb + (long)krow * ks.ss + kbb * 8);
      // V tr-image source
      const int sub = s16 / (32 * D / 8);
      const int s8 = s16 % (32 * D / 8);
      const int subidx = s8 >> 3;
      const int rem8 = s8 & 7;
      const int vrow = min(c0 + sub * 32 + 4 * (subidx / (D / 16)) + (rem8 >> 1), S - 1);
      const int d0 = 16 * (subidx % (D / 16)) + 8 * (rem8 & 1);
      vreg[it] = *(const bf16x8_t*)(v + vb + (long)vrow * vs.ss + d0);
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

  stage_load(0);
  stage_write(0);
  __syncthreads();

  const int kv_end = causal ? min(S, q0 + 256) : S;
  int cur = 0;
  for (int c0 = 0; c0 < kv_end; c0 += 64) {
    const short* kimg = kv_lds + cur * BUF;
    const unsigned tr_base = tr_base0 + (unsigned)cur * (BUF * 2);

    // T14 STAGE_LOAD: issue tile t+1's global loads before compute;
    // hipcc tracks them with counted vmcnt and waits at stage_write
    if (c0 + 64 < kv_end) stage_load(c0 + 64);

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
        sacc[sub] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc[sub], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- scale + mask + exp in place (defer-max THR=0)
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

    // ---- P pack + PV per sub-tile (batched tr reads)
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

    // T14 write-after-barrier: all waves done READING buf cur^1's
    // predecessor; the staging registers' loads are waited here by
    // hipcc's counted vmcnt, then written
    __syncthreads();
    if (c0 + 64 < kv_end) {
      stage_write(cur ^ 1);
    }
    __syncthreads();
    cur ^= 1;
  }

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

// explicit instantiation so a standalone compile exercises codegen
template __global__ void attn_fwd_8w_kernel<128>(
    const short*, const short*, const short*, short*, float*, int, int, int,
    int, float, int, Strides4, Strides4, Strides4, Strides4);

}  // namespace
