// K4 — 8-wave MFMA flash attention backward (bf16, 32x32x16).
//
// Round-2 restructure of attn_bwd_mfma.hip's dq/dkv kernels, same move
// as the forward's 8-wave rebuild (attn_fwd_8w.hip): 512 threads over a
// 256-row block (wave w owns interleaved rows base + 8*i + w), 64-row
// staged tiles shared by all 8 waves (half the staging traffic of the
// 4-wave kernels), 2 waves/SIMD with independent per-wave phases so the
// SIMD always has a runnable wave (the 4-wave kernels measured
// latency-bound: MfmaUtil 6-10%).
//
// Math identical to attn_bwd_mfma.hip (fragment maps hardware-verified
// by probe.hip):
//  dQ kernel  (wave = 32 interleaved q rows, kv tiles of 64 = 2 subs):
//    S^T  = mfma(K, Q^T); P^T = exp2(S^T*scale2 - lse2)  lane-local
//    dP^T = mfma(V, dO^T); dS^T = P^T*(dP^T - delta)*scale
//    dQ^T += mfma(K^T_cols, pack(dS^T))   [K^T by tr-read from LDS]
//  dKV kernel (wave = 32 interleaved kv rows, q tiles of 64 = 2 subs,
//              GQA loop over the group's heads):
//    S = mfma(Q, K^T); P = exp2(S*scale2 - lse2[q])
//    dV^T += mfma(pack(P^T), dO_cols); dK^T += mfma(pack(dS^T), Q_cols)
//
// exp2-fold: scale*log2e folded into the pre-softmax scale, lse scaled
// by log2e at load — one VALU multiply per score saved (the fwd writes
// lse in natural log; the bwd converts per row, not per element).
// P/dS -> bf16 fragments via v_cvt_pk_bf16_f32 (T12).

#include "../common.h"

namespace {

typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

constexpr float LOG2E = 1.4426950408889634f;

DEVINL unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// C-layout f32[16] (lane=col, regs=rows) -> two 8-k B/A fragments
DEVINL void pack_frags(const float p[16], unsigned out[2][4]) {
  #pragma unroll
  for (int s = 0; s < 2; ++s) {
    unsigned ax = cvt_pk_bf16(p[8 * s + 0], p[8 * s + 1]);
    unsigned ay = cvt_pk_bf16(p[8 * s + 2], p[8 * s + 3]);
    unsigned bx = cvt_pk_bf16(p[8 * s + 4], p[8 * s + 5]);
    unsigned by = cvt_pk_bf16(p[8 * s + 6], p[8 * s + 7]);
    auto rx = __builtin_amdgcn_permlane32_swap(ax, bx, false, false);
    auto ry = __builtin_amdgcn_permlane32_swap(ay, by, false, false);
    out[s][0] = rx[0];
    out[s][1] = ry[0];
    out[s][2] = rx[1];
    out[s][3] = ry[1];
  }
}

// cooperative glds16 stage of a 64-row global tile into TWO consecutive
// 32-row tr-images (layout in common.h), 512 threads.  32*D/8 16-B
// slots per image are a multiple of 64 for D in {64,128}, so no wave's
// slots straddle an image boundary (glds dest stays lane-linear).
template <int D>
DEVINL void stage_tr64(const short* __restrict__ src, long row_stride, int r0,
                       int S, short* lds, int tid) {
  constexpr int IMG = 32 * D / 8;        // 16-B slots per 32-row image
  constexpr int NSLOT = 2 * IMG / 512;   // per thread (D=128: 2, D=64: 1)
  #pragma unroll
  for (int it = 0; it < NSLOT; ++it) {
    const int s = it * 512 + tid;
    const int image = s / IMG;
    const int s8 = s % IMG;
    const int sub = s8 >> 3;
    const int rem8 = s8 & 7;
    const int row = min(r0 + image * 32 + 4 * (sub / (D / 16)) + (rem8 >> 1), S - 1);
    const int d0 = 16 * (sub % (D / 16)) + 8 * (rem8 & 1);
    glds16(src + (long)row * row_stride + d0, lds + (long)s * 8);
  }
}

// per-lane tr-read BASE for the column fragments of a 32-row tr-image
template <int D>
DEVINL unsigned tr_col_base(const void* img, int l) {
  const int p = l & 15;
  return lds_byte_off(img) + 128u * ((l & 31) >> 4)
         + (p >> 2) * 32 + (p & 3) * 8
         + ((l >> 5) ? 128u * 2 * (D / 16) : 0);
}

// ---------------------------------------------------------------- dQ
template <int D>
__global__ void __launch_bounds__(512, 2)
attn_bwd_dq_8w_kernel(const short* __restrict__ dout,
                      const short* __restrict__ q,
                      const short* __restrict__ k,
                      const short* __restrict__ v,
                      const float* __restrict__ lse,
                      const float* __restrict__ delta,
                      short* __restrict__ dq, int B, int H, int Hkv, int S,
                      float scale, int causal, int nbx, Strides4 dos,
                      Strides4 qs, Strides4 ks, Strides4 vs, Strides4 dqs) {
  constexpr int NSL = D / 16;
  constexpr int NDT = D / 32;
  // double-buffered 64-row K tr-image (2 x 32-row images per buffer)
  __shared__ __attribute__((aligned(16))) short ktile[2 * 64 * D];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lq = l & 31;
  const int half = l >> 5;

  int bx, bh;
  xcd_tile_map(nbx, &bx, &bh);
  const int b = bh / H, h = bh % H;
  const int hk = h / (H / Hkv);
  const int q0 = bx * 256;
  if (q0 >= S) return;

  const long qb = (long)b * qs.sb + (long)h * qs.sh;
  const long dob = (long)b * dos.sb + (long)h * dos.sh;
  const long kbb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vbb = (long)b * vs.sb + (long)hk * vs.sh;
  const long rhead = ((long)b * H + h) * S;

  const int qrow_g = q0 + 8 * lq + wid;     // interleaved q row
  const int qrow = min(qrow_g, S - 1);
  // Q/dO fragments pinned in registers (64 VGPRs at D=128): reloading
  // either per sub-tile measured dq 14-53% slower; with the mid-loop
  // scheduling split below the allocation stays at 198 regs, 0 spill.
  constexpr bool DOREG = true;
  const short* qrow_ptr = q + qb + (long)qrow * qs.ss + 8 * half;
  const short* dorow_ptr = dout + dob + (long)qrow * dos.ss + 8 * half;
  bf16x8_t qf[NSL], dof[DOREG ? NSL : 1];
  #pragma unroll
  for (int s = 0; s < NSL; ++s) {
    qf[s] = *(const bf16x8_t*)(qrow_ptr + 16 * s);
    if constexpr (DOREG) dof[s] = *(const bf16x8_t*)(dorow_ptr + 16 * s);
  }
  const float my_lse2 = lse[rhead + qrow] * LOG2E;
  const float my_delta = delta[rhead + qrow];
  const float scale2 = scale * LOG2E;

  const unsigned ktr0 = tr_col_base<D>(ktile, l);

  f32x16_t dqacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) dqacc[t] = {};

  const int kv_end = causal ? min(S, q0 + 256) : S;
  stage_tr64<D>(k + kbb, ks.ss, 0, S, ktile, tid);
  __syncthreads();
  int cur = 0;
  for (int c0 = 0; c0 < kv_end; c0 += 64) {
    // two 32-row sub-tiles per staged 64-row tile, serial (keeps one
    // sacc/dpacc pair live -> no spill at D=128)
    #pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int cs = c0 + sub * 32;
      if (cs >= kv_end) break;
      const unsigned ktr = ktr0 + (unsigned)cur * (64 * D * 2)
                           + (unsigned)sub * (32 * D * 2);

      const int krow = min(cs + lq, S - 1);
      // K A-fragments from the staged tr-image (row fragments are
      // contiguous 16-B runs of the tr layout); V has no image
      const short* kimg8 = ktile + (long)cur * 64 * D + (long)sub * 32 * D;
      const unsigned rowel = ((unsigned)(lq >> 2) * (D / 16)) * 64u +
                             (unsigned)(lq & 3) * 16u + 8u * half;
      f32x16_t sacc = {}, dpacc = {};
      __builtin_amdgcn_s_setprio(1);
      // FULL unroll (a partial '#pragma unroll N' leaves a rolled loop
      // whose qf[s] becomes a per-element compare-select tree — rule
      // 20's trap in select form, measured 14x slower); the mid-loop
      // memory-clobber asm splits the scheduling region so the
      // compiler cannot buffer all 8 k-slices' loads at once
      #pragma unroll
      for (int s = 0; s < NSL; ++s) {
        bf16x8_t kf = *(const bf16x8_t*)(kimg8 + rowel + s * 64);
        bf16x8_t vf = *(const bf16x8_t*)(v + vbb + (long)krow * vs.ss + 16 * s + 8 * half);
        const bf16x8_t dov =
            DOREG ? dof[s] : *(const bf16x8_t*)(dorow_ptr + 16 * s);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc, 0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dov, dpacc, 0, 0, 0);
        if constexpr (D > 64) {
          if (s == NSL / 2 - 1) asm volatile("" ::: "memory");
        }
      }
      __builtin_amdgcn_s_setprio(0);

      // next 64-row tile's glds: issued after the LAST ordinary-load
      // block of this tile (sub==1's kf/vf) — an ordinary load beside
      // an in-flight glds drains vmcnt(0) (guide §5 trap b); drained
      // under sub 1's softmax + pack + dQ MFMAs
      if (sub == 1 && c0 + 64 < kv_end)
        stage_tr64<D>(k + kbb, ks.ss, c0 + 64, S, ktile + (cur ^ 1) * 64 * D, tid);

      const bool need_mask = (causal && cs + 31 > qrow_g) || (cs + 31 >= S);
      float ds[16];
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
        float sv = sacc[r] * scale2;
        if (need_mask) {
          const int kg = cs + krel;
          if (kg >= S || (causal && kg > qrow_g)) sv = -1e30f;
        }
        const float p = __builtin_amdgcn_exp2f(sv - my_lse2);
        ds[r] = p * (dpacc[r] - my_delta) * scale;
      }

      unsigned dsf[2][4];
      pack_frags(ds, dsf);

      // dQ^T += K^T . dS^T per d-tile (K^T A-fragments by PAIR-batched
      // tr-read — the quad form's 16 forced-live outputs pushed the
      // pinned-Q/dO register budget over at D=128)
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int t = 0; t < NDT; ++t) {
        const unsigned qstep = 128u * (D / 16);
        auto ka = [&](int s) {
          return ktr + 128u * (unsigned)((4 * s) * (D / 16) + 2 * t);
        };
        bf16x8_pair kq = tr_read_pair(ka(0), ka(0) + qstep,
                                      ka(1), ka(1) + qstep);
        dqacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kq.f[0], *(const bf16x8_t*)dsf[0], dqacc[t], 0, 0, 0);
        dqacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            kq.f[1], *(const bf16x8_t*)dsf[1], dqacc[t], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    // drains the next tile's glds (hidden under sub 1's MFMAs)
    __syncthreads();
    cur ^= 1;
  }

  if (qrow_g < S) {
    const long obase = (long)b * dqs.sb + (long)h * dqs.sh + (long)qrow_g * dqs.ss;
    // packed 8-B stores per r-quad (consecutive d in this lane)
    typedef short bf16x4_s __attribute__((ext_vector_type(4)));
    #pragma unroll
    for (int t = 0; t < NDT; ++t)
      #pragma unroll
      for (int rq = 0; rq < 4; ++rq) {
        bf16x4_s pk;
        #pragma unroll
        for (int j = 0; j < 4; ++j)
          pk[j] = f2bf(dqacc[t][rq * 4 + j]);
        *(bf16x4_s*)(dq + obase + 32 * t + 8 * rq + 4 * half) = pk;
      }
  }
}

// --------------------------------------------------------------- dKV
template <int D>
__global__ void __launch_bounds__(512, 2)
attn_bwd_dkv_8w_kernel(const short* __restrict__ dout,
                       const short* __restrict__ q,
                       const short* __restrict__ k,
                       const short* __restrict__ v,
                       const float* __restrict__ lse,
                       const float* __restrict__ delta,
                       short* __restrict__ dk, short* __restrict__ dv,
                       int B, int H, int Hkv, int S, float scale,
                       int causal, int nbx, Strides4 dos, Strides4 qs,
                       Strides4 ks, Strides4 vs, Strides4 dks, Strides4 dvs) {
  constexpr int NSL = D / 16;
  constexpr int NDT = D / 32;
  // [buf][Q 64-row tr pair | dO 64-row tr pair], double-buffered.
  // D=128: 2*2*64*128*2B = 128 KiB — one 8-wave block per CU (which is
  // the occupancy this kernel runs at anyway: 2 waves/SIMD).
  __shared__ __attribute__((aligned(16))) short bwd_lds[2 * 2 * 64 * D];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int l = tid & 63;
  const int lk = l & 31;
  const int half = l >> 5;

  int bx, bhk;
  xcd_tile_map(nbx, &bx, &bhk);
  const int b = bhk / Hkv, hk = bhk % Hkv;
  const int group = H / Hkv;
  const int k0 = bx * 256;
  if (k0 >= S) return;

  const long kbb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vbb = (long)b * vs.sb + (long)hk * vs.sh;
  const int krow_g = k0 + 8 * lk + wid;     // interleaved kv row
  const int krow = min(krow_g, S - 1);
  const float scale2 = scale * LOG2E;

  const unsigned qtr0 = tr_col_base<D>(bwd_lds, l);

  // K^T/V^T B-fragments: D=64 pinned; D=128 reloaded from L1 per sub
  constexpr bool KVREG = (D <= 64);
  const short* krow_ptr = k + kbb + (long)krow * ks.ss + 8 * half;
  const short* vrow_ptr = v + vbb + (long)krow * vs.ss + 8 * half;
  bf16x8_t ktf[KVREG ? NSL : 1], vtf[KVREG ? NSL : 1];
  if constexpr (KVREG) {
    #pragma unroll
    for (int s = 0; s < NSL; ++s) {
      ktf[s] = *(const bf16x8_t*)(krow_ptr + 16 * s);
      vtf[s] = *(const bf16x8_t*)(vrow_ptr + 16 * s);
    }
  }

  f32x16_t dvacc[NDT], dkacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) { dvacc[t] = {}; dkacc[t] = {}; }

  for (int g = 0; g < group; ++g) {
    const int h = hk * group + g;
    const long qb = (long)b * qs.sb + (long)h * qs.sh;
    const long dob = (long)b * dos.sb + (long)h * dos.sh;
    const long rhead = ((long)b * H + h) * S;
    const int qstart = causal ? k0 : 0;
    auto stage_qdo = [&](int buf, int c0s) {
      short* base = bwd_lds + buf * (2 * 64 * D);
      stage_tr64<D>(q + qb, qs.ss, c0s, S, base, tid);
      stage_tr64<D>(dout + dob, dos.ss, c0s, S, base + 64 * D, tid);
    };
    stage_qdo(0, qstart);
    __syncthreads();
    int cur = 0;
    for (int c0 = qstart; c0 < S; c0 += 64) {
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int cs = c0 + sub * 32;
        if (cs >= S) break;
        const unsigned bufoff = (unsigned)cur * (2 * 64 * D * 2)
                                + (unsigned)sub * (32 * D * 2);
        const unsigned qtr = qtr0 + bufoff;
        const unsigned dotr = qtr + (unsigned)(64 * D * 2);

        const int qrow = min(cs + lk, S - 1);
        // Q/dO A-fragments from the staged tr-images (contiguous row
        // runs of the tr layout) — no per-sub global operand loads
        const short* qsub = bwd_lds + ((unsigned)cur * (2 * 64 * D))
                            + (unsigned)sub * 32 * D;
        const short* dosub = qsub + 64 * D;
        const unsigned rowel = ((unsigned)(lk >> 2) * (D / 16)) * 64u +
                               (unsigned)(lk & 3) * 16u + 8u * half;
        f32x16_t sacc = {}, dpacc = {};
        __builtin_amdgcn_s_setprio(1);
        if constexpr (KVREG) {
          // two independent accumulator chains interleaved (hides
          // dependent-MFMA latency; fits at D=64)
          #pragma unroll
          for (int s = 0; s < NSL; ++s) {
            bf16x8_t qf = *(const bf16x8_t*)(qsub + rowel + s * 64);
            bf16x8_t dof = *(const bf16x8_t*)(dosub + rowel + s * 64);
            sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, ktf[s], sacc, 0, 0, 0);
            dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vtf[s], dpacc, 0, 0, 0);
          }
        } else {
          // D=128: split chains with a mid-loop scheduling split (the
          // scheduler otherwise buffers all k-slices' loads -> spill)
          #pragma unroll
          for (int s = 0; s < NSL; ++s) {
            bf16x8_t qf = *(const bf16x8_t*)(qsub + rowel + s * 64);
            const bf16x8_t kv = *(const bf16x8_t*)(krow_ptr + 16 * s);
            sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kv, sacc, 0, 0, 0);
            if (s == NSL / 2 - 1) asm volatile("" ::: "memory");
          }
          #pragma unroll
          for (int s = 0; s < NSL; ++s) {
            bf16x8_t dof = *(const bf16x8_t*)(dosub + rowel + s * 64);
            const bf16x8_t vv = *(const bf16x8_t*)(vrow_ptr + 16 * s);
            dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vv, dpacc, 0, 0, 0);
            if (s == NSL / 2 - 1) asm volatile("" ::: "memory");
          }
        }
        __builtin_amdgcn_s_setprio(0);

        // next q-tile's glds after this tile's LAST ordinary-load block
        if (sub == 1 && c0 + 64 < S) stage_qdo(cur ^ 1, c0 + 64);

        // per-sub q-row constants: lane j holds row cs+j's lse/delta,
        // broadcast per-reg via shfl (2 loads/lane vs 32 scalar loads)
        const float lse2_lane = lse[rhead + qrow] * LOG2E;
        const float del_lane = delta[rhead + qrow];

        // p overwrites sacc in place
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrel = (r & 3) + 8 * (r >> 2) + 4 * half;
          const int qg = cs + qrel;
          float sv = sacc[r] * scale2;
          const bool masked = qg >= S || (causal && krow_g > qg);
          const float lse_r = __shfl(lse2_lane, qrel, 64);
          sacc[r] = masked ? 0.f : __builtin_amdgcn_exp2f(sv - lse_r);
        }

        // ds overwrites dpacc in place
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrel = (r & 3) + 8 * (r >> 2) + 4 * half;
          const float del_r = __shfl(del_lane, qrel, 64);
          dpacc[r] = sacc[r] * (dpacc[r] - del_r) * scale;
        }

        unsigned pf[2][4], dsf[2][4];
        pack_frags((const float*)&sacc, pf);
        pack_frags((const float*)&dpacc, dsf);

        // dV^T += P^T . dO_cols ; dK^T += dS^T . Q_cols — PAIR-batched
        // tr-reads (4 per wait): the quad form's 16 forced-live output
        // regs were half the dkv spill at the D=128 budget edge
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int t = 0; t < NDT; ++t) {
          const unsigned qstep = 128u * (D / 16);
          const unsigned o0 = 128u * (unsigned)(2 * t);
          const unsigned o1 = o0 + 128u * (unsigned)(4 * (D / 16));  // s=1
          bf16x8_pair dop = tr_read_pair(
              dotr + o0, dotr + o0 + qstep, dotr + o1, dotr + o1 + qstep);
          dvacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)pf[0], dop.f[0], dvacc[t], 0, 0, 0);
          dvacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)pf[1], dop.f[1], dvacc[t], 0, 0, 0);
          bf16x8_pair qp = tr_read_pair(
              qtr + o0, qtr + o0 + qstep, qtr + o1, qtr + o1 + qstep);
          dkacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)dsf[0], qp.f[0], dkacc[t], 0, 0, 0);
          dkacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)dsf[1], qp.f[1], dkacc[t], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();  // tiles consumed; next q-tile's glds drained
      cur ^= 1;
    }
  }

  // epilogue: C layout lane=col=d(within tile), regs=rows=k (interleaved)
  #pragma unroll
  for (int t = 0; t < NDT; ++t)
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int kg = k0 + 8 * krel + wid;
      if (kg < S) {
        dk[(long)b * dks.sb + (long)hk * dks.sh + (long)kg * dks.ss + 32 * t + lk]
            = f2bf(dkacc[t][r]);
        dv[(long)b * dvs.sb + (long)hk * dvs.sh + (long)kg * dvs.ss + 32 * t + lk]
            = f2bf(dvacc[t][r]);
      }
    }
}

}  // namespace

void attn_bwd_dq_8w(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                    torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                    torch::Tensor dq, int B, int H, int Hkv, int S, int D,
                    double scale, bool causal) {
  auto stream = scaffold_stream();
  const int nbx = (S + 255) / 256;
  dim3 grid(nbx * B * H), block(512);
  const Strides4 dos = strides_of(dout), qs = strides_of(q), ks = strides_of(k),
                 vs = strides_of(v), dqs = strides_of(dq);
  #define LAUNCH_DQ8(DD)                                                       \
    hipLaunchKernelGGL((attn_bwd_dq_8w_kernel<DD>), grid, block, 0, stream,    \
                       (const short*)dout.data_ptr(),                          \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),      \
                       delta.data_ptr<float>(), (short*)dq.data_ptr(),         \
                       B, H, Hkv, S, (float)scale, causal ? 1 : 0, nbx,        \
                       dos, qs, ks, vs, dqs)
  if (D == 64) LAUNCH_DQ8(64);
  else if (D == 128) LAUNCH_DQ8(128);
  else TORCH_CHECK(false, "attn_bwd_dq_8w: unsupported D ", D);
  #undef LAUNCH_DQ8
  HIP_CHECK(hipGetLastError());
}

void attn_bwd_dkv_8w(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dk, torch::Tensor dv, int B, int H,
                     int Hkv, int S, int D, double scale, bool causal) {
  auto stream = scaffold_stream();
  const int nbx = (S + 255) / 256;
  dim3 grid(nbx * B * Hkv), block(512);
  const Strides4 dos = strides_of(dout), qs = strides_of(q), ks = strides_of(k),
                 vs = strides_of(v), dks = strides_of(dk), dvs = strides_of(dv);
  #define LAUNCH_DKV8(DD)                                                      \
    hipLaunchKernelGGL((attn_bwd_dkv_8w_kernel<DD>), grid, block, 0, stream,   \
                       (const short*)dout.data_ptr(),                          \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),      \
                       delta.data_ptr<float>(), (short*)dk.data_ptr(),         \
                       (short*)dv.data_ptr(), B, H, Hkv, S, (float)scale,      \
                       causal ? 1 : 0, nbx, dos, qs, ks, vs, dks, dvs)
  if (D == 64) LAUNCH_DKV8(64);
  else if (D == 128) LAUNCH_DKV8(128);
  else TORCH_CHECK(false, "attn_bwd_dkv_8w: unsupported D ", D);
  #undef LAUNCH_DKV8
  HIP_CHECK(hipGetLastError());
}

namespace {

// ------------------------------------------------- dKV, ROLE-SPLIT
// 8 waves over a 128-row kv block: waves 0-3 compute dV for rows
// k0 + 4*lk + (w&3), waves 4-7 compute dK for the SAME rows.  The
// per-wave register budget drops enough to PIN K and V fragments at
// D=128 (the all-roles kernel spills 124 B/lane there), each role
// reads only ONE tr-image in its accumulation (dO for dV, Q for dK),
// and the whole llama-shaped grid fits in ONE residency round.  Cost:
// S = QK^T is computed by both roles (+25% MFMA issue) — bought
// against a kernel at <10% matrix-pipe occupancy.
template <int D>
__global__ void __launch_bounds__(512, 2)
attn_bwd_dkv_rs_kernel(const short* __restrict__ dout,
                       const short* __restrict__ q,
                       const short* __restrict__ k,
                       const short* __restrict__ v,
                       const float* __restrict__ lse,
                       const float* __restrict__ delta,
                       short* __restrict__ dk, short* __restrict__ dv,
                       int B, int H, int Hkv, int S, float scale,
                       int causal, int nbx, Strides4 dos, Strides4 qs,
                       Strides4 ks, Strides4 vs, Strides4 dks,
                       Strides4 dvs) {
  constexpr int NSL = D / 16;
  constexpr int NDT = D / 32;
  __shared__ __attribute__((aligned(16))) short bwd_lds[2 * 2 * 64 * D];

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int role = wid >> 2;          // 0 = dV, 1 = dK
  const int wq = wid & 3;             // row-interleave slot
  const int l = tid & 63;
  const int lk = l & 31;
  const int half = l >> 5;

  int bx, bhk;
  xcd_tile_map(nbx, &bx, &bhk);
  const int b = bhk / Hkv, hk = bhk % Hkv;
  const int group = H / Hkv;
  const int k0 = bx * 128;
  if (k0 >= S) return;

  const long kbb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vbb = (long)b * vs.sb + (long)hk * vs.sh;
  const int krow_g = k0 + 4 * lk + wq;      // interleaved kv row
  const int krow = min(krow_g, S - 1);
  const float scale2 = scale * LOG2E;

  const unsigned qtr0 = tr_col_base<D>(bwd_lds, l);

  // role-local pinned operands: dV waves pin K (for S); dK waves pin
  // K AND V (for S and dP)
  bf16x8_t ktf[NSL], vtf[NSL];
  #pragma unroll
  for (int s = 0; s < NSL; ++s) {
    ktf[s] = *(const bf16x8_t*)(k + kbb + (long)krow * ks.ss + 16 * s + 8 * half);
    if (role == 1)
      vtf[s] = *(const bf16x8_t*)(v + vbb + (long)krow * vs.ss + 16 * s + 8 * half);
  }

  f32x16_t oacc[NDT];   // dV (role 0) or dK (role 1)
  #pragma unroll
  for (int t = 0; t < NDT; ++t) oacc[t] = {};

  for (int g = 0; g < group; ++g) {
    const int h = hk * group + g;
    const long qb = (long)b * qs.sb + (long)h * qs.sh;
    const long dob = (long)b * dos.sb + (long)h * dos.sh;
    const long rhead = ((long)b * H + h) * S;
    const int qstart = causal ? k0 : 0;
    auto stage_qdo = [&](int buf, int c0s) {
      short* base = bwd_lds + buf * (2 * 64 * D);
      stage_tr64<D>(q + qb, qs.ss, c0s, S, base, tid);
      stage_tr64<D>(dout + dob, dos.ss, c0s, S, base + 64 * D, tid);
    };
    stage_qdo(0, qstart);
    __syncthreads();
    int cur = 0;
    for (int c0 = qstart; c0 < S; c0 += 64) {
      #pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int cs = c0 + sub * 32;
        if (cs >= S) break;
        const unsigned bufoff = (unsigned)cur * (2 * 64 * D * 2)
                                + (unsigned)sub * (32 * D * 2);
        const unsigned qtr = qtr0 + bufoff;
        const unsigned dotr = qtr + (unsigned)(64 * D * 2);

        const int qrow = min(cs + lk, S - 1);
        const short* qsub = bwd_lds + ((unsigned)cur * (2 * 64 * D))
                            + (unsigned)sub * 32 * D;
        const short* dosub = qsub + 64 * D;
        const unsigned rowel = ((unsigned)(lk >> 2) * (D / 16)) * 64u +
                               (unsigned)(lk & 3) * 16u + 8u * half;

        f32x16_t sacc = {}, dpacc = {};
        __builtin_amdgcn_s_setprio(1);
        if (role == 0) {
          #pragma unroll
          for (int s = 0; s < NSL; ++s) {
            bf16x8_t qf = *(const bf16x8_t*)(qsub + rowel + s * 64);
            sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, ktf[s], sacc, 0, 0, 0);
          }
        } else {
          #pragma unroll
          for (int s = 0; s < NSL; ++s) {
            bf16x8_t qf = *(const bf16x8_t*)(qsub + rowel + s * 64);
            bf16x8_t dof = *(const bf16x8_t*)(dosub + rowel + s * 64);
            sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, ktf[s], sacc, 0, 0, 0);
            dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vtf[s], dpacc, 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);

        const float lse2_lane = lse[rhead + qrow] * LOG2E;
        const float del_lane = delta[rhead + qrow];

        if (sub == 1 && c0 + 64 < S) stage_qdo(cur ^ 1, c0 + 64);

        // p (and for dK, ds) in place
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrel = (r & 3) + 8 * (r >> 2) + 4 * half;
          const int qg = cs + qrel;
          float sv = sacc[r] * scale2;
          const bool masked = qg >= S || (causal && krow_g > qg);
          const float lse_r = __shfl(lse2_lane, qrel, 64);
          sacc[r] = masked ? 0.f : __builtin_amdgcn_exp2f(sv - lse_r);
        }
        if (role == 1) {
          #pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int qrel = (r & 3) + 8 * (r >> 2) + 4 * half;
            const float del_r = __shfl(del_lane, qrel, 64);
            sacc[r] = sacc[r] * (dpacc[r] - del_r) * scale;
          }
        }

        unsigned frag[2][4];
        pack_frags((const float*)&sacc, frag);

        // role 0: dV += P^T . dO_cols ; role 1: dK += dS^T . Q_cols
        const unsigned tr = role == 0 ? dotr : qtr;
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int t = 0; t < NDT; ++t) {
          const unsigned qstep = 128u * (D / 16);
          const unsigned o0 = tr + 128u * (unsigned)(2 * t);
          const unsigned o1 = o0 + 128u * (unsigned)(4 * (D / 16));
          bf16x8_pair cq = tr_read_pair(o0, o0 + qstep, o1, o1 + qstep);
          oacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)frag[0], cq.f[0], oacc[t], 0, 0, 0);
          oacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *(const bf16x8_t*)frag[1], cq.f[1], oacc[t], 0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
      cur ^= 1;
    }
  }

  // epilogue: role 0 writes dv, role 1 writes dk
  #pragma unroll
  for (int t = 0; t < NDT; ++t)
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int kg = k0 + 4 * krel + wq;
      if (kg < S) {
        if (role == 0)
          dv[(long)b * dvs.sb + (long)hk * dvs.sh + (long)kg * dvs.ss + 32 * t + lk]
              = f2bf(oacc[t][r]);
        else
          dk[(long)b * dks.sb + (long)hk * dks.sh + (long)kg * dks.ss + 32 * t + lk]
              = f2bf(oacc[t][r]);
      }
    }
}

}  // namespace

void attn_bwd_dkv_rs(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dk, torch::Tensor dv, int B, int H,
                     int Hkv, int S, int D, double scale, bool causal) {
  auto stream = scaffold_stream();
  const int nbx = (S + 127) / 128;
  dim3 grid(nbx * B * Hkv), block(512);
  const Strides4 dos = strides_of(dout), qs = strides_of(q), ks = strides_of(k),
                 vs = strides_of(v), dks = strides_of(dk), dvs = strides_of(dv);
  #define LAUNCH_RS(DD)                                                        \
    hipLaunchKernelGGL((attn_bwd_dkv_rs_kernel<DD>), grid, block, 0, stream,   \
                       (const short*)dout.data_ptr(),                          \
                       (const short*)q.data_ptr(), (const short*)k.data_ptr(), \
                       (const short*)v.data_ptr(), lse.data_ptr<float>(),      \
                       delta.data_ptr<float>(), (short*)dk.data_ptr(),         \
                       (short*)dv.data_ptr(), B, H, Hkv, S, (float)scale,      \
                       causal ? 1 : 0, nbx, dos, qs, ks, vs, dks, dvs)
  if (D == 64) LAUNCH_RS(64);
  else if (D == 128) LAUNCH_RS(128);
  else TORCH_CHECK(false, "attn_bwd_dkv_rs: unsupported D ", D);
  #undef LAUNCH_RS
  HIP_CHECK(hipGetLastError());
}
