// K4 — MFMA flash attention backward (bf16, 32x32x16 matrix cores).
//
// Recompute-based flash bwd on the fragment maps verified by probe.hip
// (same maps as attn_fwd_mfma.hip).  delta = rowsum(dO*O) comes from the
// strided wave kernel below.
//
// v2: the transpose-operand loads (K/Q/dO "column" fragments — 32-64
// scalar 2-B global loads per tile per lane in v1, the top cost of the
// GPT-2 step at 20.7%) now come from LDS tr-images staged cooperatively
// with global_load_lds and read with ds_read_b64_tr_b16 (guide T10) —
// 2 LDS instructions per fragment instead of 16 scalar loads.  Each
// wave's rows are INTERLEAVED (row = base + 4*i + wid) so all 4 waves
// share the same causal tile range at the block's staging barriers.
//
// dQ kernel  (wave = 32 interleaved q rows, loop kv tiles):
//   S^T  = mfma(K, Q^T)            C: lane=q, regs=k   (as fwd)
//   P^T  = exp(S^T*scale - lse[q])              lane-local
//   dP^T = mfma(V, dO^T)           C: lane=q, regs=k
//   dS^T = P^T*(dP^T - delta[q])*scale          lane-local
//   dQ^T += mfma(K^T_cols, pack(dS^T))  C: lane=q, regs=d  [K^T by tr-read]
//
// dKV kernel (wave = 32 interleaved kv rows, loop q tiles, loop GQA group):
//   S    = mfma(Q, K^T)            C: lane=k, regs=q
//   P    = exp(S*scale - lse[q_reg])   (lse/delta per-REG loads, L1-hot)
//   dV^T += mfma(pack(P^T), dO_cols)   C: lane=d, regs=k  [dO by tr-read]
//   dK^T += mfma(pack(dS^T), Q_cols)   C: lane=d, regs=k  [Q by tr-read]
//   (pack() = quad pack + permlane32_swap, T12 — identical transform for
//    a C-layout operand into an A/B fragment's 8-consecutive-k order.)

#include "../common.h"

namespace {

typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

// v_cvt_pk_bf16_f32: ONE VALU op per packed pair (guide T12; the r1
// manual round-to-nearest-even bit sequence was ~7 VALU per pair).
// The asm form constrains the register allocator: the dkv D=128 kernel
// (at the 256-reg edge) spilled 264 B/lane with it, so dkv keeps the
// manual sequence (template flag); dq uses cvt_pk (0 spill there).
DEVINL unsigned pack_bf16x2_cvt(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

DEVINL unsigned pack_bf16x2(float lo, float hi) {
  return (unsigned)(unsigned short)f2bf(lo) |
         ((unsigned)(unsigned short)f2bf(hi) << 16);
}

// C-layout f32[16] (lane=col, regs=rows) -> two 8-k B/A fragments
// (slice s covers rows 16s..16s+15 in the fragment's k order)
template <bool CVT = false>
DEVINL void pack_frags(const float p[16], unsigned out[2][4]) {
  #pragma unroll
  for (int s = 0; s < 2; ++s) {
    unsigned ax = CVT ? pack_bf16x2_cvt(p[8 * s + 0], p[8 * s + 1])
                      : pack_bf16x2(p[8 * s + 0], p[8 * s + 1]);
    unsigned ay = CVT ? pack_bf16x2_cvt(p[8 * s + 2], p[8 * s + 3])
                      : pack_bf16x2(p[8 * s + 2], p[8 * s + 3]);
    unsigned bx = CVT ? pack_bf16x2_cvt(p[8 * s + 4], p[8 * s + 5])
                      : pack_bf16x2(p[8 * s + 4], p[8 * s + 5]);
    unsigned by = CVT ? pack_bf16x2_cvt(p[8 * s + 6], p[8 * s + 7])
                      : pack_bf16x2(p[8 * s + 6], p[8 * s + 7]);
    auto rx = __builtin_amdgcn_permlane32_swap(ax, bx, false, false);
    auto ry = __builtin_amdgcn_permlane32_swap(ay, by, false, false);
    out[s][0] = rx[0];
    out[s][1] = ry[0];
    out[s][2] = rx[1];
    out[s][3] = ry[1];
  }
}

// cooperative glds16 stage of a 32-row global tile into a tr-image
// (layout in common.h); rows clamped at S-1 (padded rows are masked out
// of P downstream).
template <int D>
DEVINL void stage_tr(const short* __restrict__ src, long row_stride, int r0,
                     int S, short* lds, int wid, int l) {
  constexpr int NSLOT = 32 * D / 8 / 256;
  #pragma unroll
  for (int it = 0; it < NSLOT; ++it) {
    const int s8 = (it * 4 + wid) * 64 + l;
    const int sub = s8 >> 3;
    const int rem8 = s8 & 7;
    const int row = min(r0 + 4 * (sub / (D / 16)) + (rem8 >> 1), S - 1);
    const int d0 = 16 * (sub % (D / 16)) + 8 * (rem8 & 1);
    glds16(src + (long)row * row_stride + d0, lds + (long)(it * 4 + wid) * 512);
  }
}

// per-lane tr-read BASE for the column fragments of a tr-image: the
// (t, s) fragment of rows 16s+8*half+0..7 at col 32t+(l&31) reads at
//   base + 128*((4s)*(D/16) + 2t)   [+128*(D/16) for the second quad]
// — one VGPR per image, constant offsets at the read sites (the
// 16-entry address-array version cost 32 VGPRs in dkv and spilled).
template <int D>
DEVINL unsigned tr_col_base(const void* img, int l) {
  const int p = l & 15;
  return lds_byte_off(img) + 128u * ((l & 31) >> 4)
         + (p >> 2) * 32 + (p & 3) * 8
         + ((l >> 5) ? 128u * 2 * (D / 16) : 0);
}

// delta = rowsum(dO * O) - dlse (one wave per row, stride-aware).
// dlse (may be null) is the gradient of the attention's logsumexp
// output — context-parallel partial-attention merges differentiate
// through lse (scaffold/dist/cp.py); folding it here makes the dq/dkv
// kernels' (dP - delta) term exact with zero extra passes.
template <int D>
__global__ void __launch_bounds__(256)
attn_delta_strided_kernel(const short* __restrict__ dout,
                          const short* __restrict__ o,
                          const float* __restrict__ dlse,
                          float* __restrict__ delta, int H, int S, long rows,
                          Strides4 dos, Strides4 osr) {
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const int s = (int)(row % S);
  const int h = (int)((row / S) % H);
  const long b = row / ((long)S * H);
  const long dbase = b * dos.sb + (long)h * dos.sh + (long)s * dos.ss;
  const long obase = b * osr.sb + (long)h * osr.sh + (long)s * osr.ss;
  // pairwise bf16->f32 (shl for lo, mask for hi: 1 VALU/elem — this
  // kernel measured 98% VALUBusy with the scalar 2-3 op/elem unpack)
  float acc = 0.f;
  for (int d = lane * 8; d < D; d += 64 * 8) {
    typedef unsigned uint4_t __attribute__((ext_vector_type(4)));
    uint4_t dv = *(const uint4_t*)(dout + dbase + d);
    uint4_t ov = *(const uint4_t*)(o + obase + d);
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      union { unsigned i; float f; } dl, dh, ol, oh;
      dl.i = dv[j] << 16; dh.i = dv[j] & 0xffff0000u;
      ol.i = ov[j] << 16; oh.i = ov[j] & 0xffff0000u;
      acc = fmaf(dl.f, ol.f, acc);
      acc = fmaf(dh.f, oh.f, acc);
    }
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) delta[row] = acc - (dlse ? dlse[row] : 0.f);
}

template <int D>
__global__ void __launch_bounds__(256)
attn_bwd_dq_mfma_kernel(const short* __restrict__ dout,
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
  // double-buffered tr image: tile c0+32 streams while c0 computes
  __shared__ __attribute__((aligned(16))) short ktile[2 * 32 * D];

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
  const long dob = (long)b * dos.sb + (long)h * dos.sh;
  const long kbb = (long)b * ks.sb + (long)hk * ks.sh;
  const long vbb = (long)b * vs.sb + (long)hk * vs.sh;
  const long rhead = ((long)b * H + h) * S;

  const int qrow_g = q0 + 4 * lq + wid;     // interleaved q row
  const int qrow = min(qrow_g, S - 1);
  bf16x8_t qf[NSL], dof[NSL];
  #pragma unroll
  for (int s = 0; s < NSL; ++s) {
    qf[s] = *(const bf16x8_t*)(q + qb + (long)qrow * qs.ss + 16 * s + 8 * half);
    dof[s] = *(const bf16x8_t*)(dout + dob + (long)qrow * dos.ss + 16 * s + 8 * half);
  }
  // exp2-fold: softmax in the base-2 domain (v_exp_f32 IS exp2 — one
  // VALU multiply per score saved); lse converted per row, not element
  const float my_lse2 = lse[rhead + qrow] * 1.4426950408889634f;
  const float my_delta = delta[rhead + qrow];
  const float scale2 = scale * 1.4426950408889634f;

  const unsigned ktr0 = tr_col_base<D>(ktile, l);

  f32x16_t dqacc[NDT];
  #pragma unroll
  for (int t = 0; t < NDT; ++t) dqacc[t] = {};

  const int kv_end = causal ? min(S, q0 + 128) : S;
  stage_tr<D>(k + kbb, ks.ss, 0, S, ktile, wid, l);
  __syncthreads();
  int cur = 0;
  for (int c0 = 0; c0 < kv_end; c0 += 32) {
    const unsigned ktr = ktr0 + (unsigned)cur * (32 * D * 2);

    const int krow = min(c0 + lq, S - 1);
    // K A-fragments from the staged tr-image (contiguous row runs of
    // the tr layout — same trick as dkv); V has no image, stays global
    const short* kimg = ktile + (long)cur * 32 * D;
    const unsigned rowel = ((unsigned)(lq >> 2) * (D / 16)) * 64u +
                           (unsigned)(lq & 3) * 16u + 8u * half;
    f32x16_t sacc = {}, dpacc = {};
    #pragma unroll
    for (int s = 0; s < NSL; ++s) {
      bf16x8_t kf = *(const bf16x8_t*)(kimg + rowel + s * 64);
      bf16x8_t vf = *(const bf16x8_t*)(v + vbb + (long)krow * vs.ss + 16 * s + 8 * half);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc, 0, 0, 0);
      dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[s], dpacc, 0, 0, 0);
    }

    // next tile's glds issued AFTER the block with ordinary global
    // loads (kf/vf rows) — see the fwd kernel's trap-(b) note
    if (c0 + 32 < kv_end)
      stage_tr<D>(k + kbb, ks.ss, c0 + 32, S, ktile + (cur ^ 1) * 32 * D, wid, l);

    const bool need_mask = (causal && c0 + 31 > qrow_g) || (c0 + 31 >= S);
    float ds[16];
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      float sv = sacc[r] * scale2;
      if (need_mask) {
        const int kg = c0 + krel;
        if (kg >= S || (causal && kg > qrow_g)) sv = -1e30f;
      }
      const float p = __builtin_amdgcn_exp2f(sv - my_lse2);
      ds[r] = p * (dpacc[r] - my_delta) * scale;
    }

    unsigned dsf[2][4];
    pack_frags(ds, dsf);

    // dQ^T += K^T . dS^T  per d-tile; K^T A-fragments by batched
    // tr-read (one LDS wait per 4 fragments — common.h)
    __builtin_amdgcn_s_setprio(1);
    #pragma unroll
    for (int t2 = 0; t2 < NDT; t2 += 2) {
      const unsigned qstep = 128u * (D / 16);
      auto ka = [&](int t, int s) {
        return ktr + 128u * (unsigned)((4 * s) * (D / 16) + 2 * t);
      };
      bf16x8_quad kq = tr_read_quad(
          ka(t2, 0), ka(t2, 0) + qstep, ka(t2, 1), ka(t2, 1) + qstep,
          ka(t2 + 1, 0), ka(t2 + 1, 0) + qstep,
          ka(t2 + 1, 1), ka(t2 + 1, 1) + qstep);
      dqacc[t2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          kq.f[0], *(const bf16x8_t*)dsf[0], dqacc[t2], 0, 0, 0);
      dqacc[t2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          kq.f[1], *(const bf16x8_t*)dsf[1], dqacc[t2], 0, 0, 0);
      dqacc[t2 + 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          kq.f[2], *(const bf16x8_t*)dsf[0], dqacc[t2 + 1], 0, 0, 0);
      dqacc[t2 + 1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          kq.f[3], *(const bf16x8_t*)dsf[1], dqacc[t2 + 1], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    // drains the next tile's glds (hidden under this tile's MFMAs)
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

template <int D>
__global__ void __launch_bounds__(256, 2)
attn_bwd_dkv_mfma_kernel(const short* __restrict__ dout,
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
  // [buf][Q tile | dO tile], double-buffered at BOTH head dims: tile
  // t+1's glds issue after this tile's LAST ordinary global load (the
  // lse/delta lane loads moved above the issue point — an ordinary
  // load beside an in-flight glds drains vmcnt(0), guide trap b, which
  // was also why the r1 D=128 double-buffer attempt measured
  // net-slower) and drain under the dV/dK MFMAs.  PMC r2: the
  // synchronous D=128 staging parked waves 73% of the time.
  __shared__ __attribute__((aligned(16))) short bwd_lds[2 * 2 * 32 * D];

  const int wid = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
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
  const int krow_g = k0 + 4 * lk + wid;     // interleaved kv row
  const int krow = min(krow_g, S - 1);

  const unsigned qtr0 = tr_col_base<D>(bwd_lds, l);
  const unsigned dotr0 = qtr0 + (unsigned)(32 * D * 2);

  // K^T/V^T B-fragments for S = Q.K^T / dP = dO.V^T (col=k=lane,
  // kdim=d) — contiguous loads.  D=64: pinned in registers; D=128:
  // reloaded from L1 per q-tile (64 VGPRs saved -> 2 waves/SIMD; the
  // lane's K/V rows stay L1-resident across the whole q loop).
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
      short* base = bwd_lds + buf * (2 * 32 * D);
      stage_tr<D>(q + qb, qs.ss, c0s, S, base, wid, l);
      stage_tr<D>(dout + dob, dos.ss, c0s, S, base + 32 * D, wid, l);
    };
    stage_qdo(0, qstart);
    __syncthreads();
    int cur = 0;
    for (int c0 = qstart; c0 < S; c0 += 32) {
      const unsigned bufoff = (unsigned)cur * (2 * 32 * D * 2);
      const unsigned qtr = qtr0 + bufoff;
      const unsigned dotr = dotr0 + bufoff;

      // S = Q_tile . K^T and dP = dO . V^T.  The Q/dO A-fragments come
      // from the STAGED tr-images: a row fragment (k, 16s+8h..+8) is a
      // contiguous 16-B run of the tr layout at element
      // (k>>2)*(D/16)*64 + s*64 + (k&3)*16 + 8h — so the per-tile
      // global Q/dO operand loads (an L2 round trip + vmcnt wait per
      // k-slice, a chunk of the 69%-parked time) become ds_reads of
      // data the prefetch already placed.
      const int qrow = min(c0 + lk, S - 1);
      const short* qimg = bwd_lds + (bufoff >> 1);
      const short* doimg = qimg + 32 * D;
      const unsigned rowel = ((unsigned)(lk >> 2) * (D / 16)) * 64u +
                             (unsigned)(lk & 3) * 16u + 8u * half;
      f32x16_t sacc = {}, dpacc = {};
      __builtin_amdgcn_s_setprio(1);
      // both head dims: the two accumulator chains interleave (hides
      // dependent-MFMA latency); with the operands now coming from LDS
      // the D=128 split-chain workaround for global-load buffering is
      // no longer needed
      #pragma unroll
      for (int s = 0; s < NSL; ++s) {
        bf16x8_t qf = *(const bf16x8_t*)(qimg + rowel + s * 64);
        bf16x8_t dof = *(const bf16x8_t*)(doimg + rowel + s * 64);
        const bf16x8_t kv = KVREG ? ktf[s] : *(const bf16x8_t*)(krow_ptr + 16 * s);
        const bf16x8_t vv = KVREG ? vtf[s] : *(const bf16x8_t*)(vrow_ptr + 16 * s);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qf, kv, sacc, 0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dof, vv, dpacc, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // per-tile q-row constants loaded ONCE per wave (lane j holds row
      // c0+j's values) and broadcast per-reg via shfl — 2 loads instead
      // of 32 scalar L1 loads per lane per tile, and they sit BEFORE
      // the glds issue so no ordinary load drains the prefetch (trap b)
      const float lse_lane = lse[rhead + qrow];
      const float del_lane = delta[rhead + qrow];

      // next q-tile's glds: issued after the tile's LAST ordinary
      // global load; drained under the dV/dK MFMAs + end barrier
      if (c0 + 32 < S) stage_qdo(cur ^ 1, c0 + 32);

      // p overwrites sacc in place (register reuse)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrel = (r & 3) + 8 * (r >> 2) + 4 * half;
        const int qg = c0 + qrel;
        float sv = sacc[r] * scale;
        const bool masked = qg >= S || (causal && krow_g > qg);
        const float lse_r = __shfl(lse_lane, qrel, 64);
        sacc[r] = masked ? 0.f : __expf(sv - lse_r);
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

      // dV^T += P^T . dO_cols ; dK^T += dS^T . Q_cols  (C: lane=d, regs=k)
      // column B-fragments by batched tr-read: one LDS wait per d-tile
      // pair covers both tensors' 2 k-slices (common.h)
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int t = 0; t < NDT; ++t) {
        const unsigned qstep = 128u * (D / 16);
        const unsigned o0 = 128u * (unsigned)(2 * t);
        const unsigned o1 = o0 + 128u * (unsigned)(4 * (D / 16));  // s=1
        bf16x8_quad cq = tr_read_quad(
            dotr + o0, dotr + o0 + qstep, dotr + o1, dotr + o1 + qstep,
            qtr + o0, qtr + o0 + qstep, qtr + o1, qtr + o1 + qstep);
        dvacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *(const bf16x8_t*)pf[0], cq.f[0], dvacc[t], 0, 0, 0);
        dvacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *(const bf16x8_t*)pf[1], cq.f[1], dvacc[t], 0, 0, 0);
        dkacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *(const bf16x8_t*)dsf[0], cq.f[2], dkacc[t], 0, 0, 0);
        dkacc[t] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            *(const bf16x8_t*)dsf[1], cq.f[3], dkacc[t], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      __syncthreads();  // tiles consumed; next tile's glds drained
      cur ^= 1;
    }
  }

  // epilogue: C layout lane=col=d(within tile), regs=rows=k (interleaved)
  #pragma unroll
  for (int t = 0; t < NDT; ++t)
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int krel = (r & 3) + 8 * (r >> 2) + 4 * half;
      const int kg = k0 + 4 * krel + wid;
      if (kg < S) {
        dk[(long)b * dks.sb + (long)hk * dks.sh + (long)kg * dks.ss + 32 * t + lk]
            = f2bf(dkacc[t][r]);
        dv[(long)b * dvs.sb + (long)hk * dvs.sh + (long)kg * dvs.ss + 32 * t + lk]
            = f2bf(dvacc[t][r]);
      }
    }
}

}  // namespace

// 8-wave variants (attn_bwd_8w.hip)
void attn_bwd_dq_8w(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                    torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                    torch::Tensor dq, int B, int H, int Hkv, int S, int D,
                    double scale, bool causal);
void attn_bwd_dkv_8w(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dk, torch::Tensor dv, int B, int H,
                     int Hkv, int S, int D, double scale, bool causal);
void attn_bwd_dkv_rs(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dk, torch::Tensor dv, int B, int H,
                     int Hkv, int S, int D, double scale, bool causal);

namespace {
// SCAFFOLD_ATTN_BWD = 4w | 8w | rs | auto.  rs = role-split dkv
// (dV and dK waves over the same rows) with the auto dq choice.
// Read per call so tests can exercise the kernels in one process.
int attn_bwd_mode() {
  const char* e = getenv("SCAFFOLD_ATTN_BWD");
  if (e && strcmp(e, "4w") == 0) return 4;
  if (e && strcmp(e, "8w") == 0) return 8;
  if (e && strcmp(e, "rs") == 0) return 5;
  return 0;
}
}  // namespace

void attn_bwd_mfma(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, torch::Tensor o, torch::Tensor lse,
                   torch::Tensor delta, torch::Tensor dq, torch::Tensor dk,
                   torch::Tensor dv, int B, int H, int Hkv, int S, int D,
                   double scale, bool causal, const float* dlse_ptr) {
  auto stream = scaffold_stream();
  dim3 block(256);
  const int nbx4 = (S + 127) / 128;
  dim3 grid_q(nbx4 * B * H);
  dim3 grid_kv(nbx4 * B * Hkv);
  const Strides4 dos = strides_of(dout), qs = strides_of(q), ks = strides_of(k),
                 vs = strides_of(v), dqs = strides_of(dq), dks = strides_of(dk),
                 dvs = strides_of(dv), osr = strides_of(o);
  const long rows = (long)B * H * S;
  dim3 dgrid((unsigned)((rows + 3) / 4));
  if (D == 64)
    hipLaunchKernelGGL((attn_delta_strided_kernel<64>), dgrid, block, 0, stream,
                       (const short*)dout.data_ptr(), (const short*)o.data_ptr(),
                       dlse_ptr, delta.data_ptr<float>(), H, S, rows, dos, osr);
  else
    hipLaunchKernelGGL((attn_delta_strided_kernel<128>), dgrid, block, 0, stream,
                       (const short*)dout.data_ptr(), (const short*)o.data_ptr(),
                       dlse_ptr, delta.data_ptr<float>(), H, S, rows, dos, osr);
  const int mode = attn_bwd_mode();
  // auto (re-measured r2): role-split dkv at D=128 (llama train 6.13 ->
  // 5.30 ms); 8-wave at D=64 (rs measured 0.71 vs 8w 0.63 ms there)
  if (mode == 5 || (mode == 0 && D == 128 && S >= 128)) {
    // role-split dkv + the auto dq choice
    if (S >= 256 && D <= 64) {
      attn_bwd_dq_8w(dout, q, k, v, lse, delta, dq, B, H, Hkv, S, D, scale,
                     causal);
    } else {
      #define LAUNCH_DQ4(DD)                                                   \
        hipLaunchKernelGGL((attn_bwd_dq_mfma_kernel<DD>),                      \
                           dim3(nbx4 * B * H), dim3(256), 0, stream,           \
                           (const short*)dout.data_ptr(),                      \
                           (const short*)q.data_ptr(),                         \
                           (const short*)k.data_ptr(),                         \
                           (const short*)v.data_ptr(), lse.data_ptr<float>(),  \
                           delta.data_ptr<float>(), (short*)dq.data_ptr(),     \
                           B, H, Hkv, S, (float)scale, causal ? 1 : 0, nbx4,   \
                           dos, qs, ks, vs, dqs)
      if (D == 64) LAUNCH_DQ4(64);
      else LAUNCH_DQ4(128);
      #undef LAUNCH_DQ4
    }
    attn_bwd_dkv_rs(dout, q, k, v, lse, delta, dk, dv, B, H, Hkv, S, D,
                    scale, causal);
    HIP_CHECK(hipGetLastError());
    return;
  }
  // auto: 8-wave at D=64 (measured ~even to +2%); 4-wave at D=128 where
  // the 8-wave dkv's 36 B/lane spill measured net-slower (NOTES.md r2)
  if ((mode == 8) || (mode == 0 && S >= 256 && D <= 64)) {
    attn_bwd_dq_8w(dout, q, k, v, lse, delta, dq, B, H, Hkv, S, D, scale,
                   causal);
    attn_bwd_dkv_8w(dout, q, k, v, lse, delta, dk, dv, B, H, Hkv, S, D,
                    scale, causal);
    HIP_CHECK(hipGetLastError());
    return;
  }
  #define LAUNCH_D(DD)                                                         \
    do {                                                                       \
      hipLaunchKernelGGL((attn_bwd_dq_mfma_kernel<DD>), grid_q, block, 0,      \
                         stream, (const short*)dout.data_ptr(),                \
                         (const short*)q.data_ptr(), (const short*)k.data_ptr(),\
                         (const short*)v.data_ptr(), lse.data_ptr<float>(),    \
                         delta.data_ptr<float>(), (short*)dq.data_ptr(),       \
                         B, H, Hkv, S, (float)scale, causal ? 1 : 0, nbx4,     \
                         dos, qs, ks, vs, dqs);                                \
      hipLaunchKernelGGL((attn_bwd_dkv_mfma_kernel<DD>), grid_kv, block, 0,    \
                         stream, (const short*)dout.data_ptr(),                \
                         (const short*)q.data_ptr(), (const short*)k.data_ptr(),\
                         (const short*)v.data_ptr(), lse.data_ptr<float>(),    \
                         delta.data_ptr<float>(), (short*)dk.data_ptr(),       \
                         (short*)dv.data_ptr(), B, H, Hkv, S, (float)scale,    \
                         causal ? 1 : 0, nbx4, dos, qs, ks, vs, dks, dvs);     \
    } while (0)
  if (D == 64) LAUNCH_D(64);
  else if (D == 128) LAUNCH_D(128);
  else TORCH_CHECK(false, "attn_bwd_mfma: unsupported D ", D);
  #undef LAUNCH_D
  HIP_CHECK(hipGetLastError());
}
