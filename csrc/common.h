// Common helpers for the scaffold's CDNA4 (gfx950) HIP kernels.
// Target: MI355X only — wave64, 256 CUs / 8 XCDs, 160 KiB LDS/CU.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

// torch's current stream for this device (ROCm-native name; avoids the
// hipify rewrite of at::cuda::getCurrentCUDAStream)
inline hipStream_t scaffold_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define WAVE 64
#define DEVINL __device__ __forceinline__

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));       \
  } while (0)

#define CHECK_CUDA(x) TORCH_CHECK(x.is_cuda(), #x " must be a HIP tensor")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

using bf16 = __hip_bfloat16;

// vector types for wide loads (G13: always vectorize bf16 as short4/short8)
typedef short bf16x4 __attribute__((ext_vector_type(4)));   // 8 B
typedef short bf16x8 __attribute__((ext_vector_type(8)));   // 16 B
typedef float f32x4 __attribute__((ext_vector_type(4)));    // 16 B
typedef float f32x16 __attribute__((ext_vector_type(16)));

DEVINL float bf2f(short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}

DEVINL short f2bf(float f) {
  // round-to-nearest-even bf16 conversion
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int x = c.i;
  unsigned int rounded = x + 0x7FFF + ((x >> 16) & 1);
  if ((x & 0x7F800000u) == 0x7F800000u) rounded = x;  // inf/nan passthrough
  return (short)(rounded >> 16);
}

// wave-level reductions (wave64)
DEVINL float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEVINL float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block reduction via LDS; `red` shared array must hold >= blockDim.x/64 floats
DEVINL float block_reduce_sum(float v, float* red) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) red[wid] = v;
  __syncthreads();
  v = (lane < nwaves) ? red[lane] : 0.f;
  v = wave_reduce_sum(v);
  // broadcast through lane 0 of wave 0
  if (threadIdx.x == 0) red[0] = v;
  __syncthreads();
  float out = red[0];
  __syncthreads();
  return out;
}

DEVINL float block_reduce_max(float v, float* red) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) red[wid] = v;
  __syncthreads();
  v = (lane < nwaves) ? red[lane] : -INFINITY;
  v = wave_reduce_max(v);
  if (threadIdx.x == 0) red[0] = v;
  __syncthreads();
  float out = red[0];
  __syncthreads();
  return out;
}

// Element strides of a logical (B,H,S,D) tensor with contiguous D —
// lets attention/RoPE consume permuted views (e.g. the (B,S,H,D)
// physical layout of a QKV projection) without transpose copies.
struct Strides4 {
  long sb, sh, ss;
};

inline Strides4 strides_of(const torch::Tensor& t) {
  TORCH_CHECK(t.dim() == 4 && t.stride(3) == 1,
              "expected 4-D tensor with contiguous last dim");
  return {t.stride(0), t.stride(1), t.stride(2)};
}

// ---- LDS transpose-read machinery (guide T10, §2 "ds_read_b64_tr_b16")
//
// The tr-image of a [KT rows][D cols] bf16 tile packs 4x16 subtiles
// consecutively: element (k, d) lives at
//   el = ((k>>2) * (D/16) + (d>>4)) * 64 + (k&3)*16 + (d&15)
// written coalesced (16-aligned 8-element chunks).
//
// ds_read_b64_tr_b16 semantics — measured on hardware by probe_tr
// (benchmarks/tr_probe.py, 2026-09-13), NOT the guide's ambiguous
// wording: each lane reads 8 CONTIGUOUS bytes at its own 8-B-aligned
// address; within each 16-lane group the hardware then redistributes so
// lane l receives the bf16 at positions {16*j + (l&15), j=0..3} of the
// group's 64 concatenated elements (lane p's 4 els sit at positions
// 4p..4p+3).  To hand lane l column (l&15) of a [4 k-rows][16 d-cols]
// row-major subtile, lane p=(l&15) must therefore read
//   subtile_byte_base + (p>>2)*32 + (p&3)*8.

// async global->LDS copy, 16 B per lane (guide §5 common-mistake 1);
// dest is wave-uniform base + lane*16, source is per-lane.
DEVINL void glds16(const void* g, void* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)g,
      (__attribute__((address_space(3))) void*)l, 16, 0, 0);
}

DEVINL unsigned lds_byte_off(const void* p) {
  // generic -> LDS address space; uintptr of an AS(3) pointer is the
  // LDS byte offset the DS instructions take.
  return (unsigned)(uintptr_t)(__attribute__((address_space(3))) const void*)p;
}

// two tr reads (k-quads q4 and q4+1) -> one bf16x8 A/B fragment in k-order
DEVINL bf16x8 tr_read_frag(unsigned addr_q0, unsigned addr_q1) {
  unsigned long long a, b;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(a), "=&v"(b)
      : "v"(addr_q0), "v"(addr_q1));
  union { struct { unsigned long long lo, hi; } u; bf16x8 v; } r;
  r.u.lo = a;
  r.u.hi = b;
  return r.v;
}

// FOUR fragments (8 tr reads) behind ONE lgkmcnt wait: per-fragment
// waits serialized every consumer MFMA behind a full LDS round trip
// (measured MfmaUtil 6-10% on the attention kernels); batching the
// reads keeps the LDS pipe full.  §5.7 form (i): loads + wait in one
// asm statement, early-clobber outputs.
struct bf16x8_quad { bf16x8 f[4]; };

DEVINL bf16x8_quad tr_read_quad(unsigned a0, unsigned a1, unsigned b0,
                                unsigned b1, unsigned c0, unsigned c1,
                                unsigned d0, unsigned d1) {
  unsigned long long r0, r1, r2, r3, r4, r5, r6, r7;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %8\n\t"
      "ds_read_b64_tr_b16 %1, %9\n\t"
      "ds_read_b64_tr_b16 %2, %10\n\t"
      "ds_read_b64_tr_b16 %3, %11\n\t"
      "ds_read_b64_tr_b16 %4, %12\n\t"
      "ds_read_b64_tr_b16 %5, %13\n\t"
      "ds_read_b64_tr_b16 %6, %14\n\t"
      "ds_read_b64_tr_b16 %7, %15\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r0), "=&v"(r1), "=&v"(r2), "=&v"(r3),
        "=&v"(r4), "=&v"(r5), "=&v"(r6), "=&v"(r7)
      : "v"(a0), "v"(a1), "v"(b0), "v"(b1),
        "v"(c0), "v"(c1), "v"(d0), "v"(d1));
  union { struct { unsigned long long lo, hi; } u; bf16x8 v; } t;
  bf16x8_quad q;
  t.u.lo = r0; t.u.hi = r1; q.f[0] = t.v;
  t.u.lo = r2; t.u.hi = r3; q.f[1] = t.v;
  t.u.lo = r4; t.u.hi = r5; q.f[2] = t.v;
  t.u.lo = r6; t.u.hi = r7; q.f[3] = t.v;
  return q;
}

// TWO fragments (4 tr reads) behind one wait: half the forced-live
// output registers of tr_read_quad (16 -> 8) for kernels at the VGPR
// budget edge, still batched enough to keep the LDS pipe busy.
struct bf16x8_pair { bf16x8 f[2]; };

DEVINL bf16x8_pair tr_read_pair(unsigned a0, unsigned a1, unsigned b0,
                                unsigned b1) {
  unsigned long long r0, r1, r2, r3;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %4\n\t"
      "ds_read_b64_tr_b16 %1, %5\n\t"
      "ds_read_b64_tr_b16 %2, %6\n\t"
      "ds_read_b64_tr_b16 %3, %7\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(r0), "=&v"(r1), "=&v"(r2), "=&v"(r3)
      : "v"(a0), "v"(a1), "v"(b0), "v"(b1));
  union { struct { unsigned long long lo, hi; } u; bf16x8 v; } t;
  bf16x8_pair p;
  t.u.lo = r0; t.u.hi = r1; p.f[0] = t.v;
  t.u.lo = r2; t.u.hi = r3; p.f[1] = t.v;
  return p;
}

// XCD-affinity decomposition of a 1-D launch (guide T1): the dispatcher
// places block b on XCD b%8, so mapping every tile-block of one
// (batch, head) to ids ≡ bh (mod 8) co-locates them on ONE XCD — the
// K/V (fwd) or Q/dO (bwd) windows they all stage are then fetched from
// HBM once per chip instead of once per XCD (8x staging amplification
// measured as the dominant stall class of the bwd kernels).  Falls back
// to the identity map when 8 does not divide the bh count (speed-only
// either way — correctness never depends on placement).
DEVINL void xcd_tile_map(int nbx, int* bx, int* bh) {
  const unsigned id = blockIdx.x;
  const unsigned ubx = (unsigned)nbx;
  const unsigned gy = gridDim.x / ubx;
  if (gy % 8 == 0) {
    const unsigned xcd = id % 8, idx = id / 8;
    *bh = (int)(xcd + 8 * (idx / ubx));
    *bx = (int)(idx % ubx);
  } else {
    *bx = (int)(id % ubx);
    *bh = (int)(id / ubx);
  }
}

// Grid sizing guideline 11: cap grid, grid-stride the rest.
DEVINL long grid_stride() { return (long)gridDim.x * blockDim.x; }

inline int capped_blocks(long total_threads, int block = 256, int cap = 2048) {
  long b = (total_threads + block - 1) / block;
  return (int)std::min<long>(b, cap);
}
