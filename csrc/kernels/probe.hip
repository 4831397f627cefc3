// MFMA fragment-layout probes (test-only): compute one MFMA tile from
// row-major global A/B using the ASSUMED lane->element maps, so a GPU
// test can validate the maps against torch.matmul before the big MFMA
// kernels rely on them (guide §3: always check with asymmetric B).
//
// Assumed maps (gfx950, from cdna_hip_programming.md §3):
//  mfma_f32_32x32x16_bf16: A(32x16) lane l holds row=l&31, k=8*(l>>5)+j (j=0..7)
//                          B(16x32) lane l holds col=l&31, k=8*(l>>5)+j
//                          C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
//  mfma_f32_16x16x32_bf16: A(16x32) lane l holds row=l&15, k=8*(l>>4)+j
//                          B(32x16) lane l holds col=l&15, k=8*(l>>4)+j
//                          C/D: col=lane&15, row=(lane>>4)*4+reg

#include "../common.h"

namespace {

typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef float f32x16_t __attribute__((ext_vector_type(16)));
typedef short bf16x8_t __attribute__((ext_vector_type(8)));

__global__ void probe_mfma_32x32x16(const short* __restrict__ A,  // [32][16]
                                    const short* __restrict__ B,  // [16][32]
                                    float* __restrict__ C) {      // [32][32]
  const int l = threadIdx.x & 63;
  bf16x8_t a, b;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(l & 31) * 16 + 8 * (l >> 5) + j];
    b[j] = B[(8 * (l >> 5) + j) * 32 + (l & 31)];
  }
  f32x16_t c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    C[row * 32 + (l & 31)] = c[r];
  }
}

__global__ void probe_mfma_16x16x32(const short* __restrict__ A,  // [16][32]
                                    const short* __restrict__ B,  // [32][16]
                                    float* __restrict__ C) {      // [16][16]
  const int l = threadIdx.x & 63;
  bf16x8_t a, b;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(l & 15) * 32 + 8 * (l >> 4) + j];
    b[j] = B[(8 * (l >> 4) + j) * 16 + (l & 15)];
  }
  f32x4_t c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r)
    C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

// ds_read_b64_tr_b16 semantics probe: LDS holds bf16 elements with
// value == element index (0..1023); each lane issues ONE tr read at
// base + off[lane] bytes and dumps its 4 received elements.  A host
// test infers the lane->element map from the dump (guide T10's wording
// is ambiguous between a per-lane gather and a cross-lane exchange).
__global__ void probe_tr_kernel(const int* __restrict__ off,
                                short* __restrict__ out) {
  __shared__ __attribute__((aligned(16))) short lds[1024];
  const int l = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const unsigned addr = lds_byte_off(lds) + (unsigned)off[l];
  unsigned long long r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(addr));
  #pragma unroll
  for (int j = 0; j < 4; ++j)
    out[l * 4 + j] = (short)((r >> (16 * j)) & 0xffff);
}

}  // namespace

torch::Tensor probe_tr(torch::Tensor offsets) {
  CHECK_CUDA(offsets); CHECK_CONTIG(offsets);
  TORCH_CHECK(offsets.scalar_type() == torch::kInt32 && offsets.numel() == 64);
  auto out = torch::zeros({64, 4},
      torch::TensorOptions().dtype(torch::kInt16).device(offsets.device()));
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(probe_tr_kernel, dim3(1), dim3(64), 0, stream,
                     offsets.data_ptr<int>(), (short*)out.data_ptr());
  HIP_CHECK(hipGetLastError());
  return out;
}

torch::Tensor probe_mfma(torch::Tensor A, torch::Tensor B, long which) {
  CHECK_CUDA(A); CHECK_CONTIG(A); CHECK_CONTIG(B);
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16);
  auto stream = scaffold_stream();
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(A.device());
  if (which == 0) {
    auto C = torch::zeros({32, 32}, fopt);
    hipLaunchKernelGGL(probe_mfma_32x32x16, dim3(1), dim3(64), 0, stream,
                       (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                       C.data_ptr<float>());
    HIP_CHECK(hipGetLastError());
    return C;
  }
  auto C = torch::zeros({16, 16}, fopt);
  hipLaunchKernelGGL(probe_mfma_16x16x32, dim3(1), dim3(64), 0, stream,
                     (const short*)A.data_ptr(), (const short*)B.data_ptr(),
                     C.data_ptr<float>());
  HIP_CHECK(hipGetLastError());
  return C;
}
