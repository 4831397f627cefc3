// K3 — fused softmax-cross-entropy: fwd computes per-row
// LSE online (single read of the logits) + NLL; bwd fuses
// dlogits = (softmax - onehot) * scale, recomputed from logits + lse,
// so the softmax matrix is never materialized.  bf16x8 vector loads
// (G13) with a scalar tail for V % 8 != 0.

#include "../common.h"

namespace {

template <typename T>
DEVINL float load_f(const T* p, long i);
template <> DEVINL float load_f<short>(const short* p, long i) { return bf2f(p[i]); }
template <> DEVINL float load_f<float>(const float* p, long i) { return p[i]; }

template <typename T>
DEVINL void load8v(const T* p, long i, float out[8]) {
  if constexpr (std::is_same<T, short>::value) {
    bf16x8 v = *(const bf16x8*)(p + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = bf2f(v[j]);
  } else {
    f32x4 a = *(const f32x4*)(p + i);
    f32x4 b = *(const f32x4*)(p + i + 4);
    #pragma unroll
    for (int j = 0; j < 4; ++j) { out[j] = a[j]; out[4 + j] = b[j]; }
  }
}

// one workgroup per row; online max/sum over V in one pass
template <typename T>
__global__ void __launch_bounds__(256)
softmax_xent_fwd_kernel(const T* __restrict__ logits,
                        const long* __restrict__ targets,
                        float* __restrict__ loss, float* __restrict__ lse_out,
                        int V) {
  __shared__ float red[8];
  const long row = blockIdx.x;
  const T* lr = logits + row * (long)V;
  const int V8 = V & ~7;

  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x * 8; i < V8; i += blockDim.x * 8) {
    float v[8];
    load8v(lr, i, v);
    float vm = v[0];
    #pragma unroll
    for (int j = 1; j < 8; ++j) vm = fmaxf(vm, v[j]);
    if (vm > m) {
      s *= __expf(m - vm);
      m = vm;
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) s += __expf(v[j] - m);
  }
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x) {
    float v = load_f(lr, i);
    if (v > m) {
      s = s * __expf(m - v) + 1.f;
      m = v;
    } else {
      s += __expf(v - m);
    }
  }
  float M = block_reduce_max(m, red);
  float S = block_reduce_sum(m == -INFINITY ? 0.f : s * __expf(m - M), red);
  const float lse = M + __logf(S);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    loss[row] = lse - load_f(lr, targets[row]);
  }
}

template <typename T>
__global__ void __launch_bounds__(256)
softmax_xent_bwd_kernel(const T* __restrict__ logits,
                        const long* __restrict__ targets,
                        const float* __restrict__ lse, T* __restrict__ dlogits,
                        int V, float scale) {
  const long row = blockIdx.x;
  const T* lr = logits + row * (long)V;
  T* dr = dlogits + row * (long)V;
  const float l = lse[row];
  const long tgt = targets[row];
  const int V8 = V & ~7;
  for (int i = threadIdx.x * 8; i < V8; i += blockDim.x * 8) {
    float v[8];
    load8v(lr, i, v);
    if constexpr (std::is_same<T, short>::value) {
      bf16x8 o;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = (__expf(v[j] - l) - ((long)(i + j) == tgt ? 1.f : 0.f)) * scale;
        o[j] = f2bf(d);
      }
      *(bf16x8*)(dr + i) = o;
    } else {
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        dr[i + j] = (__expf(v[j] - l) - ((long)(i + j) == tgt ? 1.f : 0.f)) * scale;
    }
  }
  for (int i = V8 + threadIdx.x; i < V; i += blockDim.x) {
    float p = __expf(load_f(lr, i) - l);
    float d = (p - ((long)i == tgt ? 1.f : 0.f)) * scale;
    if constexpr (std::is_same<T, short>::value) dr[i] = f2bf(d);
    else dr[i] = d;
  }
}

}  // namespace

std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits, torch::Tensor targets) {
  CHECK_CUDA(logits); CHECK_CONTIG(logits);
  TORCH_CHECK(targets.scalar_type() == torch::kInt64);
  const long R = logits.size(0);
  const int V = (int)logits.size(1);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(logits.device());
  auto loss = torch::empty({R}, fopt);
  auto lse = torch::empty({R}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)R), block(256);
  if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(softmax_xent_fwd_kernel<short>, grid, block, 0, stream,
                       (const short*)logits.data_ptr(), targets.data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), V);
  } else {
    hipLaunchKernelGGL(softmax_xent_fwd_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       loss.data_ptr<float>(), lse.data_ptr<float>(), V);
  }
  HIP_CHECK(hipGetLastError());
  return {loss, lse};
}

torch::Tensor softmax_xent_bwd(torch::Tensor logits, torch::Tensor targets,
                               torch::Tensor lse, double dloss) {
  CHECK_CUDA(logits); CHECK_CONTIG(logits);
  const long R = logits.size(0);
  const int V = (int)logits.size(1);
  auto dlogits = torch::empty_like(logits);
  // mean reduction: d(mean)/dlogit_row = dloss / R
  const float scale = (float)(dloss / (double)R);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)R), block(256);
  if (logits.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(softmax_xent_bwd_kernel<short>, grid, block, 0, stream,
                       (const short*)logits.data_ptr(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), (short*)dlogits.data_ptr(), V, scale);
  } else {
    hipLaunchKernelGGL(softmax_xent_bwd_kernel<float>, grid, block, 0, stream,
                       logits.data_ptr<float>(), targets.data_ptr<long>(),
                       lse.data_ptr<float>(), dlogits.data_ptr<float>(), V, scale);
  }
  HIP_CHECK(hipGetLastError());
  return dlogits;
}
