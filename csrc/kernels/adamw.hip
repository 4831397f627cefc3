// K1 — fused multi-tensor AdamW: one launch updates every
// parameter tensor: m/v update + bias correction + decoupled decay +
// fp32-master update + bf16 working-copy refresh (K8 folded in).
//
// Pure bandwidth (~30 B/element): vectorized 4-wide, grid-stride over a
// virtual concatenation of all tensors; block->tensor mapping by binary
// search in an LDS copy of the per-tensor start table.  Grid capped and
// strided (guideline 11), >> 256 workgroups to fill 8 XCDs.

#include "../common.h"

namespace {

struct AdamwArgs {
  float lr, beta1, beta2, eps, wd, bc1_inv, bc2_inv, grad_scale;
};

constexpr int MAX_TENSORS = 4096;

// meta layout (int64 words): [0..n] starts_vec4 cumsum; then per-array
// pointer blocks: master[n], grad[n], m[n], v[n], out[n], numel[n]
__global__ void __launch_bounds__(256)
adamw_mt_kernel(const long* __restrict__ meta, int n, long total_vec4,
                AdamwArgs a, int grad_bf16, int has_out,
                const int* __restrict__ step_dev) {
  __shared__ long s_starts[MAX_TENSORS + 1];
  for (int i = threadIdx.x; i <= n; i += blockDim.x) s_starts[i] = meta[i];
  __syncthreads();

  if (step_dev != nullptr) {
    // hipGraph-replay path: the step lives on DEVICE (incremented by a
    // captured op), so bias corrections stay exact under replay where
    // host-baked args would freeze at the capture step.
    const float s = (float)*step_dev;
    a.bc1_inv = 1.0f / (1.0f - __powf(a.beta1, s));
    a.bc2_inv = 1.0f / (1.0f - __powf(a.beta2, s));
  }

  const long* p_master = meta + (n + 1);
  const long* p_grad = p_master + n;
  const long* p_m = p_grad + n;
  const long* p_v = p_m + n;
  const long* p_out = p_v + n;
  const long* p_numel = p_out + n;

  const long stride = (long)gridDim.x * blockDim.x;
  for (long vi = (long)blockIdx.x * blockDim.x + threadIdx.x; vi < total_vec4;
       vi += stride) {
    // binary search: largest t with s_starts[t] <= vi
    int lo = 0, hi = n - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (s_starts[mid] <= vi) lo = mid; else hi = mid - 1;
    }
    const int t = lo;
    const long e0 = (vi - s_starts[t]) * 4;   // element offset in tensor t
    const long numel = p_numel[t];
    float* master = (float*)p_master[t];
    float* m = (float*)p_m[t];
    float* v = (float*)p_v[t];
    const int tail = (e0 + 4 > numel) ? (int)(numel - e0) : 4;

    float g[4], pm[4], mm[4], vv[4];
    if (tail == 4) {
      if (grad_bf16) {
        bf16x4 gv = *(const bf16x4*)((const short*)p_grad[t] + e0);
        #pragma unroll
        for (int j = 0; j < 4; ++j) g[j] = bf2f(gv[j]) * a.grad_scale;
      } else {
        f32x4 gv = *(const f32x4*)((const float*)p_grad[t] + e0);
        #pragma unroll
        for (int j = 0; j < 4; ++j) g[j] = gv[j] * a.grad_scale;
      }
      *(f32x4*)pm = *(const f32x4*)(master + e0);
      *(f32x4*)mm = *(const f32x4*)(m + e0);
      *(f32x4*)vv = *(const f32x4*)(v + e0);
    } else {
      for (int j = 0; j < tail; ++j) {
        g[j] = (grad_bf16 ? bf2f(((const short*)p_grad[t])[e0 + j])
                          : ((const float*)p_grad[t])[e0 + j]) * a.grad_scale;
        pm[j] = master[e0 + j];
        mm[j] = m[e0 + j];
        vv[j] = v[e0 + j];
      }
    }

    short ob[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (j >= tail) break;
      mm[j] = a.beta1 * mm[j] + (1.f - a.beta1) * g[j];
      vv[j] = a.beta2 * vv[j] + (1.f - a.beta2) * g[j] * g[j];
      float mhat = mm[j] * a.bc1_inv;
      float vhat = vv[j] * a.bc2_inv;
      pm[j] = pm[j] * (1.f - a.lr * a.wd) - a.lr * mhat / (sqrtf(vhat) + a.eps);
      ob[j] = f2bf(pm[j]);
    }

    if (tail == 4) {
      *(f32x4*)(master + e0) = *(const f32x4*)pm;
      *(f32x4*)(m + e0) = *(const f32x4*)mm;
      *(f32x4*)(v + e0) = *(const f32x4*)vv;
      if (has_out) *(bf16x4*)((short*)p_out[t] + e0) = *(const bf16x4*)ob;
    } else {
      for (int j = 0; j < tail; ++j) {
        master[e0 + j] = pm[j];
        m[e0 + j] = mm[j];
        v[e0 + j] = vv[j];
        if (has_out) ((short*)p_out[t])[e0 + j] = ob[j];
      }
    }
  }
}

}  // namespace

// Build the device meta table ONCE (pointers are stable across steps:
// grads are bucket views, masters/moments never reallocate) so the hot
// step — and a hipGraph capture of it — carries no H2D copy.
torch::Tensor adamw_build_meta(std::vector<torch::Tensor> masters,
                               std::vector<torch::Tensor> grads,
                               std::vector<torch::Tensor> exp_avgs,
                               std::vector<torch::Tensor> exp_avg_sqs,
                               std::vector<torch::Tensor> outs) {
  const int n = (int)masters.size();
  TORCH_CHECK(n > 0 && n <= MAX_TENSORS, "fused_adamw: 1..4096 tensors, got ", n);
  const bool has_out = !outs.empty();
  TORCH_CHECK(!has_out || (int)outs.size() == n);

  std::vector<long> meta;
  meta.reserve((size_t)n * 7 + 1);
  long cum = 0;
  for (int i = 0; i < n; ++i) {
    meta.push_back(cum);
    cum += (masters[i].numel() + 3) / 4;
  }
  meta.push_back(cum);
  auto push_ptrs = [&](std::vector<torch::Tensor>& ts) {
    for (int i = 0; i < n; ++i) meta.push_back((long)ts[i].data_ptr());
  };
  push_ptrs(masters);
  push_ptrs(grads);
  push_ptrs(exp_avgs);
  push_ptrs(exp_avg_sqs);
  if (has_out) push_ptrs(outs);
  else for (int i = 0; i < n; ++i) meta.push_back(0);
  for (int i = 0; i < n; ++i) meta.push_back(masters[i].numel());

  auto opts = torch::TensorOptions().dtype(torch::kInt64);
  auto meta_cpu = torch::from_blob(meta.data(), {(long)meta.size()}, opts).clone();
  return meta_cpu.to(masters[0].device());
}

void fused_adamw_run(torch::Tensor meta_dev, long n, long total_vec4,
                     double lr, double beta1, double beta2, double eps,
                     double weight_decay, long step, double grad_scale,
                     bool grad_bf16, bool has_out,
                     c10::optional<torch::Tensor> step_dev) {
  AdamwArgs a;
  a.lr = (float)lr;
  a.beta1 = (float)beta1;
  a.beta2 = (float)beta2;
  a.eps = (float)eps;
  a.wd = (float)weight_decay;
  a.bc1_inv = (float)(1.0 / (1.0 - std::pow(beta1, (double)step)));
  a.bc2_inv = (float)(1.0 / (1.0 - std::pow(beta2, (double)step)));
  a.grad_scale = (float)grad_scale;

  const int* sd = nullptr;
  if (step_dev.has_value()) {
    TORCH_CHECK(step_dev->scalar_type() == torch::kInt32 && step_dev->is_cuda());
    sd = step_dev->data_ptr<int>();
  }
  const int block = 256;
  const int grid = capped_blocks(total_vec4, block, 2048);
  auto stream = scaffold_stream();
  hipLaunchKernelGGL(adamw_mt_kernel, dim3(grid), dim3(block), 0, stream,
                     meta_dev.data_ptr<long>(), (int)n, total_vec4, a,
                     grad_bf16 ? 1 : 0, has_out ? 1 : 0, sd);
  HIP_CHECK(hipGetLastError());
}

void fused_adamw(std::vector<torch::Tensor> masters,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<torch::Tensor> outs,  // bf16 working copies or empty
                 double lr, double beta1, double beta2, double eps,
                 double weight_decay, long step, double grad_scale) {
  const int n = (int)masters.size();
  TORCH_CHECK(n > 0, "fused_adamw: empty tensor list");
  long cum = 0;
  for (int i = 0; i < n; ++i) cum += (masters[i].numel() + 3) / 4;
  auto meta_dev = adamw_build_meta(masters, grads, exp_avgs, exp_avg_sqs, outs);
  fused_adamw_run(meta_dev, n, cum, lr, beta1, beta2, eps, weight_decay, step,
                  grad_scale, grads[0].scalar_type() == torch::kBFloat16,
                  !outs.empty(), c10::nullopt);
}
