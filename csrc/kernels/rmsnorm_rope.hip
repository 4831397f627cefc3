// K7 — RMSNorm fwd/bwd + RoPE apply (Llama path).
// Same structure as layernorm.hip: one wave per row, bf16x8 vector
// loads, parallel column-reduce for dgamma.  RoPE is a vectorized
// elementwise rotate with HOST-precomputed cos/sin tables.

#include "../common.h"

namespace {

template <typename T>
DEVINL float load_f(const T* p, long i);
template <> DEVINL float load_f<short>(const short* p, long i) { return bf2f(p[i]); }
template <> DEVINL float load_f<float>(const float* p, long i) { return p[i]; }

template <typename T>
DEVINL void store_f(T* p, long i, float v);
template <> DEVINL void store_f<short>(short* p, long i, float v) { p[i] = f2bf(v); }
template <> DEVINL void store_f<float>(float* p, long i, float v) { p[i] = v; }

template <typename T, bool VEC>
DEVINL void load8(const T* p, long i, float out[8]) {
  if constexpr (VEC && std::is_same<T, short>::value) {
    bf16x8 v = *(const bf16x8*)(p + i);
    #pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = bf2f(v[j]);
  } else {
    #pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = load_f(p, i + j);
  }
}

template <typename T, bool VEC>
DEVINL void store8(T* p, long i, const float in[8]) {
  if constexpr (VEC && std::is_same<T, short>::value) {
    bf16x8 v;
    #pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = f2bf(in[j]);
    *(bf16x8*)(p + i) = v;
  } else {
    #pragma unroll
    for (int j = 0; j < 8; ++j) store_f(p, i + j, in[j]);
  }
}

template <typename T, bool VEC>
__global__ void __launch_bounds__(256)
rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                   T* __restrict__ y, float* __restrict__ rstd_out,
                   int D, float eps, long R) {
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= R) return;
  const int lane = threadIdx.x & 63;
  const T* xr = x + row * (long)D;
  T* yr = y + row * (long)D;
  float ss = 0.f;
  if constexpr (VEC) {
    for (int i = lane * 8; i < D; i += 64 * 8) {
      float v[8];
      load8<T, true>(xr, i, v);
      #pragma unroll
      for (int j = 0; j < 8; ++j) ss += v[j] * v[j];
    }
  } else {
    for (int i = lane; i < D; i += 64) {
      float v = load_f(xr, i);
      ss += v * v;
    }
  }
  ss = wave_reduce_sum(ss);
  const float rstd = rsqrtf(ss / D + eps);
  if (lane == 0) rstd_out[row] = rstd;
  if constexpr (VEC) {
    for (int i = lane * 8; i < D; i += 64 * 8) {
      float v[8], wv[8];
      load8<T, true>(xr, i, v);
      load8<T, true>(w, i, wv);
      #pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = v[j] * rstd * wv[j];
      store8<T, true>(yr, i, v);
    }
  } else {
    for (int i = lane; i < D; i += 64)
      store_f(yr, i, load_f(xr, i) * rstd * load_f(w, i));
  }
}

// dx = rstd*dyw - x * rstd^3/D * sum(dyw * x);  dyw = dy*w
template <typename T, bool VEC>
__global__ void __launch_bounds__(256)
rmsnorm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                   const T* __restrict__ w, const float* __restrict__ rstd,
                   T* __restrict__ dx, float* __restrict__ dw_part,
                   int D, int rows_per_block, long R) {
  extern __shared__ float cols[];  // [4 waves][D]
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float* mycols = cols + (size_t)wid * D;
  for (int i = threadIdx.x; i < 4 * D; i += blockDim.x) cols[i] = 0.f;
  __syncthreads();

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, R);
  for (long row = row0 + wid; row < row1; row += 4) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float rs = rstd[row];
    float s = 0.f;
    if constexpr (VEC) {
      for (int i = lane * 8; i < D; i += 64 * 8) {
        float xv[8], dyv[8], wv[8];
        load8<T, true>(xr, i, xv);
        load8<T, true>(dyr, i, dyv);
        load8<T, true>(w, i, wv);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          s += dyv[j] * wv[j] * xv[j];
          mycols[i + j] += dyv[j] * xv[j] * rs;
        }
      }
    } else {
      for (int i = lane; i < D; i += 64) {
        const float xv = load_f(xr, i);
        const float dyv = load_f(dyr, i);
        s += dyv * load_f(w, i) * xv;
        mycols[i] += dyv * xv * rs;
      }
    }
    s = wave_reduce_sum(s);
    const float c = rs * rs * rs * s / D;
    if constexpr (VEC) {
      for (int i = lane * 8; i < D; i += 64 * 8) {
        float xv[8], dyv[8], wv[8], o[8];
        load8<T, true>(xr, i, xv);
        load8<T, true>(dyr, i, dyv);
        load8<T, true>(w, i, wv);
        #pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = rs * dyv[j] * wv[j] - xv[j] * c;
        store8<T, true>(dxr, i, o);
      }
    } else {
      for (int i = lane; i < D; i += 64) {
        const float dyw = load_f(dyr, i) * load_f(w, i);
        store_f(dxr, i, rs * dyw - load_f(xr, i) * c);
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float dwv = 0.f;
    #pragma unroll
    for (int wv = 0; wv < 4; ++wv) dwv += cols[(size_t)wv * D + i];
    dw_part[(long)blockIdx.x * D + i] = dwv;
  }
}

// Two-stage column reduce (same rationale as layernorm.hip: a single
// D/64-block launch under-fills the 256-CU chip).  Stage 1 slices rows
// over gridDim.y into mid[NS][D]; stage 2 folds the NS slices.
__global__ void __launch_bounds__(256)
col_reduce1_slice_kernel(const float* __restrict__ part, float* __restrict__ mid,
                         int nrows, int C) {
  __shared__ float acc[4][64];
  const int c = threadIdx.x & 63;
  const int col = blockIdx.x * 64 + c;
  const int slice = threadIdx.x >> 6;
  const int ns = gridDim.y;
  float s = 0.f;
  if (col < C)
    for (int b = blockIdx.y * 4 + slice; b < nrows; b += ns * 4)
      s += part[(long)b * C + col];
  acc[slice][c] = s;
  __syncthreads();
  if (slice == 0 && col < C)
    mid[(long)blockIdx.y * C + col] = acc[0][c] + acc[1][c] + acc[2][c] + acc[3][c];
}

template <typename T>
__global__ void __launch_bounds__(256)
col_reduce1_kernel(const float* __restrict__ part, T* __restrict__ out,
                   int nblocks, int D) {
  __shared__ float acc[4][64];
  const int c = threadIdx.x & 63;
  const int col = blockIdx.x * 64 + c;
  const int slice = threadIdx.x >> 6;
  float s = 0.f;
  if (col < D)
    for (int b = slice; b < nblocks; b += 4) s += part[(long)b * D + col];
  acc[slice][c] = s;
  __syncthreads();
  if (slice == 0 && col < D)
    store_f(out, col, acc[0][c] + acc[1][c] + acc[2][c] + acc[3][c]);
}

// x: logical (B,H,S,D), arbitrary strides with contiguous D; cos/sin:
// (S, D/2) fp32.  Rotates interleaved pairs (x[2i], x[2i+1]); inverse
// applies the transposed rotation (bwd).
template <typename T>
__global__ void __launch_bounds__(256)
rope_kernel(const T* __restrict__ x, const float* __restrict__ cos_t,
            const float* __restrict__ sin_t, T* __restrict__ y,
            int H, int S, int D, int inverse, long total,
            Strides4 xs, Strides4 ys) {
  const int half = D / 2;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += stride) {
    const int hpos = (int)(idx % half);
    const long srow = idx / half;          // (b, h, s) flattened
    const int spos = (int)(srow % S);
    const int h = (int)((srow / S) % H);
    const long b = srow / ((long)S * H);
    const long xbase = b * xs.sb + (long)h * xs.sh + (long)spos * xs.ss + 2 * hpos;
    const long ybase = b * ys.sb + (long)h * ys.sh + (long)spos * ys.ss + 2 * hpos;
    float c = cos_t[(long)spos * half + hpos];
    float s = sin_t[(long)spos * half + hpos];
    if (inverse) s = -s;
    const float x1 = load_f(x, xbase);
    const float x2 = load_f(x, xbase + 1);
    store_f(y, ybase, x1 * c - x2 * s);
    store_f(y, ybase + 1, x1 * s + x2 * c);
  }
}

}  // namespace

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto rstd = torch::empty({R}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)((R + 3) / 4)), block(256);
  const bool vec = (D % 8) == 0;
  #define RMS_FWD(T, V)                                                        \
    hipLaunchKernelGGL((rmsnorm_fwd_kernel<T, V>), grid, block, 0, stream,     \
                       (const T*)x.data_ptr(), (const T*)w.data_ptr(),         \
                       (T*)y.data_ptr(), rstd.data_ptr<float>(), D, (float)eps, R)
  if (x.scalar_type() == torch::kBFloat16) {
    if (vec) RMS_FWD(short, true);
    else RMS_FWD(short, false);
  } else {
    if (vec) RMS_FWD(float, true);
    else RMS_FWD(float, false);
  }
  #undef RMS_FWD
  HIP_CHECK(hipGetLastError());
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  CHECK_CUDA(dy); CHECK_CONTIG(dy); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  const int target_blocks = 1024;
  const int rows_per_block = (int)std::max<long>(4, (R + target_blocks - 1) / target_blocks);
  const int nblocks = (int)((R + rows_per_block - 1) / rows_per_block);
  auto dw_part = torch::empty({nblocks, D}, fopt);
  const int NS = nblocks > 64 ? 16 : 1;
  auto mid = torch::empty({NS, (long)D}, fopt);
  const size_t lds = (size_t)4 * D * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024 - 1024, "rmsnorm_bwd: D too large for LDS path");
  auto stream = scaffold_stream();
  dim3 block(256);
  const bool vec = (D % 8) == 0;
  const int cr_grid = (D + 63) / 64;
  #define RMS_BWD(T, V)                                                        \
    do {                                                                       \
      hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, V>), dim3(nblocks), block, lds,\
                         stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(),\
                         (const T*)w.data_ptr(), rstd.data_ptr<float>(),       \
                         (T*)dx.data_ptr(), dw_part.data_ptr<float>(),         \
                         D, rows_per_block, R);                                \
      hipLaunchKernelGGL(col_reduce1_slice_kernel, dim3(cr_grid, NS), block, 0,\
                         stream, dw_part.data_ptr<float>(),                    \
                         mid.data_ptr<float>(), nblocks, D);                   \
      hipLaunchKernelGGL((col_reduce1_kernel<T>), dim3(cr_grid), block, 0,     \
                         stream, mid.data_ptr<float>(), (T*)dw.data_ptr(),     \
                         NS, D);                                               \
    } while (0)
  if (x.scalar_type() == torch::kBFloat16) {
    if (vec) RMS_BWD(short, true);
    else RMS_BWD(short, false);
  } else {
    if (vec) RMS_BWD(float, true);
    else RMS_BWD(float, false);
  }
  #undef RMS_BWD
  HIP_CHECK(hipGetLastError());
  return {dx, dw};
}

torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cos_t, torch::Tensor sin_t,
                         bool inverse) {
  CHECK_CUDA(x);
  TORCH_CHECK(x.dim() == 4, "rope expects (B,H,S,D)");
  const int B = (int)x.size(0);
  const int H = (int)x.size(1);
  const int S = (int)x.size(2);
  const int D = (int)x.size(3);
  TORCH_CHECK(cos_t.size(0) >= S && cos_t.size(1) == D / 2, "rope table shape");
  auto xc = x.stride(3) == 1 ? x : x.contiguous();
  // output in (B,S,H,D) physical layout: feeds the strided attention path
  // and the un-transpose view backward without copies
  auto y = torch::empty({B, S, H, D}, x.options()).permute({0, 2, 1, 3});
  auto stream = scaffold_stream();
  const long total = (long)B * H * S * (D / 2);
  dim3 grid(capped_blocks(total, 256, 2048)), block(256);
  const Strides4 xs = strides_of(xc), ys = strides_of(y);
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rope_kernel<short>, grid, block, 0, stream,
                       (const short*)xc.data_ptr(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), (short*)y.data_ptr(),
                       H, S, D, inverse ? 1 : 0, total, xs, ys);
  } else {
    hipLaunchKernelGGL(rope_kernel<float>, grid, block, 0, stream,
                       xc.data_ptr<float>(), cos_t.data_ptr<float>(),
                       sin_t.data_ptr<float>(), y.data_ptr<float>(),
                       H, S, D, inverse ? 1 : 0, total, xs, ys);
  }
  HIP_CHECK(hipGetLastError());
  return y;
}
