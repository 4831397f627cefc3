// K2 — LayerNorm fwd/bwd.  One WAVE per row (4 rows per
// 256-thread block), wave64 shuffle reductions, bf16x8 vector loads
// (G13) when D % 8 == 0; scalar fallback otherwise.  dgamma/dbeta via
// per-block partials + a parallel column-reduce (64 cols x 4 row-slices
// per block) — the serial version was 22% of a GPT-2 step (see
// profiles/gpt2_kernel_stats_r1_before_fixes.csv).

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

// 8-wide vector access helpers: V=1 -> vectorized bf16x8/f32 path
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
layernorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                     const T* __restrict__ b, T* __restrict__ y,
                     float* __restrict__ mean_out, float* __restrict__ rstd_out,
                     int D, float eps, long R) {
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= R) return;
  const int lane = threadIdx.x & 63;
  const T* xr = x + row * (long)D;
  T* yr = y + row * (long)D;

  float sum = 0.f, sumsq = 0.f;
  if constexpr (VEC) {
    for (int i = lane * 8; i < D; i += 64 * 8) {
      float v[8];
      load8<T, true>(xr, i, v);
      #pragma unroll
      for (int j = 0; j < 8; ++j) { sum += v[j]; sumsq += v[j] * v[j]; }
    }
  } else {
    for (int i = lane; i < D; i += 64) {
      float v = load_f(xr, i);
      sum += v;
      sumsq += v * v;
    }
  }
  sum = wave_reduce_sum(sum);
  sumsq = wave_reduce_sum(sumsq);
  const float mean = sum / D;
  const float var = sumsq / D - mean * mean;
  const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  if constexpr (VEC) {
    for (int i = lane * 8; i < D; i += 64 * 8) {
      float v[8], wv[8], bv[8];
      load8<T, true>(xr, i, v);
      load8<T, true>(w, i, wv);
      load8<T, true>(b, i, bv);
      #pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (v[j] - mean) * rstd * wv[j] + bv[j];
      store8<T, true>(yr, i, v);
    }
  } else {
    for (int i = lane; i < D; i += 64) {
      float v = (load_f(xr, i) - mean) * rstd;
      store_f(yr, i, v * load_f(w, i) + load_f(b, i));
    }
  }
}

// dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat)); dyw = dy * w
// 4 waves per block, each wave owns whole rows; per-wave LDS column
// accumulators merged at block end into fp32 partials.
template <typename T, bool VEC>
__global__ void __launch_bounds__(256)
layernorm_bwd_dx_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                        const T* __restrict__ w, const float* __restrict__ mean,
                        const float* __restrict__ rstd, T* __restrict__ dx,
                        float* __restrict__ part,  // [nblocks][2D]: dw | db
                        int D, int rows_per_block, long R) {
  extern __shared__ float cols[];  // [4 waves][2][D]
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float* mycols = cols + (size_t)wid * 2 * D;
  for (int i = threadIdx.x; i < 4 * 2 * D; i += blockDim.x) cols[i] = 0.f;
  __syncthreads();

  const long row0 = (long)blockIdx.x * rows_per_block;
  const long row1 = min(row0 + rows_per_block, R);
  for (long row = row0 + wid; row < row1; row += 4) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float mu = mean[row], rs = rstd[row];

    float s1 = 0.f, s2 = 0.f;
    if constexpr (VEC) {
      for (int i = lane * 8; i < D; i += 64 * 8) {
        float xv[8], dyv[8], wv[8];
        load8<T, true>(xr, i, xv);
        load8<T, true>(dyr, i, dyv);
        load8<T, true>(w, i, wv);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xh = (xv[j] - mu) * rs;
          const float dyw = dyv[j] * wv[j];
          s1 += dyw;
          s2 += dyw * xh;
          mycols[i + j] += dyv[j] * xh;
          mycols[D + i + j] += dyv[j];
        }
      }
    } else {
      for (int i = lane; i < D; i += 64) {
        const float xh = (load_f(xr, i) - mu) * rs;
        const float dyv = load_f(dyr, i);
        const float dyw = dyv * load_f(w, i);
        s1 += dyw;
        s2 += dyw * xh;
        mycols[i] += dyv * xh;
        mycols[D + i] += dyv;
      }
    }
    s1 = wave_reduce_sum(s1) / D;
    s2 = wave_reduce_sum(s2) / D;
    if constexpr (VEC) {
      for (int i = lane * 8; i < D; i += 64 * 8) {
        float xv[8], dyv[8], wv[8], o[8];
        load8<T, true>(xr, i, xv);
        load8<T, true>(dyr, i, dyv);
        load8<T, true>(w, i, wv);
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const float xh = (xv[j] - mu) * rs;
          o[j] = rs * (dyv[j] * wv[j] - s1 - xh * s2);
        }
        store8<T, true>(dxr, i, o);
      }
    } else {
      for (int i = lane; i < D; i += 64) {
        const float xh = (load_f(xr, i) - mu) * rs;
        const float dyw = load_f(dyr, i) * load_f(w, i);
        store_f(dxr, i, rs * (dyw - s1 - xh * s2));
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float dwv = 0.f, dbv = 0.f;
    #pragma unroll
    for (int wv = 0; wv < 4; ++wv) {
      dwv += cols[(size_t)wv * 2 * D + i];
      dbv += cols[(size_t)wv * 2 * D + D + i];
    }
    part[(long)blockIdx.x * 2 * D + i] = dwv;
    part[(long)blockIdx.x * 2 * D + D + i] = dbv;
  }
}

// Two-stage parallel column reduce of a [nrows][C] fp32 partial matrix.
// Stage 1 slices the rows over gridDim.y so the launch fills the chip
// (the old single-stage version ran C/64 blocks = 12 CUs at D=768 and
// was 7% of the GPT-2 step); stage 2 folds the NS slice sums and splits
// the columns into the dw / db outputs.
__global__ void __launch_bounds__(256)
col_reduce_slice_kernel(const float* __restrict__ part, float* __restrict__ mid,
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
col_reduce_final_kernel(const float* __restrict__ mid, T* __restrict__ dw,
                        T* __restrict__ db, int nrows, int D) {
  __shared__ float acc[4][64];
  const int c = threadIdx.x & 63;
  const int col = blockIdx.x * 64 + c;
  const int slice = threadIdx.x >> 6;
  const int C = 2 * D;
  float s = 0.f;
  if (col < C)
    for (int b = slice; b < nrows; b += 4) s += mid[(long)b * C + col];
  acc[slice][c] = s;
  __syncthreads();
  if (slice == 0 && col < C) {
    const float v = acc[0][c] + acc[1][c] + acc[2][c] + acc[3][c];
    if (col < D) store_f(dw, col, v);
    else store_f(db, col - D, v);
  }
}

}  // namespace

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  CHECK_CUDA(x); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto y = torch::empty_like(x);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto mean = torch::empty({R}, fopt);
  auto rstd = torch::empty({R}, fopt);
  auto stream = scaffold_stream();
  dim3 grid((unsigned)((R + 3) / 4)), block(256);
  const bool vec = (D % 8) == 0;
  #define LN_FWD(T, V, PX)                                                     \
    hipLaunchKernelGGL((layernorm_fwd_kernel<T, V>), grid, block, 0, stream,   \
                       (const T*)x.data_ptr(), (const T*)w.data_ptr(),         \
                       (const T*)b.data_ptr(), (T*)y.data_ptr(),               \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), D,      \
                       (float)eps, R)
  if (x.scalar_type() == torch::kBFloat16) {
    if (vec) LN_FWD(short, true, );
    else LN_FWD(short, false, );
  } else {
    if (vec) LN_FWD(float, true, );
    else LN_FWD(float, false, );
  }
  #undef LN_FWD
  HIP_CHECK(hipGetLastError());
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  CHECK_CUDA(dy); CHECK_CONTIG(dy); CHECK_CONTIG(x);
  const long R = x.size(0);
  const int D = (int)x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::empty_like(w);
  auto db = torch::empty_like(w);
  auto fopt = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());

  const int target_blocks = 1024;  // >> 256 CUs
  const int rows_per_block = (int)std::max<long>(4, (R + target_blocks - 1) / target_blocks);
  const int nblocks = (int)((R + rows_per_block - 1) / rows_per_block);
  auto part = torch::empty({nblocks, 2L * D}, fopt);
  const int NS = nblocks > 64 ? 16 : 1;  // row slices for the chip-filling stage
  auto mid = torch::empty({NS, 2L * D}, fopt);
  const size_t lds = (size_t)4 * 2 * D * sizeof(float);
  TORCH_CHECK(lds <= 160 * 1024 - 1024, "layernorm_bwd: D too large for LDS path");

  auto stream = scaffold_stream();
  dim3 block(256);
  const bool vec = (D % 8) == 0;
  const int cr_grid = (2 * D + 63) / 64;
  #define LN_BWD(T, V)                                                         \
    do {                                                                       \
      hipLaunchKernelGGL((layernorm_bwd_dx_kernel<T, V>), dim3(nblocks), block,\
                         lds, stream, (const T*)dy.data_ptr(),                 \
                         (const T*)x.data_ptr(), (const T*)w.data_ptr(),       \
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),       \
                         (T*)dx.data_ptr(), part.data_ptr<float>(),            \
                         D, rows_per_block, R);                                \
      hipLaunchKernelGGL(col_reduce_slice_kernel, dim3(cr_grid, NS), block, 0, \
                         stream, part.data_ptr<float>(), mid.data_ptr<float>(),\
                         nblocks, 2 * D);                                      \
      hipLaunchKernelGGL((col_reduce_final_kernel<T>), dim3(cr_grid), block, 0,\
                         stream, mid.data_ptr<float>(), (T*)dw.data_ptr(),     \
                         (T*)db.data_ptr(), NS, D);                            \
    } while (0)
  if (x.scalar_type() == torch::kBFloat16) {
    if (vec) LN_BWD(short, true);
    else LN_BWD(short, false);
  } else {
    if (vec) LN_BWD(float, true);
    else LN_BWD(float, false);
  }
  #undef LN_BWD
  HIP_CHECK(hipGetLastError());
  return {dx, dw, db};
}
