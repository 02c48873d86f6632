// Fused LayerNorm / RMSNorm, gfx950. bf16/fp32 I/O, fp32 stats.
// Memory-bound: the row kernels read each operand ONCE into registers
// (16-B vector loads), keep it there across the stats reduction and the
// write pass, and optionally fuse the residual add:
//   fwd: s = x (+ res); y = LN(s); s written out for the residual stream
//   bwd: dx = ln_bwd(dy) (+ dsum)   — the residual-stream gradient join
// One 256-thread block per row (grid-stride); compile-time (VEC, ITERS)
// covers the transformer hidden sizes (1024/2048/4096/5120/8192); a
// generic two-pass kernel backstops any other H.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;

template <typename T, int VEC>
DEV_INLINE void load_vec(const T* p, float* v) {
  if constexpr (sizeof(T) == 2) {
    if constexpr (VEC == 8) {
      short8v pk = *reinterpret_cast<const short8v*>(p);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = bf_raw2f(((unsigned short*)&pk)[j]);
    } else {
      short4v pk = *reinterpret_cast<const short4v*>(p);
#pragma unroll
      for (int j = 0; j < 4; ++j) v[j] = bf_raw2f(((unsigned short*)&pk)[j]);
    }
  } else {
#pragma unroll
    for (int j = 0; j < VEC; j += 4) {
      float4v pk = *reinterpret_cast<const float4v*>(p + j);
#pragma unroll
      for (int k = 0; k < 4; ++k) v[j + k] = ((float*)&pk)[k];
    }
  }
}

template <typename T, int VEC>
DEV_INLINE void store_vec(T* p, const float* v) {
  if constexpr (sizeof(T) == 2) {
    if constexpr (VEC == 8) {
      short8v out;
#pragma unroll
      for (int j = 0; j < 8; j += 2) {
        unsigned int u = cvt_pk_bf16(v[j], v[j + 1]);
        ((unsigned short*)&out)[j] = (unsigned short)u;
        ((unsigned short*)&out)[j + 1] = (unsigned short)(u >> 16);
      }
      *reinterpret_cast<short8v*>(p) = out;
    } else {
      short4v out;
#pragma unroll
      for (int j = 0; j < 4; j += 2) {
        unsigned int u = cvt_pk_bf16(v[j], v[j + 1]);
        ((unsigned short*)&out)[j] = (unsigned short)u;
        ((unsigned short*)&out)[j + 1] = (unsigned short)(u >> 16);
      }
      *reinterpret_cast<short4v*>(p) = out;
    }
  } else {
#pragma unroll
    for (int j = 0; j < VEC; j += 4) {
      float4v out;
#pragma unroll
      for (int k = 0; k < 4; ++k) ((float*)&out)[k] = v[j + k];
      *reinterpret_cast<float4v*>(p + j) = out;
    }
  }
}

// ---------------- forward (register-resident fast path) ----------------
// H == BLOCK * VEC * ITERS. Optional fused residual: s = x + res.
template <typename T, int VEC, int ITERS>
__global__ __launch_bounds__(BLOCK) void ln_fwd_reg_kernel(
    const T* __restrict__ x, const T* __restrict__ res,
    const T* __restrict__ w, const T* __restrict__ b, T* __restrict__ y,
    T* __restrict__ sum_out, float* __restrict__ mean,
    float* __restrict__ rstd, int N, int H, float eps, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  // w / b are row-invariant: load once per block
  float wv[ITERS][VEC], bv[ITERS][VEC];
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    int i = (it * BLOCK + threadIdx.x) * VEC;
    load_vec<T, VEC>(w + i, wv[it]);
    if (b) load_vec<T, VEC>(b + i, bv[it]);
  }
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    float v[ITERS][VEC];
    float s = 0.f, ss = 0.f;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * BLOCK + threadIdx.x) * VEC;
      load_vec<T, VEC>(xr + i, v[it]);
      if (res) {
        float rv[VEC];
        load_vec<T, VEC>(res + (long)row * H + i, rv);
#pragma unroll
        for (int j = 0; j < VEC; ++j) v[it][j] += rv[j];
        store_vec<T, VEC>(sum_out + (long)row * H + i, v[it]);
        if constexpr (sizeof(T) == 2) {
          // normalize the ROUNDED sum — backward recomputes xhat from
          // the stored bf16 s, so stats must see the same values
#pragma unroll
          for (int j = 0; j < VEC; ++j) v[it][j] = bf_raw2f(f2bf_raw(v[it][j]));
        }
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j) { s += v[it][j]; ss += v[it][j] * v[it][j]; }
    }
    s = block_reduce_sum<BLOCK>(s, sred);
    __syncthreads();
    ss = block_reduce_sum<BLOCK>(ss, sred);
    float mu = rms ? 0.f : s / H;
    float var = ss / H - mu * mu;
    float rs = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (!rms && mean) mean[row] = mu;
      rstd[row] = rs;
    }
    T* yr = y + (long)row * H;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * BLOCK + threadIdx.x) * VEC;
      float o[VEC];
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = (v[it][j] - mu) * rs * wv[it][j] + (b ? bv[it][j] : 0.f);
      store_vec<T, VEC>(yr + i, o);
    }
    __syncthreads();
  }
}

// generic fallback (two passes over x, any H % 4 == 0)
template <typename T, int VEC>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                              const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              T* __restrict__ sum_out, float* __restrict__ mean,
                              float* __restrict__ rstd,
                              int N, int H, float eps, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    float s = 0.f, ss = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float v[VEC];
      load_vec<T, VEC>(xr + i, v);
      if (res) {
        float rv[VEC];
        load_vec<T, VEC>(res + (long)row * H + i, rv);
#pragma unroll
        for (int j = 0; j < VEC; ++j) v[j] += rv[j];
        store_vec<T, VEC>(sum_out + (long)row * H + i, v);
        if constexpr (sizeof(T) == 2) {
#pragma unroll
          for (int j = 0; j < VEC; ++j) v[j] = bf_raw2f(f2bf_raw(v[j]));
        }
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j) { s += v[j]; ss += v[j] * v[j]; }
    }
    s = block_reduce_sum<BLOCK>(s, sred);
    __syncthreads();
    ss = block_reduce_sum<BLOCK>(ss, sred);
    float mu = rms ? 0.f : s / H;
    float var = ss / H - mu * mu;
    float rs = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (!rms && mean) mean[row] = mu;
      rstd[row] = rs;
    }
    const T* sr = res ? sum_out + (long)row * H : xr;
    T* yr = y + (long)row * H;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float v[VEC], wv[VEC], o[VEC];
      load_vec<T, VEC>(sr + i, v);
      load_vec<T, VEC>(w + i, wv);
      if (b) {
        float bv[VEC];
        load_vec<T, VEC>(b + i, bv);
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] = (v[j] - mu) * rs * wv[j] + bv[j];
      } else {
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] = (v[j] - mu) * rs * wv[j];
      }
      store_vec<T, VEC>(yr + i, o);
    }
    __syncthreads();
  }
}

// ---------------- backward: dx (row-parallel, register-resident) -------
// Optional fused addend (residual-stream gradient): dx += dsum.
template <typename T, int VEC, int ITERS>
__global__ __launch_bounds__(BLOCK) void ln_bwd_dx_reg_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, const T* __restrict__ dsum,
    T* __restrict__ dx, int N, int H, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  float wv[ITERS][VEC];
#pragma unroll
  for (int it = 0; it < ITERS; ++it)
    load_vec<T, VEC>(w + (it * BLOCK + threadIdx.x) * VEC, wv[it]);
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    const T* dyr = dy + (long)row * H;
    float mu = rms ? 0.f : mean[row];
    float rs = rstd[row];
    float xh[ITERS][VEC], dw[ITERS][VEC];
    float c1 = 0.f, c2 = 0.f;  // mean(dy*w), mean(dy*w*xhat)
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * BLOCK + threadIdx.x) * VEC;
      float xv[VEC], dv[VEC];
      load_vec<T, VEC>(xr + i, xv);
      load_vec<T, VEC>(dyr + i, dv);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        xh[it][j] = (xv[j] - mu) * rs;
        dw[it][j] = dv[j] * wv[it][j];
        c1 += dw[it][j];
        c2 += dw[it][j] * xh[it][j];
      }
    }
    c1 = block_reduce_sum<BLOCK>(c1, sred) / H;
    __syncthreads();
    c2 = block_reduce_sum<BLOCK>(c2, sred) / H;
    T* dxr = dx + (long)row * H;
#pragma unroll
    for (int it = 0; it < ITERS; ++it) {
      int i = (it * BLOCK + threadIdx.x) * VEC;
      float o[VEC];
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        o[j] = rms ? (dw[it][j] - xh[it][j] * c2) * rs
                   : (dw[it][j] - c1 - xh[it][j] * c2) * rs;
      if (dsum) {
        float av[VEC];
        load_vec<T, VEC>(dsum + (long)row * H + i, av);
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] += av[j];
      }
      store_vec<T, VEC>(dxr + i, o);
    }
    __syncthreads();
  }
}

// generic fallback
template <typename T, int VEC>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                 const T* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 const T* __restrict__ dsum,
                                 T* __restrict__ dx, int N, int H, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    const T* dyr = dy + (long)row * H;
    float mu = rms ? 0.f : mean[row];
    float rs = rstd[row];
    float c1 = 0.f, c2 = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float xv[VEC], dv[VEC], wv[VEC];
      load_vec<T, VEC>(xr + i, xv);
      load_vec<T, VEC>(dyr + i, dv);
      load_vec<T, VEC>(w + i, wv);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xhj = (xv[j] - mu) * rs;
        float dwj = dv[j] * wv[j];
        c1 += dwj;
        c2 += dwj * xhj;
      }
    }
    c1 = block_reduce_sum<BLOCK>(c1, sred) / H;
    __syncthreads();
    c2 = block_reduce_sum<BLOCK>(c2, sred) / H;
    T* dxr = dx + (long)row * H;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float xv[VEC], dv[VEC], wv[VEC], o[VEC];
      load_vec<T, VEC>(xr + i, xv);
      load_vec<T, VEC>(dyr + i, dv);
      load_vec<T, VEC>(w + i, wv);
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xhj = (xv[j] - mu) * rs;
        o[j] = rms ? (dv[j] * wv[j] - xhj * c2) * rs
                   : (dv[j] * wv[j] - c1 - xhj * c2) * rs;
      }
      if (dsum) {
        float av[VEC];
        load_vec<T, VEC>(dsum + (long)row * H + i, av);
#pragma unroll
        for (int j = 0; j < VEC; ++j) o[j] += av[j];
      }
      store_vec<T, VEC>(dxr + i, o);
    }
    __syncthreads();
  }
}

// ---------------- backward: dw/db (column-parallel, chunked atomics) ----
template <typename T>
__global__ void ln_bwd_dwdb_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ rstd,
                                   float* __restrict__ dw, float* __restrict__ db,
                                   int N, int H, int rows_per_chunk, bool rms) {
  // grid: (ceil(H/BLOCK), nchunks); thread owns one column in one row-chunk
  int col = blockIdx.x * BLOCK + threadIdx.x;
  if (col >= H) return;
  int r0 = blockIdx.y * rows_per_chunk;
  int r1 = min(N, r0 + rows_per_chunk);
  float sw = 0.f, sb = 0.f;
  for (int r = r0; r < r1; ++r) {
    float xv, dv;
    if constexpr (sizeof(T) == 2) {
      xv = bf_raw2f(((const unsigned short*)x)[(long)r * H + col]);
      dv = bf_raw2f(((const unsigned short*)dy)[(long)r * H + col]);
    } else {
      xv = ((const float*)x)[(long)r * H + col];
      dv = ((const float*)dy)[(long)r * H + col];
    }
    float mu = rms ? 0.f : mean[r];
    float xh = (xv - mu) * rstd[r];
    sw += dv * xh;
    sb += dv;
  }
  atomicAdd(&dw[col], sw);
  if (db) atomicAdd(&db[col], sb);
}

// (VEC, ITERS) fast-path table for transformer hidden sizes
#define LN_FOR_EACH_SHAPE(F) \
  F(4, 1)  /* H = 1024 */    \
  F(8, 1)  /* H = 2048 */    \
  F(8, 2)  /* H = 4096 */    \
  F(4, 5)  /* H = 5120 */    \
  F(8, 4)  /* H = 8192 */

template <typename T>
void ln_fwd_launch(const T* x, const T* res, const T* w, const T* b, T* y,
                   T* sum_out, float* mean, float* rstd, int N, int H,
                   float eps, bool rms, hipStream_t stream) {
  int grid = std::min(N, 4096);
  TORCH_CHECK(H % 4 == 0, "hidden size must be divisible by 4");
#define LN_FWD_CASE(V, I)                                                   \
  if (H == BLOCK * V * I) {                                                 \
    hipLaunchKernelGGL((ln_fwd_reg_kernel<T, V, I>), dim3(grid),            \
                       dim3(BLOCK), 0, stream, x, res, w, b, y, sum_out,    \
                       mean, rstd, N, H, eps, rms);                         \
    return;                                                                 \
  }
  LN_FOR_EACH_SHAPE(LN_FWD_CASE)
#undef LN_FWD_CASE
  if (H % (8 * 4) == 0)
    hipLaunchKernelGGL((ln_fwd_kernel<T, 8>), dim3(grid), dim3(BLOCK), 0,
                       stream, x, res, w, b, y, sum_out, mean, rstd, N, H,
                       eps, rms);
  else
    hipLaunchKernelGGL((ln_fwd_kernel<T, 4>), dim3(grid), dim3(BLOCK), 0,
                       stream, x, res, w, b, y, sum_out, mean, rstd, N, H,
                       eps, rms);
}

template <typename T>
void ln_bwd_dx_launch(const T* dy, const T* x, const T* w, const float* mean,
                      const float* rstd, const T* dsum, T* dx, int N, int H,
                      bool rms, hipStream_t stream) {
  int grid = std::min(N, 4096);
#define LN_BWD_CASE(V, I)                                                   \
  if (H == BLOCK * V * I) {                                                 \
    hipLaunchKernelGGL((ln_bwd_dx_reg_kernel<T, V, I>), dim3(grid),         \
                       dim3(BLOCK), 0, stream, dy, x, w, mean, rstd, dsum,  \
                       dx, N, H, rms);                                      \
    return;                                                                 \
  }
  LN_FOR_EACH_SHAPE(LN_BWD_CASE)
#undef LN_BWD_CASE
  if (H % (8 * 4) == 0)
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 8>), dim3(grid), dim3(BLOCK), 0,
                       stream, dy, x, w, mean, rstd, dsum, dx, N, H, rms);
  else
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 4>), dim3(grid), dim3(BLOCK), 0,
                       stream, dy, x, w, mean, rstd, dsum, dx, N, H, rms);
}

}  // namespace

// ---------------- torch bindings ----------------

static std::vector<torch::Tensor> ln_fwd_common(torch::Tensor x,
                                                c10::optional<torch::Tensor> res,
                                                torch::Tensor w,
                                                c10::optional<torch::Tensor> b,
                                                double eps, bool rms) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  int N = x.size(0), H = x.size(1);
  auto y = torch::empty_like(x);
  torch::Tensor sum_out;
  bool has_res = res.has_value();
  if (has_res) {
    TORCH_CHECK(res->sizes() == x.sizes() && res->is_contiguous());
    sum_out = torch::empty_like(x);
  }
  auto mean = rms ? torch::Tensor()
                  : torch::empty({N}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    using T = __hip_bfloat16;
    ln_fwd_launch<T>((const T*)x.data_ptr(),
                     has_res ? (const T*)res->data_ptr() : nullptr,
                     (const T*)w.data_ptr(),
                     b.has_value() ? (const T*)b->data_ptr() : nullptr,
                     (T*)y.data_ptr(),
                     has_res ? (T*)sum_out.data_ptr() : nullptr,
                     rms ? nullptr : mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), N, H, (float)eps, rms, stream);
  } else {
    ln_fwd_launch<float>(x.data_ptr<float>(),
                         has_res ? res->data_ptr<float>() : nullptr,
                         w.data_ptr<float>(),
                         b.has_value() ? b->data_ptr<float>() : nullptr,
                         y.data_ptr<float>(),
                         has_res ? sum_out.data_ptr<float>() : nullptr,
                         rms ? nullptr : mean.data_ptr<float>(),
                         rstd.data_ptr<float>(), N, H, (float)eps, rms,
                         stream);
  }
  std::vector<torch::Tensor> out;
  out.push_back(y);
  if (!rms) out.push_back(mean);
  out.push_back(rstd);
  if (has_res) out.push_back(sum_out);
  return out;
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  return ln_fwd_common(x, c10::nullopt, w, b, eps, false);
}

// residual-fused: returns {y, mean, rstd, sum} with sum = x + res
std::vector<torch::Tensor> layernorm_fwd_residual(torch::Tensor x,
                                                  torch::Tensor res,
                                                  torch::Tensor w,
                                                  torch::Tensor b,
                                                  double eps) {
  return ln_fwd_common(x, res, w, b, eps, false);
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  return ln_fwd_common(x, c10::nullopt, w, c10::nullopt, eps, true);
}

static std::vector<torch::Tensor> ln_bwd_impl(torch::Tensor dy, torch::Tensor x,
                                              torch::Tensor w,
                                              torch::Tensor mean_or_empty,
                                              torch::Tensor rstd,
                                              c10::optional<torch::Tensor> dsum,
                                              bool rms) {
  int N = x.size(0), H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat));
  auto db = rms ? torch::Tensor()
                : torch::zeros({H}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const float* mean_p = rms ? nullptr : mean_or_empty.data_ptr<float>();
  // chunk count balances fill (>= 4 blocks/CU) against atomic contention
  int rows_per_chunk = std::max(16, N / 256);
  int nchunks = (N + rows_per_chunk - 1) / rows_per_chunk;
  dim3 grid2((H + BLOCK - 1) / BLOCK, nchunks);
  if (x.scalar_type() == torch::kBFloat16) {
    using T = __hip_bfloat16;
    ln_bwd_dx_launch<T>((const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                        (const T*)w.data_ptr(), mean_p,
                        rstd.data_ptr<float>(),
                        dsum.has_value() ? (const T*)dsum->data_ptr() : nullptr,
                        (T*)dx.data_ptr(), N, H, rms, stream);
    hipLaunchKernelGGL((ln_bwd_dwdb_kernel<T>), grid2, dim3(BLOCK), 0, stream,
                       (const T*)dy.data_ptr(), (const T*)x.data_ptr(), mean_p,
                       rstd.data_ptr<float>(), dw.data_ptr<float>(),
                       rms ? nullptr : db.data_ptr<float>(), N, H,
                       rows_per_chunk, rms);
  } else {
    ln_bwd_dx_launch<float>(dy.data_ptr<float>(), x.data_ptr<float>(),
                            w.data_ptr<float>(), mean_p,
                            rstd.data_ptr<float>(),
                            dsum.has_value() ? dsum->data_ptr<float>() : nullptr,
                            dx.data_ptr<float>(), N, H, rms, stream);
    hipLaunchKernelGGL((ln_bwd_dwdb_kernel<float>), grid2, dim3(BLOCK), 0,
                       stream, dy.data_ptr<float>(), x.data_ptr<float>(), mean_p,
                       rstd.data_ptr<float>(), dw.data_ptr<float>(),
                       rms ? nullptr : db.data_ptr<float>(), N, H,
                       rows_per_chunk, rms);
  }
  if (rms) return {dx, dw};
  return {dx, dw, db};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  return ln_bwd_impl(dy, x, w, mean, rstd, c10::nullopt, false);
}

// residual-fused: dx = ln_bwd(dy) + dsum (the residual-stream gradient)
std::vector<torch::Tensor> layernorm_bwd_residual(torch::Tensor dy,
                                                  torch::Tensor x,
                                                  torch::Tensor w,
                                                  torch::Tensor mean,
                                                  torch::Tensor rstd,
                                                  torch::Tensor dsum) {
  return ln_bwd_impl(dy, x, w, mean, rstd, dsum, false);
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  return ln_bwd_impl(dy, x, w, torch::Tensor(), rstd, c10::nullopt, true);
}
