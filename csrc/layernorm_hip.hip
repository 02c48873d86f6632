#include "hip/hip_runtime.h"
// Fused LayerNorm / RMSNorm, gfx950. bf16/fp32 I/O, fp32 stats.
// Memory-bound: vectorized short8 (16 B/lane) loads per Guideline 13;
// one 256-thread block per row, grid-stride over rows.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;

// ---------------- forward ----------------
template <typename T, int VEC>
__global__ void ln_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                              const T* __restrict__ b, T* __restrict__ y,
                              float* __restrict__ mean, float* __restrict__ rstd,
                              int N, int H, float eps, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    float s = 0.f, ss = 0.f;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float v[VEC];
      if constexpr (sizeof(T) == 2) {
        short4v pk = *reinterpret_cast<const short4v*>(xr + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) v[j] = bf_raw2f(((unsigned short*)&pk)[j]);
      } else {
        float4v pk = *reinterpret_cast<const float4v*>(xr + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) v[j] = ((float*)&pk)[j];
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
    T* yr = y + (long)row * H;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
      float v[VEC], wv[VEC], bv[VEC];
      if constexpr (sizeof(T) == 2) {
        short4v pk = *reinterpret_cast<const short4v*>(xr + i);
        short4v wp = *reinterpret_cast<const short4v*>(w + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          v[j] = bf_raw2f(((unsigned short*)&pk)[j]);
          wv[j] = bf_raw2f(((unsigned short*)&wp)[j]);
        }
        if (!rms) {
          short4v bp = *reinterpret_cast<const short4v*>(b + i);
#pragma unroll
          for (int j = 0; j < VEC; ++j) bv[j] = bf_raw2f(((unsigned short*)&bp)[j]);
        }
        short4v out;
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float o = (v[j] - mu) * rs * wv[j] + (rms ? 0.f : bv[j]);
          ((unsigned short*)&out)[j] = f2bf_raw(o);
        }
        *reinterpret_cast<short4v*>(yr + i) = out;
      } else {
        float4v pk = *reinterpret_cast<const float4v*>(xr + i);
        float4v wp = *reinterpret_cast<const float4v*>(w + i);
        float4v bp;
        if (!rms) bp = *reinterpret_cast<const float4v*>(b + i);
        float4v out;
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float o = (((float*)&pk)[j] - mu) * rs * ((float*)&wp)[j] +
                    (rms ? 0.f : ((float*)&bp)[j]);
          ((float*)&out)[j] = o;
        }
        *reinterpret_cast<float4v*>(yr + i) = out;
      }
    }
    __syncthreads();
  }
}

// ---------------- backward: dx (row-parallel) ----------------
template <typename T, int VEC>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                 const T* __restrict__ w,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 T* __restrict__ dx, int N, int H, bool rms) {
  __shared__ float sred[BLOCK / WAVE];
  for (int row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (long)row * H;
    const T* dyr = dy + (long)row * H;
    float mu = rms ? 0.f : mean[row];
    float rs = rstd[row];
    float c1 = 0.f, c2 = 0.f;  // mean(dy*w), mean(dy*w*xhat)
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xv, dv, wv;
        if constexpr (sizeof(T) == 2) {
          xv = bf_raw2f(((const unsigned short*)xr)[i + j]);
          dv = bf_raw2f(((const unsigned short*)dyr)[i + j]);
          wv = bf_raw2f(((const unsigned short*)w)[i + j]);
        } else {
          xv = ((const float*)xr)[i + j];
          dv = ((const float*)dyr)[i + j];
          wv = ((const float*)w)[i + j];
        }
        float xh = (xv - mu) * rs;
        float dw = dv * wv;
        c1 += dw;
        c2 += dw * xh;
      }
    }
    c1 = block_reduce_sum<BLOCK>(c1, sred) / H;
    __syncthreads();
    c2 = block_reduce_sum<BLOCK>(c2, sred) / H;
    T* dxr = dx + (long)row * H;
    for (int i = threadIdx.x * VEC; i < H; i += BLOCK * VEC) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float xv, dv, wv;
        if constexpr (sizeof(T) == 2) {
          xv = bf_raw2f(((const unsigned short*)xr)[i + j]);
          dv = bf_raw2f(((const unsigned short*)dyr)[i + j]);
          wv = bf_raw2f(((const unsigned short*)w)[i + j]);
        } else {
          xv = ((const float*)xr)[i + j];
          dv = ((const float*)dyr)[i + j];
          wv = ((const float*)w)[i + j];
        }
        float xh = (xv - mu) * rs;
        float o = rms ? (dv * wv - xh * c2) * rs
                      : (dv * wv - c1 - xh * c2) * rs;
        if constexpr (sizeof(T) == 2)
          ((unsigned short*)dxr)[i + j] = f2bf_raw(o);
        else
          ((float*)dxr)[i + j] = o;
      }
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

template <typename T>
void ln_fwd_launch(const T* x, const T* w, const T* b, T* y, float* mean,
                   float* rstd, int N, int H, float eps, bool rms,
                   hipStream_t stream) {
  int grid = std::min(N, 2048);
  TORCH_CHECK(H % 4 == 0, "hidden size must be divisible by 4");
  hipLaunchKernelGGL((ln_fwd_kernel<T, 4>), dim3(grid), dim3(BLOCK), 0, stream,
                     x, w, b, y, mean, rstd, N, H, eps, rms);
}

}  // namespace

// ---------------- torch bindings ----------------

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  int N = x.size(0), H = x.size(1);
  auto y = torch::empty_like(x);
  auto mean = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    ln_fwd_launch<__hip_bfloat16>(
        (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
        (const __hip_bfloat16*)b.data_ptr(), (__hip_bfloat16*)y.data_ptr(),
        mean.data_ptr<float>(), rstd.data_ptr<float>(), N, H, (float)eps,
        false, stream);
  } else {
    ln_fwd_launch<float>(x.data_ptr<float>(), w.data_ptr<float>(),
                         b.data_ptr<float>(), y.data_ptr<float>(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(), N, H,
                         (float)eps, false, stream);
  }
  return {y, mean, rstd};
}

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  int N = x.size(0), H = x.size(1);
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    ln_fwd_launch<__hip_bfloat16>(
        (const __hip_bfloat16*)x.data_ptr(), (const __hip_bfloat16*)w.data_ptr(),
        nullptr, (__hip_bfloat16*)y.data_ptr(), nullptr,
        rstd.data_ptr<float>(), N, H, (float)eps, true, stream);
  } else {
    ln_fwd_launch<float>(x.data_ptr<float>(), w.data_ptr<float>(), nullptr,
                         y.data_ptr<float>(), nullptr, rstd.data_ptr<float>(),
                         N, H, (float)eps, true, stream);
  }
  return {y, rstd};
}

static std::vector<torch::Tensor> ln_bwd_impl(torch::Tensor dy, torch::Tensor x,
                                              torch::Tensor w,
                                              torch::Tensor mean_or_empty,
                                              torch::Tensor rstd, bool rms) {
  int N = x.size(0), H = x.size(1);
  auto dx = torch::empty_like(x);
  auto dw = torch::zeros({H}, x.options().dtype(torch::kFloat));
  auto db = rms ? torch::Tensor()
                : torch::zeros({H}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  int grid = std::min(N, 2048);
  const float* mean_p = rms ? nullptr : mean_or_empty.data_ptr<float>();
  // enough (col-tile x row-chunk) blocks to fill 256 CUs; each thread owns
  // one column over rows_per_chunk rows (coalesced across the 256 lanes)
  int rows_per_chunk = 16;
  int nchunks = (N + rows_per_chunk - 1) / rows_per_chunk;
  dim3 grid2((H + BLOCK - 1) / BLOCK, nchunks);
  if (x.scalar_type() == torch::kBFloat16) {
    using T = __hip_bfloat16;
    hipLaunchKernelGGL((ln_bwd_dx_kernel<T, 4>), dim3(grid), dim3(BLOCK), 0,
                       stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(),
                       (const T*)w.data_ptr(), mean_p, rstd.data_ptr<float>(),
                       (T*)dx.data_ptr(), N, H, rms);
    hipLaunchKernelGGL((ln_bwd_dwdb_kernel<T>), grid2, dim3(BLOCK), 0, stream,
                       (const T*)dy.data_ptr(), (const T*)x.data_ptr(), mean_p,
                       rstd.data_ptr<float>(), dw.data_ptr<float>(),
                       rms ? nullptr : db.data_ptr<float>(), N, H,
                       rows_per_chunk, rms);
  } else {
    hipLaunchKernelGGL((ln_bwd_dx_kernel<float, 4>), dim3(grid), dim3(BLOCK), 0,
                       stream, dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), mean_p, rstd.data_ptr<float>(),
                       dx.data_ptr<float>(), N, H, rms);
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
  return ln_bwd_impl(dy, x, w, mean, rstd, false);
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  return ln_bwd_impl(dy, x, w, torch::Tensor(), rstd, true);
}
