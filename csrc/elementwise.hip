// Elementwise fused kernels: bias+gelu (fwd/bwd), flat AdamW, RoPE.
// All memory-bound: vectorized loads (G13), grid-stride (G11).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;
constexpr float GELU_K = 0.7978845608028654f;  // sqrt(2/pi)

// tanh via the hardware exp unit: tanh(y) = 1 - 2/(exp(2y)+1).
// libm tanhf lowers to a branchy polynomial (~20 VALU ops); this is 4,
// exact to ~1 ulp fp32 for the gelu argument range.
DEV_INLINE float fast_tanh(float y) {
  return 1.f - 2.f / (__expf(2.f * y) + 1.f);
}
DEV_INLINE float gelu_tanh(float x) {
  float t = fast_tanh(GELU_K * (x + 0.044715f * x * x * x));
  return 0.5f * x * (1.f + t);
}
DEV_INLINE float gelu_tanh_grad(float x) {
  float t = fast_tanh(GELU_K * (x + 0.044715f * x * x * x));
  return 0.5f * (1.f + t) +
         0.5f * x * (1.f - t * t) * GELU_K * (1.f + 3.f * 0.044715f * x * x);
}

// ---- bias + gelu: x [N, H] + bias[H] ----
// 2D grid: blockIdx.x covers a column span (the bias chunk loads ONCE per
// block), blockIdx.y strides rows — no 64-bit div/mod per element.
template <typename T, bool HAS_BIAS, int VEC>
__global__ __launch_bounds__(BLOCK) void bias_gelu_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ bias, T* __restrict__ y,
    long N, int H) {
  const int col = (blockIdx.x * BLOCK + threadIdx.x) * VEC;
  if (col >= H) return;
  float bv[VEC];
  if constexpr (HAS_BIAS) vload<T, VEC>(bias + col, bv);
  for (long r = blockIdx.y; r < N; r += gridDim.y) {
    const long i = r * H + col;
    float v[VEC];
    vload<T, VEC>(x + i, v);
    float o[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      if constexpr (HAS_BIAS) v[j] += bv[j];
      o[j] = gelu_tanh(v[j]);
    }
    vstore<T, VEC>(y + i, o);
  }
}

template <typename T, bool HAS_BIAS, int VEC>
__global__ __launch_bounds__(BLOCK) void bias_gelu_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ bias, T* __restrict__ dx, long N, int H) {
  const int col = (blockIdx.x * BLOCK + threadIdx.x) * VEC;
  if (col >= H) return;
  float bv[VEC];
  if constexpr (HAS_BIAS) vload<T, VEC>(bias + col, bv);
  for (long r = blockIdx.y; r < N; r += gridDim.y) {
    const long i = r * H + col;
    float v[VEC], d[VEC];
    vload<T, VEC>(x + i, v);
    vload<T, VEC>(dy + i, d);
    float o[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      if constexpr (HAS_BIAS) v[j] += bv[j];
      o[j] = d[j] * gelu_tanh_grad(v[j]);
    }
    vstore<T, VEC>(dx + i, o);
  }
}

// ---- column sum: x [N, H] -> fp32 [H] (dbias; chunked atomics) ----
template <typename T, int CV>
__global__ __launch_bounds__(BLOCK) void colsum_kernel(
    const T* __restrict__ x, float* __restrict__ out, long N, int H,
    int rows_per_chunk) {
  const int col = (blockIdx.x * BLOCK + threadIdx.x) * CV;
  if (col >= H) return;
  long r0 = (long)blockIdx.y * rows_per_chunk;
  long r1 = r0 + rows_per_chunk < N ? r0 + rows_per_chunk : N;
  float s[CV];
#pragma unroll
  for (int j = 0; j < CV; ++j) s[j] = 0.f;
  for (long r = r0; r < r1; ++r) {
    float v[CV];
    vload<T, CV>(x + r * H + col, v);
#pragma unroll
    for (int j = 0; j < CV; ++j) s[j] += v[j];
  }
#pragma unroll
  for (int j = 0; j < CV; ++j) atomicAdd(&out[col + j], s[j]);
}

// ---- fused grad sumsq (found_inf + global-norm in ONE pass) ----
// out[slot] += sum(x^2) in fp32; NaN/Inf grads propagate into the sum,
// so !isfinite(out) IS the found_inf flag (reference
// check_finite_and_unscale, distributed/apis/amp.py:212-216)
template <typename T>
__global__ void grad_sumsq_kernel(const T* __restrict__ x, long n,
                                  float* __restrict__ out, long slot) {
  __shared__ float sred[BLOCK / WAVE];
  float s = 0.f;
  constexpr int V = 4;
  for (long i = ((long)blockIdx.x * BLOCK + threadIdx.x) * V; i < n;
       i += (long)gridDim.x * BLOCK * V) {
    float v[V];
    vload<T, V>(x + i, v);
#pragma unroll
    for (int j = 0; j < V; ++j) s += v[j] * v[j];
  }
  s = block_reduce_sum<BLOCK>(s, sred);
  if (threadIdx.x == 0) atomicAdd(&out[slot], s);
}

void launch_colsum(const void* x, float* out, long N, int H, bool bf16,
                   hipStream_t stream) {
  constexpr int CV = 4;
  int rows_per_chunk = (int)std::max<long>(32, N / 128);
  int nchunks = (int)((N + rows_per_chunk - 1) / rows_per_chunk);
  dim3 grid((H + BLOCK * CV - 1) / (BLOCK * CV), nchunks);
  if (bf16)
    hipLaunchKernelGGL((colsum_kernel<__hip_bfloat16, CV>), grid, dim3(BLOCK),
                       0, stream, (const __hip_bfloat16*)x, out, N, H,
                       rows_per_chunk);
  else
    hipLaunchKernelGGL((colsum_kernel<float, CV>), grid, dim3(BLOCK), 0,
                       stream, (const float*)x, out, N, H, rows_per_chunk);
}

// ---- AdamW over flat fp32 state, low-precision model copy + grads ----
template <typename TM, typename TG>  // model dtype, grad dtype
__global__ void adamw_flat_kernel(float* __restrict__ master,
                                  const TG* __restrict__ grad,
                                  float* __restrict__ m, float* __restrict__ v,
                                  TM* __restrict__ model, long n, float lr,
                                  float beta1, float beta2, float eps,
                                  float wd, float bc1, float bc2) {
  const int VEC = 4;
  for (long i = ((long)blockIdx.x * BLOCK + threadIdx.x) * VEC; i < n;
       i += (long)gridDim.x * BLOCK * VEC) {
    float4v pm = *reinterpret_cast<float4v*>(master + i);
    float4v pg;
    if constexpr (sizeof(TG) == 2) {
      short4v graw = *reinterpret_cast<const short4v*>((const unsigned short*)grad + i);
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        ((float*)&pg)[j] = bf_raw2f(((unsigned short*)&graw)[j]);
    } else {
      pg = *reinterpret_cast<const float4v*>((const float*)grad + i);
    }
    float4v pmm = *reinterpret_cast<float4v*>(m + i);
    float4v pvv = *reinterpret_cast<float4v*>(v + i);
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float g = ((float*)&pg)[j];
      float mm = beta1 * ((float*)&pmm)[j] + (1.f - beta1) * g;
      float vv = beta2 * ((float*)&pvv)[j] + (1.f - beta2) * g * g;
      float p = ((float*)&pm)[j];
      p *= (1.f - lr * wd);
      float denom = sqrtf(vv / bc2) + eps;
      p -= lr * (mm / bc1) / denom;
      ((float*)&pmm)[j] = mm;
      ((float*)&pvv)[j] = vv;
      ((float*)&pm)[j] = p;
    }
    *reinterpret_cast<float4v*>(master + i) = pm;
    *reinterpret_cast<float4v*>(m + i) = pmm;
    *reinterpret_cast<float4v*>(v + i) = pvv;
    if (model) {
      if constexpr (sizeof(TM) == 2) {
        short4v out;
#pragma unroll
        for (int j = 0; j < VEC; ++j)
          ((unsigned short*)&out)[j] = f2bf_raw(((float*)&pm)[j]);
        *reinterpret_cast<short4v*>((unsigned short*)model + i) = out;
      } else {
        *reinterpret_cast<float4v*>((float*)model + i) = pm;
      }
    }
  }
}

// ---- RoPE: x [B, H, S, D], cos/sin [S, D/2] fp32, interleaved pairs ----
template <typename T>
__global__ void rope_kernel(const T* __restrict__ x, const float* __restrict__ cs,
                            const float* __restrict__ sn, T* __restrict__ y,
                            long rows, int S, int D) {
  // one row = one (b, h, s); lanes cover D/2 pairs
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    int s = (int)(row % S);
    const T* xr = x + row * D;
    T* yr = y + row * D;
    const float* cr = cs + (long)s * (D / 2);
    const float* sr = sn + (long)s * (D / 2);
    for (int p = threadIdx.x; p < D / 2; p += BLOCK) {
      float x1, x2;
      if constexpr (sizeof(T) == 2) {
        x1 = bf_raw2f(((const unsigned short*)xr)[2 * p]);
        x2 = bf_raw2f(((const unsigned short*)xr)[2 * p + 1]);
      } else {
        x1 = ((const float*)xr)[2 * p];
        x2 = ((const float*)xr)[2 * p + 1];
      }
      float c = cr[p], s_ = sr[p];
      float o1 = x1 * c - x2 * s_;
      float o2 = x2 * c + x1 * s_;
      if constexpr (sizeof(T) == 2) {
        ((unsigned short*)yr)[2 * p] = f2bf_raw(o1);
        ((unsigned short*)yr)[2 * p + 1] = f2bf_raw(o2);
      } else {
        ((float*)yr)[2 * p] = o1;
        ((float*)yr)[2 * p + 1] = o2;
      }
    }
  }
}

long grid_for(long total, int per_block) {
  long g = (total + per_block - 1) / per_block;
  return std::min<long>(g, 2048);
}

}  // namespace

torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  long total = x.numel();
  int H = x.size(-1);
  long N = total / H;
  TORCH_CHECK(H % 4 == 0, "last dim must be divisible by 4");
  bool has_bias = bias.defined() && bias.numel() > 0;
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  bool v8 = H % 8 == 0;
  int vec = v8 ? 8 : 4;
  dim3 grid((H + BLOCK * vec - 1) / (BLOCK * vec),
            (unsigned)std::min<long>(N, 16384));
#define LAUNCH_BG(T, HB, V)                                                  \
  hipLaunchKernelGGL((bias_gelu_fwd_kernel<T, HB, V>), grid, dim3(BLOCK), 0, \
                     stream, (const T*)x.data_ptr(),                         \
                     has_bias ? (const T*)bias.data_ptr() : nullptr,         \
                     (T*)y.data_ptr(), N, H)
#define PICK_BG(T, HB) do { if (v8) LAUNCH_BG(T, HB, 8); else LAUNCH_BG(T, HB, 4); } while (0)
  if (x.scalar_type() == torch::kBFloat16) {
    if (has_bias) PICK_BG(__hip_bfloat16, true);
    else PICK_BG(__hip_bfloat16, false);
  } else {
    if (has_bias) PICK_BG(float, true);
    else PICK_BG(float, false);
  }
#undef PICK_BG
#undef LAUNCH_BG
  return y;
}

std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  long total = x.numel();
  int H = x.size(-1);
  long N = total / H;
  TORCH_CHECK(H % 4 == 0, "last dim must be divisible by 4");
  bool has_bias = bias.defined() && bias.numel() > 0;
  auto dx = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  bool v8 = H % 8 == 0;
  int vec = v8 ? 8 : 4;
  dim3 grid((H + BLOCK * vec - 1) / (BLOCK * vec),
            (unsigned)std::min<long>(N, 16384));
#define LAUNCH_BG(T, HB, V)                                                  \
  hipLaunchKernelGGL((bias_gelu_bwd_kernel<T, HB, V>), grid, dim3(BLOCK), 0, \
                     stream, (const T*)dy.data_ptr(), (const T*)x.data_ptr(), \
                     has_bias ? (const T*)bias.data_ptr() : nullptr,         \
                     (T*)dx.data_ptr(), N, H)
#define PICK_BG(T, HB) do { if (v8) LAUNCH_BG(T, HB, 8); else LAUNCH_BG(T, HB, 4); } while (0)
  if (x.scalar_type() == torch::kBFloat16) {
    if (has_bias) PICK_BG(__hip_bfloat16, true);
    else PICK_BG(__hip_bfloat16, false);
  } else {
    if (has_bias) PICK_BG(float, true);
    else PICK_BG(float, false);
  }
#undef PICK_BG
#undef LAUNCH_BG
  torch::Tensor db;
  if (has_bias) {
    // fused column sum (fp32 accumulate) — replaces the materialized
    // .to(float).sum(0), which cost two extra full passes over dx
    db = torch::zeros({H}, x.options().dtype(torch::kFloat));
    launch_colsum(dx.data_ptr(), db.data_ptr<float>(), N, H,
                  x.scalar_type() == torch::kBFloat16, stream);
  }
  return {dx, db};
}

// accumulate sum(x^2) into out[slot] (fp32). x flat, numel % 4 == 0.
void grad_sumsq(torch::Tensor x, torch::Tensor out, long slot) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kFloat);
  long n = x.numel();
  TORCH_CHECK(n % 4 == 0);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(grid_for(n, BLOCK * 4));
  if (x.scalar_type() == torch::kBFloat16)
    hipLaunchKernelGGL((grad_sumsq_kernel<__hip_bfloat16>), grid,
                       dim3(BLOCK), 0, stream,
                       (const __hip_bfloat16*)x.data_ptr(), n,
                       out.data_ptr<float>(), slot);
  else
    hipLaunchKernelGGL((grad_sumsq_kernel<float>), grid, dim3(BLOCK), 0,
                       stream, (const float*)x.data_ptr(), n,
                       out.data_ptr<float>(), slot);
}

// standalone column sum for linear dbias: x [N, H] -> fp32 [H]
torch::Tensor colsum(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  long N = x.size(0);
  int H = x.size(1);
  TORCH_CHECK(H % 4 == 0, "H must be divisible by 4");
  auto out = torch::zeros({H}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  launch_colsum(x.data_ptr(), out.data_ptr<float>(), N, H,
                x.scalar_type() == torch::kBFloat16, stream);
  return out;
}

void adamw_flat(torch::Tensor master, torch::Tensor grad, torch::Tensor m,
                torch::Tensor v, torch::Tensor model, double lr, double beta1,
                double beta2, double eps, double wd, long step) {
  TORCH_CHECK(master.is_cuda() && master.scalar_type() == torch::kFloat);
  long n = master.numel();
  TORCH_CHECK(n % 4 == 0);
  bool has_model = model.defined() && model.numel() > 0;
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(grid_for(n, BLOCK * 4));
  bool model_bf16 = has_model && model.scalar_type() == torch::kBFloat16;
  bool grad_bf16 = grad.scalar_type() == torch::kBFloat16;
#define LAUNCH_ADAMW(TM, TG, MP, GP)                                          \
  hipLaunchKernelGGL((adamw_flat_kernel<TM, TG>), grid, dim3(BLOCK), 0,       \
                     stream, master.data_ptr<float>(), (const TG*)(GP),       \
                     m.data_ptr<float>(), v.data_ptr<float>(), (TM*)(MP), n,  \
                     (float)lr, (float)beta1, (float)beta2, (float)eps,       \
                     (float)wd, bc1, bc2)
  void* mp = has_model ? model.data_ptr() : nullptr;
  const void* gp = grad.data_ptr();
  if (model_bf16 && grad_bf16) LAUNCH_ADAMW(__hip_bfloat16, __hip_bfloat16, mp, gp);
  else if (model_bf16) LAUNCH_ADAMW(__hip_bfloat16, float, mp, gp);
  else if (grad_bf16) LAUNCH_ADAMW(float, __hip_bfloat16, mp, gp);
  else LAUNCH_ADAMW(float, float, mp, gp);
#undef LAUNCH_ADAMW
}

torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cos, torch::Tensor sin) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4);
  int S = x.size(2), D = x.size(3);
  long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid((unsigned)std::min<long>(rows, 8192));
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL((rope_kernel<__hip_bfloat16>), grid, dim3(BLOCK), 0,
                       stream, (const __hip_bfloat16*)x.data_ptr(),
                       cos.data_ptr<float>(), sin.data_ptr<float>(),
                       (__hip_bfloat16*)y.data_ptr(), rows, S, D);
  } else {
    hipLaunchKernelGGL((rope_kernel<float>), grid, dim3(BLOCK), 0, stream,
                       x.data_ptr<float>(), cos.data_ptr<float>(),
                       sin.data_ptr<float>(), y.data_ptr<float>(), rows, S, D);
  }
  return y;
}
