// Fused scale+causal-mask+softmax and cross-entropy kernels, gfx950.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;

// ---- fused scale + causal mask + softmax on [B, H, Sq, Sk] ----
// One block per (b*h, qi) row; valid columns = 0 .. qi + (Sk - Sq).
template <typename T>
__global__ void softmax_causal_fwd_kernel(const T* __restrict__ s,
                                          T* __restrict__ y, int Sq, int Sk,
                                          float scale) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;  // (bh * Sq + qi)
  int qi = (int)(row % Sq);
  int valid = qi + (Sk - Sq) + 1;  // causal prefix length
  const T* sr = s + row * Sk;
  T* yr = y + row * Sk;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < valid; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                               : ((const float*)sr)[i];
    mx = fmaxf(mx, v * scale);
  }
  mx = block_reduce_max<BLOCK>(mx, sred);
  __syncthreads();
  float sum = 0.f;
  for (int i = threadIdx.x; i < valid; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                               : ((const float*)sr)[i];
    sum += __expf(v * scale - mx);
  }
  sum = block_reduce_sum<BLOCK>(sum, sred);
  float inv = 1.f / sum;
  for (int i = threadIdx.x; i < Sk; i += BLOCK) {
    float o = 0.f;
    if (i < valid) {
      float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                                 : ((const float*)sr)[i];
      o = __expf(v * scale - mx) * inv;
    }
    if constexpr (sizeof(T) == 2)
      ((unsigned short*)yr)[i] = f2bf_raw(o);
    else
      ((float*)yr)[i] = o;
  }
}

// ---- fused scale + bias-add + softmax (non-causal) on [R, K] ----
// bias broadcasts with period `inner` over an outer stride `outer`:
// bias_row = (row / outer) * inner + row % inner — covers the folding
// model's [B, 1, h, Q, K] pair-bias against [B, S, h, Q, K] logits
// (reference fused_gate_attention, protein_folding/attentions.py:126).
template <typename T>
__global__ void softmax_bias_fwd_kernel(const T* __restrict__ s,
                                        const T* __restrict__ bias,
                                        T* __restrict__ y, int K,
                                        long inner, long outer,
                                        float scale) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;
  long brow = (row / outer) * inner + row % inner;
  const T* sr = s + row * K;
  const T* br = bias + brow * K;
  T* yr = y + row * K;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < K; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                               : ((const float*)sr)[i];
    float b = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)br)[i])
                               : ((const float*)br)[i];
    mx = fmaxf(mx, v * scale + b);
  }
  mx = block_reduce_max<BLOCK>(mx, sred);
  __syncthreads();
  float sum = 0.f;
  for (int i = threadIdx.x; i < K; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                               : ((const float*)sr)[i];
    float b = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)br)[i])
                               : ((const float*)br)[i];
    sum += __expf(v * scale + b - mx);
  }
  sum = block_reduce_sum<BLOCK>(sum, sred);
  float inv = 1.f / sum;
  for (int i = threadIdx.x; i < K; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)sr)[i])
                               : ((const float*)sr)[i];
    float b = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)br)[i])
                               : ((const float*)br)[i];
    float o = __expf(v * scale + b - mx) * inv;
    if constexpr (sizeof(T) == 2)
      ((unsigned short*)yr)[i] = f2bf_raw(o);
    else
      ((float*)yr)[i] = o;
  }
}

// ds = y * (dy - sum(dy*y)) * scale
template <typename T>
__global__ void softmax_causal_bwd_kernel(const T* __restrict__ dy,
                                          const T* __restrict__ y,
                                          T* __restrict__ ds, int Sk,
                                          float scale) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;
  const T* dyr = dy + row * Sk;
  const T* yr = y + row * Sk;
  T* dsr = ds + row * Sk;
  float dot = 0.f;
  for (int i = threadIdx.x; i < Sk; i += BLOCK) {
    float d = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)dyr)[i])
                               : ((const float*)dyr)[i];
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)yr)[i])
                               : ((const float*)yr)[i];
    dot += d * v;
  }
  dot = block_reduce_sum<BLOCK>(dot, sred);
  for (int i = threadIdx.x; i < Sk; i += BLOCK) {
    float d = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)dyr)[i])
                               : ((const float*)dyr)[i];
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)yr)[i])
                               : ((const float*)yr)[i];
    float o = v * (d - dot) * scale;
    if constexpr (sizeof(T) == 2)
      ((unsigned short*)dsr)[i] = f2bf_raw(o);
    else
      ((float*)dsr)[i] = o;
  }
}

// ---- cross entropy: logits [N, V] -> loss [N], lse [N] (fp32) ----
template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ loss, float* __restrict__ lse,
                              int V, long ignore_index) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;
  const T* lr = logits + row * (long)V;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)lr)[i])
                               : ((const float*)lr)[i];
    mx = fmaxf(mx, v);
  }
  mx = block_reduce_max<BLOCK>(mx, sred);
  __syncthreads();
  float sum = 0.f;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)lr)[i])
                               : ((const float*)lr)[i];
    sum += __expf(v - mx);
  }
  sum = block_reduce_sum<BLOCK>(sum, sred);
  if (threadIdx.x == 0) {
    float l = logf(sum) + mx;
    lse[row] = l;
    long lab = labels[row];
    if (lab == ignore_index) {
      loss[row] = 0.f;
    } else {
      float p = (sizeof(T) == 2)
                    ? bf_raw2f(((const unsigned short*)lr)[lab])
                    : ((const float*)lr)[lab];
      loss[row] = l - p;
    }
  }
}

// dlogits = (softmax - onehot) * dloss, 0 for ignored rows
template <typename T>
__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const T* __restrict__ logits,
                              const long* __restrict__ labels,
                              const float* __restrict__ lse,
                              T* __restrict__ dlogits, int V,
                              long ignore_index) {
  long row = blockIdx.x;
  const T* lr = logits + row * (long)V;
  T* dr = dlogits + row * (long)V;
  long lab = labels[row];
  float dl = (lab == ignore_index) ? 0.f : dloss[row];
  float l = lse[row];
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)lr)[i])
                               : ((const float*)lr)[i];
    float p = __expf(v - l);
    float g = (p - (i == (int)lab ? 1.f : 0.f)) * dl;
    if constexpr (sizeof(T) == 2)
      ((unsigned short*)dr)[i] = f2bf_raw(g);
    else
      ((float*)dr)[i] = g;
  }
}

// ---- vocab-parallel CE helpers: local stats in one pass each ----
template <typename T>
__global__ void row_max_kernel(const T* __restrict__ x, float* __restrict__ out,
                               int V) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;
  const T* xr = x + row * (long)V;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)xr)[i])
                               : ((const float*)xr)[i];
    mx = fmaxf(mx, v);
  }
  mx = block_reduce_max<BLOCK>(mx, sred);
  if (threadIdx.x == 0) out[row] = mx;
}

template <typename T>
__global__ void row_sumexp_kernel(const T* __restrict__ x,
                                  const float* __restrict__ gmax,
                                  float* __restrict__ out, int V) {
  __shared__ float sred[BLOCK / WAVE];
  long row = blockIdx.x;
  const T* xr = x + row * (long)V;
  float m = gmax[row];
  float s = 0.f;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)xr)[i])
                               : ((const float*)xr)[i];
    s += __expf(v - m);
  }
  s = block_reduce_sum<BLOCK>(s, sred);
  if (threadIdx.x == 0) out[row] = s;
}

// picked[row] = logits[row, label-start] if label in [start, end) else 0
template <typename T>
__global__ void gather_label_logit_kernel(const T* __restrict__ x,
                                          const long* __restrict__ labels,
                                          float* __restrict__ out, long N,
                                          int V, long start,
                                          long ignore_index) {
  long row = (long)blockIdx.x * BLOCK + threadIdx.x;
  if (row >= N) return;
  long lab = labels[row];
  float v = 0.f;
  if (lab != ignore_index && lab >= start && lab < start + V) {
    long j = lab - start;
    v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)(x + row * (long)V))[j])
                         : ((const float*)(x + row * (long)V))[j];
  }
  out[row] = v;
}

// vocab-parallel CE backward: dlogits = (exp(l - lse) - onehot_local) * dloss
template <typename T>
__global__ void vp_ce_bwd_kernel(const float* __restrict__ dloss,
                                 const T* __restrict__ logits,
                                 const long* __restrict__ labels,
                                 const float* __restrict__ lse,
                                 T* __restrict__ dlogits, int V, long start,
                                 long ignore_index) {
  long row = blockIdx.x;
  const T* lr = logits + row * (long)V;
  T* dr = dlogits + row * (long)V;
  long lab = labels[row];
  bool valid = lab != ignore_index;
  float dl = valid ? dloss[row] : 0.f;
  float l = lse[row];
  long local = (valid && lab >= start && lab < start + V) ? lab - start : -1;
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    float v = (sizeof(T) == 2) ? bf_raw2f(((const unsigned short*)lr)[i])
                               : ((const float*)lr)[i];
    float p = __expf(v - l);
    float g = (p - (i == (int)local ? 1.f : 0.f)) * dl;
    if constexpr (sizeof(T) == 2)
      ((unsigned short*)dr)[i] = f2bf_raw(g);
    else
      ((float*)dr)[i] = g;
  }
}

}  // namespace

#define DISPATCH_T(tensor, ...)                         \
  if ((tensor).scalar_type() == torch::kBFloat16) {     \
    using T = __hip_bfloat16;                           \
    __VA_ARGS__;                                        \
  } else {                                              \
    using T = float;                                    \
    __VA_ARGS__;                                        \
  }

torch::Tensor softmax_causal_fwd(torch::Tensor s, double scale) {
  TORCH_CHECK(s.is_cuda() && s.is_contiguous() && s.dim() == 4);
  int Sq = s.size(2), Sk = s.size(3);
  long rows = s.numel() / Sk;
  auto y = torch::empty_like(s);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(s, hipLaunchKernelGGL((softmax_causal_fwd_kernel<T>), dim3(rows),
                                   dim3(BLOCK), 0, stream,
                                   (const T*)s.data_ptr(), (T*)y.data_ptr(),
                                   Sq, Sk, (float)scale));
  return y;
}

// softmax(s*scale + bias) rowwise over the last dim; bias broadcast
// with period `inner` / outer stride `outer` (see kernel comment)
torch::Tensor softmax_bias_fwd(torch::Tensor s, torch::Tensor bias,
                               long inner, long outer, double scale) {
  TORCH_CHECK(s.is_cuda() && s.is_contiguous() && bias.is_contiguous());
  int K = s.size(-1);
  TORCH_CHECK(bias.size(-1) == K);
  long rows = s.numel() / K;
  auto y = torch::empty_like(s);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(s, hipLaunchKernelGGL((softmax_bias_fwd_kernel<T>), dim3(rows),
                                   dim3(BLOCK), 0, stream,
                                   (const T*)s.data_ptr(),
                                   (const T*)bias.data_ptr(),
                                   (T*)y.data_ptr(), K, inner, outer,
                                   (float)scale));
  return y;
}

torch::Tensor softmax_causal_bwd(torch::Tensor dy, torch::Tensor y,
                                 double scale) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous() && dy.is_contiguous());
  int Sk = y.size(-1);
  long rows = y.numel() / Sk;
  auto ds = torch::empty_like(y);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(y, hipLaunchKernelGGL((softmax_causal_bwd_kernel<T>), dim3(rows),
                                   dim3(BLOCK), 0, stream,
                                   (const T*)dy.data_ptr(),
                                   (const T*)y.data_ptr(), (T*)ds.data_ptr(),
                                   Sk, (float)scale));
  return ds;
}

std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor labels,
                                             long ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  long N = logits.size(0);
  int V = logits.size(1);
  auto loss = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(logits, hipLaunchKernelGGL((ce_fwd_kernel<T>), dim3(N),
                                        dim3(BLOCK), 0, stream,
                                        (const T*)logits.data_ptr(),
                                        labels.data_ptr<long>(),
                                        loss.data_ptr<float>(),
                                        lse.data_ptr<float>(), V,
                                        ignore_index));
  return {loss, lse};
}

torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logits,
                                torch::Tensor labels, torch::Tensor lse,
                                long ignore_index) {
  long N = logits.size(0);
  int V = logits.size(1);
  auto dl = torch::empty_like(logits);
  auto dlf = dloss.to(torch::kFloat).contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(logits, hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(N),
                                        dim3(BLOCK), 0, stream,
                                        dlf.data_ptr<float>(),
                                        (const T*)logits.data_ptr(),
                                        labels.data_ptr<long>(),
                                        lse.data_ptr<float>(),
                                        (T*)dl.data_ptr(), V, ignore_index));
  return dl;
}

torch::Tensor row_max(torch::Tensor x) {
  long N = x.size(0);
  int V = x.size(1);
  auto out = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(x, hipLaunchKernelGGL((row_max_kernel<T>), dim3(N), dim3(BLOCK), 0,
                                   stream, (const T*)x.data_ptr(),
                                   out.data_ptr<float>(), V));
  return out;
}

torch::Tensor row_sumexp(torch::Tensor x, torch::Tensor gmax) {
  long N = x.size(0);
  int V = x.size(1);
  auto out = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(x, hipLaunchKernelGGL((row_sumexp_kernel<T>), dim3(N), dim3(BLOCK),
                                   0, stream, (const T*)x.data_ptr(),
                                   gmax.data_ptr<float>(),
                                   out.data_ptr<float>(), V));
  return out;
}

torch::Tensor gather_label_logit(torch::Tensor x, torch::Tensor labels,
                                 long start, long ignore_index) {
  long N = x.size(0);
  int V = x.size(1);
  auto out = torch::empty({N}, x.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(x, hipLaunchKernelGGL((gather_label_logit_kernel<T>),
                                   dim3((N + BLOCK - 1) / BLOCK),
                                   dim3(BLOCK), 0, stream,
                                   (const T*)x.data_ptr(),
                                   labels.data_ptr<long>(),
                                   out.data_ptr<float>(), N, V, start,
                                   ignore_index));
  return out;
}

torch::Tensor vp_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                        torch::Tensor labels, torch::Tensor lse, long start,
                        long ignore_index) {
  long N = logits.size(0);
  int V = logits.size(1);
  auto dl = torch::empty_like(logits);
  auto dlf = dloss.to(torch::kFloat).contiguous();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_T(logits, hipLaunchKernelGGL((vp_ce_bwd_kernel<T>), dim3(N),
                                        dim3(BLOCK), 0, stream,
                                        dlf.data_ptr<float>(),
                                        (const T*)logits.data_ptr(),
                                        labels.data_ptr<long>(),
                                        lse.data_ptr<float>(),
                                        (T*)dl.data_ptr(), V, start,
                                        ignore_index));
  return dl;
}
