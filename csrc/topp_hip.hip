#include "hip/hip_runtime.h"
// Top-p (nucleus) sampling, gfx950.
//
// Replaces ppfleetx/ops/topp_sampling.cu:377 (CUB segmented radix sort +
// block prefix scan + ballot cutoff). rocPRIM radix sort runs via
// torch.sort on the wrapper side; this kernel does the block-wide
// inclusive scan over sorted probs, the nucleus cutoff, and the draw.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;
constexpr int NW = BLOCK / WAVE;

// First index i in [0, n) whose inclusive cumulative sum >= threshold;
// n-1 if the total never reaches it. Whole block participates.
DEV_INLINE int find_first_crossing(const float* __restrict__ pr, int n,
                                   float threshold, float* wtot /*LDS[NW+1]*/) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  float running = 0.f;
  int result = -1;
  for (int base = 0; base < n; base += BLOCK) {
    const int i = base + threadIdx.x;
    const float v = (i < n) ? pr[i] : 0.f;
    // wave-inclusive scan
    float x = v;
#pragma unroll
    for (int off = 1; off < WAVE; off <<= 1) {
      float y = __shfl_up(x, off, WAVE);
      if (lane >= off) x += y;
    }
    if (lane == WAVE - 1) wtot[wid] = x;  // wave total
    __syncthreads();
    float wave_off = 0.f, chunk_tot = 0.f;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      if (w < wid) wave_off += wtot[w];
      chunk_tot += wtot[w];
    }
    __syncthreads();  // wtot reads done before it is reused below
    const float incl = running + wave_off + x;
    float cand = (i < n && incl >= threshold) ? (float)i : 1e30f;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      cand = fminf(cand, __shfl_xor(cand, off, WAVE));
    if (lane == 0) wtot[wid] = cand;
    __syncthreads();
    float best = 1e30f;
#pragma unroll
    for (int w = 0; w < NW; ++w) best = fminf(best, wtot[w]);
    __syncthreads();
    if (best < 1e30f) { result = (int)best; break; }
    running += chunk_tot;
  }
  return result < 0 ? n - 1 : result;
}

// sorted_p [B, V] descending (fp32), sorted_idx [B, V]; u [B] uniform draws.
__global__ void topp_select_kernel(const float* __restrict__ sorted_p,
                                   const long* __restrict__ sorted_idx,
                                   const float* __restrict__ top_p,
                                   const float* __restrict__ u,
                                   long* __restrict__ out_id,
                                   float* __restrict__ out_p, int V) {
  __shared__ float wtot[NW + 1];
  __shared__ float sred[NW];
  const long row = blockIdx.x;
  const float* pr = sorted_p + row * (long)V;

  // 1) nucleus = first prefix reaching top_p (always >= 1 token)
  int ncut = find_first_crossing(pr, V, top_p[row], wtot) + 1;
  __syncthreads();

  // 2) nucleus mass
  float mass = 0.f;
  for (int i = threadIdx.x; i < ncut; i += BLOCK) mass += pr[i];
  mass = block_reduce_sum<BLOCK>(mass, sred);
  __syncthreads();

  // 3) draw within the nucleus
  const float target = u[row] * mass;
  int pick = find_first_crossing(pr, ncut, target, wtot);
  if (threadIdx.x == 0) {
    out_id[row] = sorted_idx[row * (long)V + pick];
    out_p[row] = pr[pick];
  }
}

}  // namespace

std::vector<torch::Tensor> topp_select(torch::Tensor sorted_p,
                                       torch::Tensor sorted_idx,
                                       torch::Tensor top_p, torch::Tensor u) {
  TORCH_CHECK(sorted_p.is_cuda() && sorted_p.dim() == 2 &&
              sorted_p.scalar_type() == torch::kFloat);
  long B = sorted_p.size(0);
  int V = sorted_p.size(1);
  auto out_id = torch::empty({B, 1}, sorted_p.options().dtype(torch::kLong));
  auto out_p = torch::empty({B, 1}, sorted_p.options());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(topp_select_kernel, dim3(B), dim3(BLOCK), 0, stream,
                     sorted_p.data_ptr<float>(), sorted_idx.data_ptr<long>(),
                     top_p.data_ptr<float>(), u.data_ptr<float>(),
                     out_id.data_ptr<long>(), out_p.data_ptr<float>(), V);
  return {out_id, out_p};
}
