// MoE token dispatch, gfx950: histogram + rank + row scatter in two
// small kernels, replacing the torch argsort/index_select pipeline
// (paddlefleetx_amd/models/moe/moe_layer.py dispatch; reference
// global_scatter/_assign_pos, ppfleetx moe/comm_ops.py:28-118 +
// moe/utils.py:97-107).
//
// Slot -> expert ids may contain -1 (capacity-dropped): those slots are
// excluded. Ranks come from per-expert atomics, so intra-expert order is
// unstable — harmless: expert FFNs act row-wise and the combine uses the
// inverse of the SAME permutation.
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int BLOCK = 256;

__global__ void moe_count_rank_kernel(const long* __restrict__ expert_ids,
                                      int* __restrict__ counts,
                                      int* __restrict__ rank, long S) {
  long i = (long)blockIdx.x * BLOCK + threadIdx.x;
  if (i >= S) return;
  long e = expert_ids[i];
  rank[i] = (e >= 0) ? atomicAdd(&counts[e], 1) : -1;
}

// dest[off[e] + rank[i]] = i for active slots; also copy the token row.
template <typename T, int VEC>
__global__ void moe_scatter_kernel(const long* __restrict__ expert_ids,
                                   const int* __restrict__ rank,
                                   const int* __restrict__ offsets,
                                   const T* __restrict__ x,
                                   T* __restrict__ out,
                                   long* __restrict__ sel_sorted,
                                   long S, int D, int top_k) {
  // one 16-lane group per slot: copies the D-wide row vectorized
  constexpr int GROUPS = BLOCK / 16;
  long i = (long)blockIdx.x * GROUPS + threadIdx.x / 16;
  if (i >= S) return;
  long e = expert_ids[i];
  if (e < 0) return;
  int lane = threadIdx.x % 16;
  long pos = offsets[e] + rank[i];
  if (lane == 0) sel_sorted[pos] = i;
  const T* src = x + (i / top_k) * (long)D;
  T* dst = out + pos * (long)D;
  for (int c = lane * VEC; c < D; c += 16 * VEC) {
    float v[VEC];
    vload<T, VEC>(src + c, v);
    vstore<T, VEC>(dst + c, v);
  }
}

}  // namespace

// x [T, D]; expert_ids [S = T*top_k] int64 (-1 = dropped)
// returns {dispatched [A, D], sel_sorted [A] int64, counts [E] int64}
std::vector<torch::Tensor> moe_dispatch(torch::Tensor x,
                                        torch::Tensor expert_ids,
                                        long num_experts, long top_k) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(expert_ids.scalar_type() == torch::kLong &&
              expert_ids.is_contiguous());
  long S = expert_ids.numel();
  int D = x.size(1);
  TORCH_CHECK(D % 4 == 0, "d_model must be divisible by 4");
  auto stream = at::hip::getCurrentHIPStream();
  auto counts32 = torch::zeros({num_experts},
                               x.options().dtype(torch::kInt));
  auto rank = torch::empty({S}, x.options().dtype(torch::kInt));
  hipLaunchKernelGGL(moe_count_rank_kernel,
                     dim3((S + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, stream,
                     expert_ids.data_ptr<long>(), counts32.data_ptr<int>(),
                     rank.data_ptr<int>(), S);
  auto offsets64 = torch::cumsum(counts32, 0) - counts32;
  auto offsets = offsets64.to(torch::kInt);
  // total active = S minus dropped; cheap device sum + sync (the host
  // needs the per-expert counts for the a2a splits anyway)
  long active = (long)counts32.sum().item<int>();
  auto out = torch::empty({active, (long)D}, x.options());
  auto sel_sorted = torch::empty({active},
                                 x.options().dtype(torch::kLong));
  constexpr int GROUPS = BLOCK / 16;
  dim3 grid((S + GROUPS - 1) / GROUPS);
#define LAUNCH_SCATTER(T, V)                                                \
  hipLaunchKernelGGL((moe_scatter_kernel<T, V>), grid, dim3(BLOCK), 0,      \
                     stream, expert_ids.data_ptr<long>(),                   \
                     rank.data_ptr<int>(), offsets.data_ptr<int>(),         \
                     (const T*)x.data_ptr(), (T*)out.data_ptr(),            \
                     sel_sorted.data_ptr<long>(), S, D, (int)top_k)
  if (x.scalar_type() == torch::kBFloat16) {
    if (D % 8 == 0) LAUNCH_SCATTER(__hip_bfloat16, 8);
    else LAUNCH_SCATTER(__hip_bfloat16, 4);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat);
    if (D % 8 == 0) LAUNCH_SCATTER(float, 8);
    else LAUNCH_SCATTER(float, 4);
  }
#undef LAUNCH_SCATTER
  return {out, sel_sorted, counts32.to(torch::kLong)};
}
