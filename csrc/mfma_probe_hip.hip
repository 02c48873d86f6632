#include "hip/hip_runtime.h"
// Debug probe: one 16x16x32 bf16 MFMA with the fragment layout assumed
// across attention.hip. Tested on-GPU against torch.matmul so a layout
// mistake shows up in ONE 16x16 tile instead of inside the attention
// kernels (guide §3: asymmetric-B check).
#include "common.h"

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf8;
typedef __attribute__((ext_vector_type(4))) float f4;

// a [16, 32] bf16 row-major, b [32, 16] bf16 row-major -> c [16, 16] f32
__global__ void mfma16_probe_kernel(const __hip_bfloat16* __restrict__ a,
                                    const __hip_bfloat16* __restrict__ b,
                                    float* __restrict__ c) {
  const int lane = threadIdx.x % WAVE;
  // A[row][k]: row = lane&15, k = (lane>>4)*8 + e
  union { short4v v[2]; bf8 f; } ua, ub;
  int arow = lane & 15, k0 = (lane >> 4) * 8;
  ua.v[0] = *reinterpret_cast<const short4v*>((const unsigned short*)a + arow * 32 + k0);
  ua.v[1] = *reinterpret_cast<const short4v*>((const unsigned short*)a + arow * 32 + k0 + 4);
  // B[k][col]: col = lane&15, k = (lane>>4)*8 + e  (strided reads)
  int bcol = lane & 15;
  unsigned short tmp[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) tmp[e] = ((const unsigned short*)b)[(k0 + e) * 16 + bcol];
  ub.v[0] = *reinterpret_cast<short4v*>(tmp);
  ub.v[1] = *reinterpret_cast<short4v*>(tmp + 4);
  f4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ua.f, ub.f, acc, 0, 0, 0);
  // C[row][col]: col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int r = 0; r < 4; ++r)
    c[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

}  // namespace

torch::Tensor mfma_gemm16_probe(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
  auto c = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma16_probe_kernel, dim3(1), dim3(WAVE), 0, stream,
                     (const __hip_bfloat16*)a.contiguous().data_ptr(),
                     (const __hip_bfloat16*)b.contiguous().data_ptr(),
                     c.data_ptr<float>());
  return c;
}
