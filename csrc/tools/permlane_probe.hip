// Standalone probe: print the exact lane conventions of
// v_permlane16_swap_b32 / v_permlane32_swap_b32 on gfx950.
// Build on a GPU box:  hipcc --offload-arch=gfx950 permlane_probe.hip -o probe && ./probe
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned* o16a, unsigned* o16b, unsigned* o32a,
                      unsigned* o32b) {
  unsigned x = threadIdx.x;          // lane id
  unsigned y = threadIdx.x + 1000;   // distinct second operand
  auto r16 = __builtin_amdgcn_permlane16_swap(x, y, false, false);
  o16a[threadIdx.x] = r16[0];
  o16b[threadIdx.x] = r16[1];
  auto r32 = __builtin_amdgcn_permlane32_swap(x, y, false, false);
  o32a[threadIdx.x] = r32[0];
  o32b[threadIdx.x] = r32[1];
}

int main() {
  unsigned *a, *b, *c, *d;
  hipMallocManaged(&a, 64 * 4); hipMallocManaged(&b, 64 * 4);
  hipMallocManaged(&c, 64 * 4); hipMallocManaged(&d, 64 * 4);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, a, b, c, d);
  hipDeviceSynchronize();
  printf("lane:  "); for (int i = 0; i < 64; ++i) printf("%4d", i);
  printf("\np16[0]:"); for (int i = 0; i < 64; ++i) printf("%4u", a[i]);
  printf("\np16[1]:"); for (int i = 0; i < 64; ++i) printf("%4u", b[i] % 1000);
  printf("\np32[0]:"); for (int i = 0; i < 64; ++i) printf("%4u", c[i]);
  printf("\np32[1]:"); for (int i = 0; i < 64; ++i) printf("%4u", d[i] % 1000);
  printf("\n(x = lane, y = lane+1000; %%1000 shown for the y-origin reg)\n");
  return 0;
}
