// Common helpers for gfx950 (MI355X / CDNA4) kernels.
// Wavefront = 64 lanes; LDS 160 KiB/CU; vectorize bf16 as short4/short8.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) short short2v;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(2))) float float2v;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// MFMA fragment types (gfx950 16x16x32 bf16: 8 bf16 in = 4 VGPRs, 4 f32 acc)
typedef __attribute__((ext_vector_type(8))) __bf16 mfma_bf16x8;
typedef __attribute__((ext_vector_type(4))) float mfma_f32x4;

DEV_INLINE float bf2f(__hip_bfloat16 x) { return __bfloat162float(x); }
DEV_INLINE __hip_bfloat16 f2bf(float x) { return __float2bfloat16(x); }

DEV_INLINE float bf_raw2f(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}
DEV_INLINE unsigned short f2bf_raw(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  unsigned int lsb = (v.i >> 16) & 1u;
  v.i += 0x7fffu + lsb;  // round-to-nearest-even
  return (unsigned short)(v.i >> 16);
}

#ifdef __HIP_DEVICE_COMPILE__
// Pack two f32 into two bf16 (RTNE) in ONE VALU op — the hand-rolled
// f2bf_raw costs ~4 integer ops per value (guide T12: no builtin on
// gfx950, use asm). lo -> bits [15:0], hi -> bits [31:16].
DEV_INLINE unsigned int cvt_pk_bf16(float lo, float hi) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}
#else
DEV_INLINE unsigned int cvt_pk_bf16(float lo, float hi) {
  return (unsigned int)f2bf_raw(lo) | ((unsigned int)f2bf_raw(hi) << 16);
}
#endif

// ---- wave reductions (64-lane) ----
DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
DEV_INLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}
// reduce within contiguous 16-lane groups
DEV_INLINE float group16_reduce_sum(float x) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}
DEV_INLINE float group16_reduce_max(float x) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// ---- block reductions via LDS (expects <= 16 waves) ----
template <int BLOCK>
DEV_INLINE float block_reduce_sum(float x, float* lds_scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  float r = (threadIdx.x < NW) ? lds_scratch[threadIdx.x] : 0.f;
#pragma unroll
  for (int off = NW / 2; off > 0; off >>= 1) r += __shfl_xor(r, off, WAVE);
  r = __shfl(r, 0, WAVE);
  if (threadIdx.x == 0) lds_scratch[0] = r;
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
DEV_INLINE float block_reduce_max(float x, float* lds_scratch) {
  constexpr int NW = BLOCK / WAVE;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) lds_scratch[wid] = x;
  __syncthreads();
  float r = (threadIdx.x < NW) ? lds_scratch[threadIdx.x] : -INFINITY;
#pragma unroll
  for (int off = NW / 2; off > 0; off >>= 1) r = fmaxf(r, __shfl_xor(r, off, WAVE));
  r = __shfl(r, 0, WAVE);
  if (threadIdx.x == 0) lds_scratch[0] = r;
  __syncthreads();
  r = lds_scratch[0];
  __syncthreads();
  return r;
}

// ---- Philox4x32-10 counter-based RNG (dropout; torch's generator) ----
struct philox4 { unsigned x, y, z, w; };

DEV_INLINE philox4 philox4x32_10(unsigned c0, unsigned c1, unsigned c2,
                                 unsigned c3, unsigned k0, unsigned k1) {
  const unsigned M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const unsigned W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int i = 0; i < 10; ++i) {
    unsigned hi0 = __umulhi(M0, c0), lo0 = M0 * c0;
    unsigned hi1 = __umulhi(M1, c2), lo1 = M1 * c2;
    unsigned n0 = hi1 ^ c1 ^ k0;
    unsigned n1 = lo1;
    unsigned n2 = hi0 ^ c3 ^ k1;
    unsigned n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}

// ---- width-generic vector load/store (bf16 raw-short or fp32), VEC 4/8 ----
template <typename T, int VEC>
DEV_INLINE void vload(const T* p, float* v) {
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
      float4v pk = *reinterpret_cast<const float4v*>((const float*)p + j);
#pragma unroll
      for (int k = 0; k < 4; ++k) v[j + k] = ((float*)&pk)[k];
    }
  }
}

template <typename T, int VEC>
DEV_INLINE void vstore(T* p, const float* v) {
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
      *reinterpret_cast<float4v*>((float*)p + j) = out;
    }
  }
}

#define HIP_CHECK(x)                                                       \
  do {                                                                     \
    hipError_t _e = (x);                                                   \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
    }                                                                      \
  } while (0)
