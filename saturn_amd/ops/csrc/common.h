// Common device helpers for saturn_amd CDNA4 (gfx950) kernels.
//
// Written directly for MI355X: wave64, 4xSIMD-32 CUs, 160 KiB LDS,
// HBM3E-bound elementwise ops vectorized to 8x bf16 per lane (guide G13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

namespace samd {

typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

__device__ __forceinline__ float bf2f(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ __hip_bfloat16 f2bf(float x) {
  return __float2bfloat16(x);
}

// dtype-generic float conversion (torch builds define
// __HIP_NO_HALF_CONVERSIONS__, so C-style casts on __half don't compile).
template <typename T>
__device__ __forceinline__ float toF(T v) { return (float)v; }
template <>
__device__ __forceinline__ float toF<__half>(__half v) { return __half2float(v); }
template <>
__device__ __forceinline__ float toF<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <typename T>
__device__ __forceinline__ T fromF(float v) { return (T)v; }
template <>
__device__ __forceinline__ __half fromF<__half>(float v) {
  return __float2half(v);
}
template <>
__device__ __forceinline__ __hip_bfloat16 fromF<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// bf16 stored as ushort bit pattern <-> float
__device__ __forceinline__ float us2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}
__device__ __forceinline__ unsigned short f2us(float f) {
  union { unsigned int i; float f; } c;
  c.f = f;
  // round-to-nearest-even bf16
  unsigned int lsb = (c.i >> 16) & 1u;
  unsigned int rounded = c.i + 0x7fffu + lsb;
  return (unsigned short)(rounded >> 16);
}

// Wave-wide reductions (64 lanes).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}
__device__ __forceinline__ float wave_allsum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;  // valid in all lanes
}
__device__ __forceinline__ float wave_allmax(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block reduction across waves through LDS; BLOCK <= 1024.
template <int BLOCK>
__device__ __forceinline__ float block_sum(float v, float* lds /*BLOCK/WAVE*/) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = 0.f;
  if (wid == 0) {
    r = (lane < BLOCK / WAVE) ? lds[lane] : 0.f;
    r = wave_sum(r);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
__device__ __forceinline__ float block_max(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = -INFINITY;
  if (wid == 0) {
    r = (lane < BLOCK / WAVE) ? lds[lane] : -INFINITY;
    r = wave_max(r);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

}  // namespace samd

#define SAMD_CHECK_HIP(expr)                                        \
  do {                                                              \
    hipError_t _e = (expr);                                         \
    if (_e != hipSuccess) {                                         \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));     \
    }                                                               \
  } while (0)
