// Common helpers for the agentainer_amd CDNA4 (gfx950) HIP kernels.
//
// All kernels in this directory are written MI355X-first: wave64,
// vectorized 16-B/lane global access, LDS staging where reuse exists,
// MFMA for matmul-shaped work. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",         \
                  __FILE__, ":", __LINE__);                                    \
    }                                                                          \
  } while (0)

// bf16 <-> f32 scalar converts (device)
__device__ __forceinline__ float bf2f(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
__device__ __forceinline__ __hip_bfloat16 f2bf(float x) {
  return __float2bfloat16(x);
}

// Treat 8 bf16 as one 16-byte vector load/store unit.
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short bf16x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bits2f(short s) {
  union {
    unsigned u;
    float f;
  } cv;
  cv.u = ((unsigned)(unsigned short)s) << 16;
  return cv.f;
}

__device__ __forceinline__ short f2bits(float f) {
  // round-to-nearest-even bf16 truncation
  union {
    float f;
    unsigned u;
  } cv;
  cv.f = f;
  unsigned lsb = (cv.u >> 16) & 1u;
  unsigned rounded = cv.u + 0x7fffu + lsb;
  return (short)(rounded >> 16);
}

// OCP e4m3 <-> f32 (fp8 KV cache, scale-free storage). On gfx950 these
// lower to v_cvt_pk ops.
#include <hip/hip_fp8.h>
typedef unsigned char u8x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ unsigned char f2fp8(float x) {
  __hip_fp8_e4m3 v(x);
  return (unsigned char)v.__x;
}

__device__ __forceinline__ float fp82f(unsigned char b) {
  __hip_fp8_e4m3 v;
  v.__x = b;
  return (float)v;
}

// Fast path: 8 e4m3 bytes -> 8 f32 in 4 v_cvt_pk_f32_fp8 (the scalar
// __hip_fp8 float operator lowers to a software sequence on ROCm 7.2,
// which made the first fp8 decode-attention build VALU-bound).
typedef float f32x2_t __attribute__((ext_vector_type(2)));

__device__ __forceinline__ void fp8x8_to_f32(const u8x8 v, float* out) {
  int w0, w1;
  __builtin_memcpy(&w0, &v, 4);
  __builtin_memcpy(&w1, reinterpret_cast<const char*>(&v) + 4, 4);
  const f32x2_t p0 = __builtin_amdgcn_cvt_pk_f32_fp8(w0, false);
  const f32x2_t p1 = __builtin_amdgcn_cvt_pk_f32_fp8(w0, true);
  const f32x2_t p2 = __builtin_amdgcn_cvt_pk_f32_fp8(w1, false);
  const f32x2_t p3 = __builtin_amdgcn_cvt_pk_f32_fp8(w1, true);
  out[0] = p0.x; out[1] = p0.y; out[2] = p1.x; out[3] = p1.y;
  out[4] = p2.x; out[5] = p2.y; out[6] = p3.x; out[7] = p3.y;
}

// Wave-wide f32 reductions (64 lanes).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
