// Common helpers for timm_amd gfx950 HIP kernels.
// CDNA4-only: wave64, MFMA 16x16x32 bf16, LDS 160KB/CU. No CUDA compat paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64

// ---- vector types for wide loads ----
typedef short  short8 __attribute__((ext_vector_type(8)));   // 8 x bf16/f16 = 16B
typedef float  f32x4  __attribute__((ext_vector_type(4)));
typedef short  bf16x8 __attribute__((ext_vector_type(8)));
typedef short  bf16x4 __attribute__((ext_vector_type(4)));

// ---- dtype conversion helpers (element as ushort storage for 16-bit) ----
template <typename T> struct Elem;

template <> struct Elem<float> {
  using storage = float;
  static __device__ __forceinline__ float  to_f32(float v) { return v; }
  static __device__ __forceinline__ float  from_f32(float v) { return v; }
};

template <> struct Elem<__hip_bfloat16> {
  using storage = __hip_bfloat16;
  static __device__ __forceinline__ float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
  static __device__ __forceinline__ __hip_bfloat16 from_f32(float v) { return __float2bfloat16(v); }
};

template <> struct Elem<__half> {
  using storage = __half;
  static __device__ __forceinline__ float to_f32(__half v) { return __half2float(v); }
  static __device__ __forceinline__ __half from_f32(float v) { return __float2half(v); }
};

// bf16 bit helpers for short-typed registers
static __device__ __forceinline__ float bf16s_to_f32(short s) {
  union { unsigned int u; float f; } cvt;
  cvt.u = ((unsigned int)(unsigned short)s) << 16;
  return cvt.f;
}
static __device__ __forceinline__ short f32_to_bf16s(float f) {
  union { unsigned int u; float f; } cvt;
  cvt.f = f;
  unsigned int u = cvt.u;
  // round-to-nearest-even
  unsigned int lsb = (u >> 16) & 1;
  u += 0x7fff + lsb;
  return (short)(u >> 16);
}

// ---- wave reductions (64-wide) ----
static __device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
static __device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}
// reduce across a 16-lane group (lanes with the same (lane>>4))
static __device__ __forceinline__ float group16_reduce_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
static __device__ __forceinline__ float group16_reduce_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t _e = hipGetLastError();                                       \
    if (_e != hipSuccess) {                                                  \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(_e)); \
    }                                                                        \
  } while (0)

static inline int cdiv(int a, int b) { return (a + b - 1) / b; }
