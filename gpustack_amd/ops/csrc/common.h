// Common device helpers for gpustack_amd CDNA4 (gfx950 / MI355X) kernels.
//
// Design notes (MI355X-first):
//  - wavefront = 64 lanes; all cross-lane idioms use 64-wide shuffles.
//  - bf16 is the serving dtype; all reductions accumulate in f32.
//  - memory-bound kernels vectorize loads as 16 B/lane (u16x8 for bf16).
#pragma once

#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE_SIZE 64

typedef __attribute__((ext_vector_type(2))) unsigned short u16x2;
typedef __attribute__((ext_vector_type(4))) unsigned short u16x4;
typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) float f32x8;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

// bf16 (stored as unsigned short) <-> f32
__device__ __forceinline__ float bf2f(unsigned short u) {
  union { float f; unsigned int i; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

// round-to-nearest-even f32 -> bf16
__device__ __forceinline__ unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int x = v.i;
  if ((x & 0x7fffffffu) > 0x7f800000u) return 0x7fc0; // NaN
  unsigned int round = 0x7fffu + ((x >> 16) & 1u);
  return (unsigned short)((x + round) >> 16);
}

typedef __attribute__((ext_vector_type(8))) unsigned char u8x8;
typedef __attribute__((ext_vector_type(4))) unsigned char u8x4;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

// OCP fp8 e4m3 (gfx950-native v_cvt_f32_fp8 / v_cvt_pk_fp8_f32)
__device__ __forceinline__ float fp8_to_f32(unsigned char u) {
  __hip_fp8_e4m3 v;
  v.__x = u;
  return (float)v;
}

__device__ __forceinline__ unsigned char f32_to_fp8(float f) {
  __hip_fp8_e4m3 v(f);
  return v.__x;
}

__device__ __forceinline__ void fp8x8_to_f32(const u8x8 v, float* out) {
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = fp8_to_f32(v[i]);
}

__device__ __forceinline__ void bf8_to_f32(const u16x8 v, float* out) {
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f(v[i]);
}

__device__ __forceinline__ u16x8 f32_to_bf8(const float* in) {
  u16x8 v;
#pragma unroll
  for (int i = 0; i < 8; ++i) v[i] = f2bf(in[i]);
  return v;
}

// Full-wave (64-lane) butterfly reductions.
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int m = 32; m >= 1; m >>= 1) x += __shfl_xor(x, m, WAVE_SIZE);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int m = 32; m >= 1; m >>= 1) x = fmaxf(x, __shfl_xor(x, m, WAVE_SIZE));
  return x;
}

// Reduction across a 16-lane subgroup (lanes with equal l>>4).
__device__ __forceinline__ float group16_reduce_sum(float x) {
#pragma unroll
  for (int m = 8; m >= 1; m >>= 1) x += __shfl_xor(x, m, WAVE_SIZE);
  return x;
}

#define HIP_CHECK_KERNEL()                                        \
  do {                                                            \
    hipError_t e_ = hipGetLastError();                            \
    if (e_ != hipSuccess) {                                       \
      TORCH_CHECK(false, "HIP kernel launch failed: ",            \
                  hipGetErrorString(e_));                         \
    }                                                             \
  } while (0)

#define DEVICE_INLINE __device__ __forceinline__

static inline int ceil_div(int a, int b) { return (a + b - 1) / b; }
