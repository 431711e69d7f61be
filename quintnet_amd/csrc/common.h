// Common helpers for the quintnet_amd CDNA4 (gfx950) kernel library.
// Wave size is 64 on CDNA — every cross-lane idiom below assumes it.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define QN_WAVE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));                \
    }                                                                          \
  } while (0)

// ---- vector types -----------------------------------------------------------
typedef __attribute__((ext_vector_type(2))) float  f32x2;
typedef __attribute__((ext_vector_type(4))) float  f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short  s16x4;
typedef __attribute__((ext_vector_type(8))) short  s16x8;
typedef __attribute__((ext_vector_type(2))) short  s16x2;
typedef __attribute__((ext_vector_type(4))) int    i32x4;

// ---- dtype conversion -------------------------------------------------------
__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

__device__ __forceinline__ unsigned short f32_to_bf16(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  unsigned int x = v.i;
  // round-to-nearest-even
  unsigned int lsb = (x >> 16) & 1u;
  x += 0x7fffu + lsb;
  return (unsigned short)(x >> 16);
}

// 8-wide bf16 load/store helpers (guide G13: scalar bf16 is ~2-2.5x)
__device__ __forceinline__ void ld8_f32(const unsigned short* p, float* out) {
  s16x8 v = *reinterpret_cast<const s16x8*>(p);
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = bf16_to_f32((unsigned short)v[j]);
}

__device__ __forceinline__ void st8_f32(unsigned short* p, const float* in) {
  s16x8 v;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = (short)f32_to_bf16(in[j]);
  *reinterpret_cast<s16x8*>(p) = v;
}

// generic scalar load/store as float, templated on element type
template <typename T> __device__ __forceinline__ float ld_as_f32(const T* p);
template <> __device__ __forceinline__ float ld_as_f32<float>(const float* p) { return *p; }
template <> __device__ __forceinline__ float ld_as_f32<unsigned short>(const unsigned short* p) {
  return bf16_to_f32(*p);
}

template <typename T> __device__ __forceinline__ void st_from_f32(T* p, float v);
template <> __device__ __forceinline__ void st_from_f32<float>(float* p, float v) { *p = v; }
template <> __device__ __forceinline__ void st_from_f32<unsigned short>(unsigned short* p, float v) {
  *p = f32_to_bf16(v);
}

// ---- wave reductions (64-lane) ---------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, QN_WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, QN_WAVE));
  return v;
}

// block reduction via LDS (block = nwaves x 64)
// Result is returned to EVERY thread of the block (broadcast through LDS).
template <int MAX_WAVES>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = (blockDim.x + QN_WAVE - 1) / QN_WAVE;
    float acc = 0.f;
    for (int i = 0; i < nwaves; ++i) acc += lds_scratch[i];
    lds_scratch[0] = acc;
  }
  __syncthreads();
  v = lds_scratch[0];
  __syncthreads();  // protect scratch for back-to-back reductions
  return v;
}

template <int MAX_WAVES>
__device__ __forceinline__ float block_reduce_max(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (QN_WAVE - 1);
  const int wave = threadIdx.x / QN_WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds_scratch[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = (blockDim.x + QN_WAVE - 1) / QN_WAVE;
    float acc = -INFINITY;
    for (int i = 0; i < nwaves; ++i) acc = fmaxf(acc, lds_scratch[i]);
    lds_scratch[0] = acc;
  }
  __syncthreads();
  v = lds_scratch[0];
  __syncthreads();
  return v;
}

// tanh-approx GELU matching torch.nn.functional.gelu(approximate="tanh")
__device__ __forceinline__ float gelu_tanh(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float t = tanhf(c * (x + 0.044715f * x * x * x));
  return 0.5f * x * (1.0f + t);
}
