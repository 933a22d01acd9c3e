// Common helpers for the dinov3_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA; block size is a multiple of 64 everywhere.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64
#define DEV_INLINE __device__ __forceinline__

// 16-byte vector of 8 bf16 values (guideline 13: always vectorize bf16 I/O).
typedef __attribute__((ext_vector_type(8))) short short8_t;
typedef __attribute__((ext_vector_type(4))) float float4_t;

DEV_INLINE float bf16_to_f32(short u) {
  union { float f; unsigned int i; } v;
  v.i = ((unsigned int)(unsigned short)u) << 16;
  return v.f;
}

DEV_INLINE short f32_to_bf16(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int rounding_bias = 0x7FFF + ((v.i >> 16) & 1);
  return (short)((v.i + rounding_bias) >> 16);
}

// ---- type traits: load/store element at fp32 compute precision ----
template <typename T> struct ScalarOps;

template <> struct ScalarOps<float> {
  DEV_INLINE static float load(const float* p) { return *p; }
  DEV_INLINE static void store(float* p, float v) { *p = v; }
};

template <> struct ScalarOps<__hip_bfloat16> {
  DEV_INLINE static float load(const __hip_bfloat16* p) {
    return bf16_to_f32(*reinterpret_cast<const short*>(p));
  }
  DEV_INLINE static void store(__hip_bfloat16* p, float v) {
    *reinterpret_cast<short*>(p) = f32_to_bf16(v);
  }
};

template <> struct ScalarOps<_Float16> {
  DEV_INLINE static float load(const _Float16* p) { return (float)(*p); }
  DEV_INLINE static void store(_Float16* p, float v) { *p = (_Float16)v; }
};

// ---- 8-element vectorized load/store (16B for bf16/fp16, 2x16B for f32) ----
template <typename T> struct Vec8 {
  DEV_INLINE static void load(T* dst, const T* src) {
    reinterpret_cast<float4_t*>(dst)[0] = reinterpret_cast<const float4_t*>(src)[0];
    reinterpret_cast<float4_t*>(dst)[1] = reinterpret_cast<const float4_t*>(src)[1];
  }
  DEV_INLINE static void store(T* dst, const T* src) {
    reinterpret_cast<float4_t*>(dst)[0] = reinterpret_cast<const float4_t*>(src)[0];
    reinterpret_cast<float4_t*>(dst)[1] = reinterpret_cast<const float4_t*>(src)[1];
  }
};

template <> struct Vec8<__hip_bfloat16> {
  DEV_INLINE static void load(__hip_bfloat16* dst, const __hip_bfloat16* src) {
    *reinterpret_cast<float4_t*>(dst) = *reinterpret_cast<const float4_t*>(src);
  }
  DEV_INLINE static void store(__hip_bfloat16* dst, const __hip_bfloat16* src) {
    *reinterpret_cast<float4_t*>(dst) = *reinterpret_cast<const float4_t*>(src);
  }
};

// ---- wave + block reductions ----
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE_SIZE);
  return v;  // valid in lane 0 of the wave
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE_SIZE));
  return v;
}

// Block-wide sum for blockDim.x <= 1024 (<=16 waves). `lds` needs >= 16 floats.
DEV_INLINE float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) v = wave_reduce_sum(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  return lds[0];
}

DEV_INLINE float block_reduce_max(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) v = wave_reduce_max(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  return lds[0];
}

#define HIP_CHECK_LAUNCH()                                                    \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) {                                                    \
      printf("HIP launch error: %s at %s:%d\n", hipGetErrorString(e),         \
             __FILE__, __LINE__);                                             \
    }                                                                         \
  } while (0)
