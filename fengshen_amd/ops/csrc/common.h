// Common device helpers for fengshen_amd CDNA4 (gfx950) kernels.
// Wave64 reductions, bf16 vector load/store (guide §6 G13: ALWAYS vectorize
// bf16 as short4/short8 — scalar bf16 loads cost ~2-2.5x).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE_SIZE 64
#define FS_MAX_BLOCKS 2048  // memory-bound grid cap (guide §6 G11)

using bf16_t = __hip_bfloat16;
using fp16_t = __half;

typedef short short8_t __attribute__((ext_vector_type(8)));
typedef float float4_t __attribute__((ext_vector_type(4)));

// ---- scalar conversions -------------------------------------------------
template <typename T> __device__ __forceinline__ float to_f32(T v);
template <> __device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ float to_f32<bf16_t>(bf16_t v) {
  return __bfloat162float(v);
}
template <> __device__ __forceinline__ float to_f32<fp16_t>(fp16_t v) {
  return __half2float(v);
}

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ bf16_t from_f32<bf16_t>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ fp16_t from_f32<fp16_t>(float v) {
  return __float2half(v);
}

// ---- vectorized 8-element load/store (16B for bf16/fp16, 32B for f32) ----
template <typename T>
__device__ __forceinline__ void load8(const T* p, float (&out)[8]) {
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = to_f32<T>(p[i]);
}
template <>
__device__ __forceinline__ void load8<bf16_t>(const bf16_t* p, float (&out)[8]) {
  short8_t v = *reinterpret_cast<const short8_t*>(p);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    unsigned short u = (unsigned short)v[i];
    unsigned int w = ((unsigned int)u) << 16;
    out[i] = __uint_as_float(w);
  }
}
template <>
__device__ __forceinline__ void load8<fp16_t>(const fp16_t* p, float (&out)[8]) {
  short8_t v = *reinterpret_cast<const short8_t*>(p);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    __half_raw h; h.x = (unsigned short)v[i];
    out[i] = __half2float(__half(h));
  }
}

template <typename T>
__device__ __forceinline__ void store8(T* p, const float (&in)[8]) {
#pragma unroll
  for (int i = 0; i < 8; ++i) p[i] = from_f32<T>(in[i]);
}
template <>
__device__ __forceinline__ void store8<bf16_t>(bf16_t* p, const float (&in)[8]) {
  short8_t v;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    v[i] = (short)(__hip_bfloat16_raw(__float2bfloat16(in[i])).x);
  }
  *reinterpret_cast<short8_t*>(p) = v;
}
template <>
__device__ __forceinline__ void store8<fp16_t>(fp16_t* p, const float (&in)[8]) {
  short8_t v;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    __half_raw h(__float2half(in[i]));
    v[i] = (short)h.x;
  }
  *reinterpret_cast<short8_t*>(p) = v;
}

// ---- wave & block reductions --------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}
__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// block reduce across up to 16 waves (block <= 1024 threads)
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nw = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) v = wave_reduce_sum(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}
__device__ __forceinline__ float block_reduce_max(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nw = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nw) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) v = wave_reduce_max(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

#define HIP_CHECK_LAST()                                                   \
  do {                                                                     \
    hipError_t e = hipGetLastError();                                      \
    if (e != hipSuccess) {                                                 \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__,    \
             __LINE__);                                                    \
    }                                                                      \
  } while (0)
