// Shared kernel infrastructure for gfx950 (CDNA4).  Wave size 64.
// Dtype helpers, vectorized load types, wave/block reductions.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>
#include <stdio.h>
#include <algorithm>

#define WAVE 64
#define DDL_CHECK(x)                                                          \
  do {                                                                        \
    hipError_t e_ = (x);                                                      \
    if (e_ != hipSuccess) {                                                   \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_),       \
              __FILE__, __LINE__);                                            \
      abort();                                                                \
    }                                                                         \
  } while (0)

using bf16 = __hip_bfloat16;
using f16 = __half;

// ---- scalar conversions ---------------------------------------------------
__device__ __forceinline__ float to_f32(float v) { return v; }
__device__ __forceinline__ float to_f32(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ float to_f32(f16 v) { return __half2float(v); }

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ bf16 from_f32<bf16>(float v) { return __float2bfloat16(v); }
template <> __device__ __forceinline__ f16 from_f32<f16>(float v) { return __float2half(v); }

// ---- vector types: process 8 elements per lane (16B for bf16/f16,
//      32B as two uint4 for f32) ------------------------------------------
template <typename T> struct Vec8 {
  T v[8];
};

template <typename T>
__device__ __forceinline__ Vec8<T> load8(const T* p) {
  Vec8<T> r;
  if constexpr (sizeof(T) == 2) {
    uint4 u = *reinterpret_cast<const uint4*>(p);
    *reinterpret_cast<uint4*>(r.v) = u;
  } else {
    uint4 u0 = *reinterpret_cast<const uint4*>(p);
    uint4 u1 = *reinterpret_cast<const uint4*>(p + 4);
    *reinterpret_cast<uint4*>(r.v) = u0;
    *reinterpret_cast<uint4*>(r.v + 4) = u1;
  }
  return r;
}

template <typename T>
__device__ __forceinline__ void store8(T* p, const Vec8<T>& r) {
  if constexpr (sizeof(T) == 2) {
    *reinterpret_cast<uint4*>(p) = *reinterpret_cast<const uint4*>(r.v);
  } else {
    *reinterpret_cast<uint4*>(p) = *reinterpret_cast<const uint4*>(r.v);
    *reinterpret_cast<uint4*>(p + 4) = *reinterpret_cast<const uint4*>(r.v + 4);
  }
}

// ---- reductions -----------------------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// Block reduction over up to 1024 threads (sum).  `scratch` >= blockDim/64
// floats.  Returns the total in every thread.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) total += scratch[i];
  return total;
}

// ---- grid helpers ---------------------------------------------------------
__device__ __forceinline__ int64_t grid_stride() {
  return (int64_t)gridDim.x * blockDim.x;
}
__device__ __forceinline__ int64_t global_tid() {
  return (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
}

// XCD-aware bijective block remap (guide §5.5 T1): contiguous chunks per XCD.
__device__ __forceinline__ unsigned xcd_swizzle(unsigned bid, unsigned nwg) {
  const unsigned NX = 8;
  if (nwg < NX * 2) return bid;
  unsigned xcd = bid % NX, idx = bid / NX;
  unsigned q = nwg / NX, r = nwg % NX;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// ceil-div
__host__ __device__ __forceinline__ int64_t cdiv(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}
