// Common helpers for ray_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA4; all block sizes are multiples of 64.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16;
typedef __hip_bfloat162 bf162;

// short8: 8 bf16 = 16 B per lane (coalescing sweet spot, guide G13).
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(2))) float float2v;

DEV_INLINE float bf2f(short s) {
  union { unsigned int u; float f; } cv;
  cv.u = ((unsigned int)(unsigned short)s) << 16;
  return cv.f;
}

DEV_INLINE short f2bf(float f) {
  union { float f; unsigned int u; } cv;
  cv.f = f;
  unsigned int lsb = (cv.u >> 16) & 1;
  cv.u += 0x7fff + lsb;  // round-to-nearest-even
  return (short)(cv.u >> 16);
}

// Wave-wide reduction (64 lanes).
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// Block reduction using LDS; BLOCK must be a multiple of 64, <=1024.
template <int BLOCK>
DEV_INLINE float block_sum(float v, float* lds /* BLOCK/64 floats */) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  v = wave_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = 0.f;
  if (wid == 0) {
    r = (lane < BLOCK / 64) ? lds[lane] : 0.f;
    r = wave_sum(r);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
DEV_INLINE float block_max(float v, float* lds) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  v = wave_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = -INFINITY;
  if (wid == 0) {
    r = (lane < BLOCK / 64) ? lds[lane] : -INFINITY;
    r = wave_max(r);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

#define HIP_CHECK_KERNEL()                                    \
  do {                                                        \
    hipError_t err_ = hipGetLastError();                      \
    if (err_ != hipSuccess)                                   \
      throw std::runtime_error(hipGetErrorString(err_));      \
  } while (0)
