// Shared helpers for quoracle_amd CDNA4 (gfx950) kernels.
//
// Wave64 reductions and bf16 vector I/O idioms follow
// /opt/skills/guides/cdna_hip_programming.md: 64-wide wavefronts (not 32),
// bf16x8 vector loads via uint4, fp32 accumulation throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;

__device__ __forceinline__ float bf2f(bf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ bf16 f2bf(float v) { return __float2bfloat16(v); }

// Unpack a uint32 holding two bf16 values (little-endian: lo = element 0).
__device__ __forceinline__ void unpack_bf16x2(unsigned int u, float &lo, float &hi) {
  unsigned short a = (unsigned short)(u & 0xffffu);
  unsigned short b = (unsigned short)(u >> 16);
  lo = __bfloat162float(*reinterpret_cast<__hip_bfloat16 *>(&a));
  hi = __bfloat162float(*reinterpret_cast<__hip_bfloat16 *>(&b));
}

__device__ __forceinline__ unsigned int pack_bf16x2(float lo, float hi) {
  bf16 a = f2bf(lo), b = f2bf(hi);
  unsigned short ua = *reinterpret_cast<unsigned short *>(&a);
  unsigned short ub = *reinterpret_cast<unsigned short *>(&b);
  return (unsigned int)ua | ((unsigned int)ub << 16);
}

// Wave-wide (64-lane) sum/max via xor shuffles.
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block-wide reduce for up to 1024 threads (multiple of 64), via LDS scratch.
// `scratch` needs blockDim.x/64 floats. Result valid in all threads.
__device__ __forceinline__ float block_sum(float v, float *scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nw = blockDim.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

__device__ __forceinline__ float block_max(float v, float *scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nw = blockDim.x / WAVE;
  v = wave_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float m = -INFINITY;
#pragma unroll 4
  for (int i = 0; i < nw; ++i) m = fmaxf(m, scratch[i]);
  __syncthreads();
  return m;
}

#define HIP_CHECK_KERNEL()                                                    \
  do {                                                                        \
    hipError_t err_ = hipGetLastError();                                      \
    if (err_ != hipSuccess)                                                   \
      throw std::runtime_error(std::string("HIP kernel launch failed: ") +    \
                               hipGetErrorString(err_));                      \
  } while (0)
