// Shared helpers for the npf CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define NPF_WAVE 64  // CDNA wavefront width (NOT 32)

__device__ __forceinline__ float wave_reduce_sum(float x) {
  #pragma unroll
  for (int off = NPF_WAVE / 2; off > 0; off >>= 1)
    x += __shfl_down(x, off, NPF_WAVE);
  return x;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float x) {
  #pragma unroll
  for (int off = NPF_WAVE / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_down(x, off, NPF_WAVE));
  return x;  // valid in lane 0
}

__device__ __forceinline__ float wave_allreduce_sum(float x) {
  #pragma unroll
  for (int off = NPF_WAVE / 2; off > 0; off >>= 1)
    x += __shfl_xor(x, off, NPF_WAVE);
  return x;  // valid in all lanes
}

__device__ __forceinline__ float wave_allreduce_max(float x) {
  #pragma unroll
  for (int off = NPF_WAVE / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, NPF_WAVE));
  return x;  // valid in all lanes
}

// block-level reduce-sum into lane 0 of wave 0 (needs 16 floats of smem)
__device__ __forceinline__ float block_reduce_sum(float x, float* smem16) {
  const int lane = threadIdx.x & (NPF_WAVE - 1);
  const int wid = threadIdx.x / NPF_WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) smem16[wid] = x;
  __syncthreads();
  const int nw = (blockDim.x + NPF_WAVE - 1) / NPF_WAVE;
  x = (threadIdx.x < nw) ? smem16[threadIdx.x] : 0.f;
  if (wid == 0) x = wave_reduce_sum(x);
  return x;  // valid in thread 0
}

// generic load/store as float for fp32 / bf16 tensors
template <typename T>
__device__ __forceinline__ float ldf(const T* p) { return static_cast<float>(*p); }
template <>
__device__ __forceinline__ float ldf<__hip_bfloat16>(const __hip_bfloat16* p) {
  return __bfloat162float(*p);
}
template <typename T>
__device__ __forceinline__ void stf(T* p, float v) { *p = static_cast<T>(v); }
template <>
__device__ __forceinline__ void stf<__hip_bfloat16>(__hip_bfloat16* p, float v) {
  *p = __float2bfloat16(v);
}
