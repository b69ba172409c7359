// Common device helpers for the CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#define IBP_CHECK_HIP(cmd)                                                    \
  do {                                                                        \
    hipError_t e = (cmd);                                                     \
    if (e != hipSuccess) {                                                    \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(e), " at ",        \
                  __FILE__, ":", __LINE__);                                   \
    }                                                                         \
  } while (0)

namespace ibp {

constexpr int kWave = 64;

// ---- bf16 <-> fp32 -------------------------------------------------------
__device__ __forceinline__ float b2f(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ __hip_bfloat16 f2b(float v) {
  return __float2bfloat16(v);
}

// bf16 stored as ushort for raw vector loads
__device__ __forceinline__ float us2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = (unsigned int)u << 16;
  return c.f;
}
__device__ __forceinline__ unsigned short f2us(float f) {
  // round-to-nearest-even bf16
  union { float f; unsigned int i; } c;
  c.f = f;
  unsigned int lsb = (c.i >> 16) & 1u;
  unsigned int rounded = c.i + 0x7fffu + lsb;
  return (unsigned short)(rounded >> 16);
}

// generic scalar load/store templated over torch scalar types
template <typename T> __device__ __forceinline__ float ldf(const T* p) {
  return static_cast<float>(*p);
}
template <> __device__ __forceinline__ float ldf<__hip_bfloat16>(const __hip_bfloat16* p) {
  return b2f(*p);
}
template <typename T> __device__ __forceinline__ void stf(T* p, float v) {
  *p = static_cast<T>(v);
}
template <> __device__ __forceinline__ void stf<__hip_bfloat16>(__hip_bfloat16* p, float v) {
  *p = f2b(v);
}

// ---- wave / block reductions --------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// block reduction via LDS; blockDim.x must be a multiple of 64 and <= 1024
__device__ __forceinline__ float block_reduce_sum(float v, float* lds /*>= 16 floats*/) {
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  int nw = (blockDim.x + 63) >> 6;
  v = (threadIdx.x < (unsigned)nw) ? lds[threadIdx.x] : 0.0f;
  if (wid == 0) {
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  }
  return v;  // valid in thread 0
}

__device__ __forceinline__ float leaky(float v, float slope) {
  return v > 0.0f ? v : v * slope;
}

// grid helper: number of blocks capped for grid-stride loops
inline int grid_1d(long long total, int block, int cap = 2048) {
  long long g = (total + block - 1) / block;
  return (int)(g < cap ? (g > 0 ? g : 1) : cap);
}

}  // namespace ibp
