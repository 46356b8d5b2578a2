// Common device helpers for colossalai_amd HIP kernels (gfx950 / CDNA4 only).
//
// Conventions:
//  - wavefront = 64 lanes, hard-coded (per CDNA4 guide: neither
//    __AMDGCN_WAVEFRONT_SIZE nor warpSize-as-constexpr is reliable).
//  - bf16 is the primary working dtype; accumulation is fp32.
//  - memory-bound kernels ALWAYS load bf16 vectorized (short4/short8
//    reinterpret) — hipcc does not auto-vectorize scalar bf16 loads.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

namespace cai {

// ---------------------------------------------------------------- vec types
typedef float float4v __attribute__((ext_vector_type(4)));
typedef float float2v __attribute__((ext_vector_type(2)));
typedef short short8 __attribute__((ext_vector_type(8)));   // 8 x bf16 = 16 B
typedef short short4v __attribute__((ext_vector_type(4)));  // 8 B
typedef short short2v __attribute__((ext_vector_type(2)));
// MFMA fragment types (gfx950)
typedef short bf16x8 __attribute__((ext_vector_type(8)));   // A/B frag: 8 bf16 in 4 VGPRs
typedef float f32x4 __attribute__((ext_vector_type(4)));    // 16x16 C/D frag
typedef float f32x16 __attribute__((ext_vector_type(16)));  // 32x32 C/D frag

using bf16 = __hip_bfloat16;

DEV_INLINE float bf2f(unsigned short u) {
  union { unsigned int u32; float f; } cvt;
  cvt.u32 = ((unsigned int)u) << 16;
  return cvt.f;
}

DEV_INLINE unsigned short f2bf(float f) {
  union { float f; unsigned int u32; } cvt;
  cvt.f = f;
  // round-to-nearest-even
  unsigned int u = cvt.u32;
  unsigned int rounding = 0x7FFF + ((u >> 16) & 1);
  u += rounding;
  return (unsigned short)(u >> 16);
}

// ------------------------------------------------------------- reductions
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// Block reduce: requires blockDim.x % 64 == 0, <= 1024 threads.
// smem must hold blockDim.x/64 floats.
DEV_INLINE float block_reduce_sum(float v, float* smem) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nwaves) ? smem[threadIdx.x] : 0.0f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off);
  }
  if (threadIdx.x == 0) smem[0] = v;
  __syncthreads();
  v = smem[0];
  __syncthreads();
  return v;
}

DEV_INLINE float block_reduce_max(float v, float* smem) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nwaves) ? smem[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  }
  if (threadIdx.x == 0) smem[0] = v;
  __syncthreads();
  v = smem[0];
  __syncthreads();
  return v;
}

// ------------------------------------------------------- grid sizing helper
// Memory-bound kernels: cap grid and stride (guide §6 G11).
inline int capped_grid(long total_blocks, int cap = 2048) {
  return (int)(total_blocks < cap ? total_blocks : cap);
}

}  // namespace cai

#define HIP_CHECK_LAST()                                                        \
  do {                                                                          \
    hipError_t e = hipGetLastError();                                           \
    if (e != hipSuccess) {                                                      \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e));   \
    }                                                                           \
  } while (0)
