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
  // the native cast is ONE op (and adjacent pairs fuse into
  // v_cvt_pk_bf16_f32, RNE); the old manual round-to-nearest-even
  // bit-twiddle cost 4 VALU ops per element
  __bf16 h = (__bf16)f;
  return *reinterpret_cast<unsigned short*>(&h);
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

// ------------------------------------------- subtiled LDS tiles + tr-read
// Shared by flash_attn.hip / grouped_gemm.hip. "Subtiled row-major":
// [COLS/16 groups][ROWS][16 bf16], 32 B row pitch, 16 B pad between groups.
// Row-slice fragment reads (8 consecutive cols at fixed row) are single
// 16 B vector loads; column fragments use ds_read_b64_tr_b16 (HW-verified
// weave semantics: per-lane 64-bit reads, 16-lane redistribution
// out[l][j] = pool[(l&15)+16j] — see flash_attn.hip header / tr16_probe).
typedef __bf16 bf16x8_v __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4_v __attribute__((ext_vector_type(4)));

template <int ROWS>
__host__ __device__ constexpr int sub_pitch() { return ROWS * 32 + 16; }

template <int ROWS>
DEV_INLINE int sub_off(int row, int col) {
  return (col >> 4) * sub_pitch<ROWS>() + row * 32 + (col & 15) * 2;
}

template <int ROWS, int COLS>
__host__ __device__ constexpr int sub_bytes() { return (COLS / 16) * sub_pitch<ROWS>(); }

DEV_INLINE bf16x8_v ld_g16b(const unsigned short* p) {
  return *reinterpret_cast<const bf16x8_v*>(p);
}

DEV_INLINE void st_lds16b(char* lds, int byte, bf16x8_v v) {
  *reinterpret_cast<bf16x8_v*>(lds + byte) = v;
}

DEV_INLINE bf16x8_v ld_lds16b(const char* lds, int byte) {
  return *reinterpret_cast<const bf16x8_v*>(lds + byte);
}

DEV_INLINE bf16x4_v ld_tr16(const char* lds, int byte) {
  typedef __attribute__((address_space(3))) bf16x4_v* lds_v4p;
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_v4p)(lds + byte));
}

// Full 8-element column fragment from a subtiled tile: lane l receives
// tile[k_base + (l>>5)*8 + e][cg_base*16 + (l&31)] for e = 0..7 — the A/B
// fragment of mfma_32x32x16 with the tile's ROW axis as the MFMA k-dim.
template <int ROWS>
DEV_INLINE bf16x8_v ld_frag_tr(const char* tile, int lane, int k_base, int cg_base) {
  const int m = lane & 15;
  const int cg = cg_base + ((lane >> 4) & 1);
  const int row = k_base + ((lane >> 5) * 8) + (m >> 2);
  const int base = cg * sub_pitch<ROWS>() + (m & 3) * 8;
  bf16x4_v lo = ld_tr16(tile, base + row * 32);
  bf16x4_v hi = ld_tr16(tile, base + (row + 4) * 32);
  bf16x8_v r;
  r[0] = lo[0]; r[1] = lo[1]; r[2] = lo[2]; r[3] = lo[3];
  r[4] = hi[0]; r[5] = hi[1]; r[6] = hi[2]; r[7] = hi[3];
  return r;
}

// ------------------------------------------------------- grid sizing helper
// Memory-bound kernels: cap grid and stride (guide §6 G11).
inline int capped_grid(long total_blocks, int cap = 2048) {
  return (int)(total_blocks < cap ? total_blocks : cap);
}


// XCD-aware blockIdx remap (guide T1, bijective m204 form): dispatch-order
// block f lands on XCD f%8; remapping logical ids so each XCD owns a
// CONTIGUOUS chunk of the (y-major) grid makes neighbor blocks — which
// share K/V (attention) or operand panels (GEMM) — hit that XCD's
// private 4 MiB L2 instead of re-streaming HBM.
// 3-D variant: z (e.g. expert) is the slowest logical axis, so contiguous
// per-XCD chunks stay within one z and share its operand panels.
DEV_INLINE void xcd_swizzle_xyz(int& x, int& y, int& z) {
  const int gx = (int)gridDim.x, gy = (int)gridDim.y;
  const int f = ((int)blockIdx.z * gy + (int)blockIdx.y) * gx + (int)blockIdx.x;
  const int nwg = gx * gy * (int)gridDim.z;
  const int q = nwg >> 3, r = nwg & 7, xcd = f & 7, i = f >> 3;
  const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  x = swz % gx;
  y = (swz / gx) % gy;
  z = swz / (gx * gy);
}

DEV_INLINE void xcd_swizzle_xy(int& x, int& y) {
  const int gx = (int)gridDim.x;
  const int f = (int)blockIdx.y * gx + (int)blockIdx.x;
  const int nwg = gx * (int)gridDim.y;
  const int q = nwg >> 3, r = nwg & 7, xcd = f & 7, i = f >> 3;
  const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + i;
  x = swz % gx;
  y = swz / gx;
}

}  // namespace cai

#define HIP_CHECK_LAST()                                                        \
  do {                                                                          \
    hipError_t e = hipGetLastError();                                           \
    if (e != hipSuccess) {                                                      \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e));   \
    }                                                                           \
  } while (0)
