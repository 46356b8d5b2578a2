// MoE combine kernels, gfx950 (reference equivalent:
// extensions/csrc/kernel/cuda/moe_kernel.cu combine_forward/backward).
//
// The dispatch side (rows sorted by destination expert) is a plain gather
// that hipBLASLt-adjacent torch index_select already does at HBM speed;
// the COMBINE side is where fusion pays: un-permute + routing-weight
// multiply + top-k sum collapse into one pass so the [T, k, H] slot tensor
// is never materialized. One 256-thread block per token, bf16 rows
// streamed 16 B/lane, fp32 accumulation; the backward reuses the token's
// dout row for both dy (scatter by the inverse permutation — each y row
// written exactly once, no atomics) and dw (block reduction per slot).
//
// Index convention (set up in Python with one argsort):
//   inv[t*k + j] = row of y holding token t's j-th routed copy
//   topw[t, j]   = routing weight (fp32)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

constexpr int MOE_BLOCK = 256;

__global__ __launch_bounds__(MOE_BLOCK) void moe_combine_fwd_kernel(
    const unsigned short* __restrict__ Y,   // [N, H]
    const int* __restrict__ inv,            // [T*k]
    const float* __restrict__ W,            // [T, k]
    unsigned short* __restrict__ OUT,       // [T, H]
    long T, long H, int k) {
  for (long t = blockIdx.x; t < T; t += gridDim.x) {
    for (long h = threadIdx.x * 8L; h < H; h += (long)blockDim.x * 8L) {
      float acc[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = 0.0f;
      for (int j = 0; j < k; ++j) {
        const float w = W[t * k + j];
        const long row = inv[t * k + j];
        short8 yv = *reinterpret_cast<const short8*>(Y + row * H + h);
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += w * bf2f((unsigned short)yv[e]);
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) OUT[t * H + h + e] = f2bf(acc[e]);
    }
  }
}

__global__ __launch_bounds__(MOE_BLOCK) void moe_combine_bwd_kernel(
    const unsigned short* __restrict__ DOUT,  // [T, H]
    const unsigned short* __restrict__ Y,     // [N, H]
    const int* __restrict__ inv,              // [T*k]
    const float* __restrict__ W,              // [T, k]
    unsigned short* __restrict__ DY,          // [N, H]
    float* __restrict__ DW,                   // [T, k]
    long T, long H, int k) {
  __shared__ float red[MOE_BLOCK / 64];
  for (long t = blockIdx.x; t < T; t += gridDim.x) {
    for (int j = 0; j < k; ++j) {
      const float w = W[t * k + j];
      const long row = inv[t * k + j];
      float dot = 0.0f;
      for (long h = threadIdx.x * 8L; h < H; h += (long)blockDim.x * 8L) {
        short8 dv = *reinterpret_cast<const short8*>(DOUT + t * H + h);
        short8 yv = *reinterpret_cast<const short8*>(Y + row * H + h);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float d = bf2f((unsigned short)dv[e]);
          dot += d * bf2f((unsigned short)yv[e]);
          DY[row * H + h + e] = f2bf(w * d);
        }
      }
      // block-reduce dot -> DW[t, j]
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) dot += __shfl_xor(dot, off);
      const int wv = threadIdx.x / 64;
      if ((threadIdx.x & 63) == 0) red[wv] = dot;
      __syncthreads();
      if (threadIdx.x == 0) {
        float s = 0.0f;
        for (int i = 0; i < MOE_BLOCK / 64; ++i) s += red[i];
        DW[t * k + j] = s;
      }
      __syncthreads();
    }
  }
}

at::Tensor moe_combine_fwd(at::Tensor y, at::Tensor inv, at::Tensor topw) {
  TORCH_CHECK(y.dim() == 2 && y.scalar_type() == at::kBFloat16 && y.is_contiguous(), "y must be [N,H] bf16");
  TORCH_CHECK(inv.scalar_type() == at::kInt && inv.is_contiguous());
  TORCH_CHECK(topw.dim() == 2 && topw.scalar_type() == at::kFloat && topw.is_contiguous());
  const long T = topw.size(0), H = y.size(1);
  const int k = (int)topw.size(1);
  TORCH_CHECK(H % 8 == 0, "moe_combine: H must be a multiple of 8");
  TORCH_CHECK(inv.numel() == T * k);
  auto out = at::empty({T, H}, y.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid(T, 8192);
  hipLaunchKernelGGL(moe_combine_fwd_kernel, dim3(grid), dim3(MOE_BLOCK), 0, stream.stream(),
                     (const unsigned short*)y.data_ptr(), inv.data_ptr<int>(),
                     topw.data_ptr<float>(), (unsigned short*)out.data_ptr(), T, H, k);
  HIP_CHECK_LAST();
  return out;
}

std::vector<at::Tensor> moe_combine_bwd(at::Tensor dout, at::Tensor y, at::Tensor inv, at::Tensor topw) {
  TORCH_CHECK(dout.is_contiguous() && y.is_contiguous() && inv.is_contiguous() && topw.is_contiguous());
  const long T = topw.size(0), H = y.size(1);
  const int k = (int)topw.size(1);
  auto dy = at::empty_like(y);
  auto dw = at::empty_like(topw);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid(T, 8192);
  hipLaunchKernelGGL(moe_combine_bwd_kernel, dim3(grid), dim3(MOE_BLOCK), 0, stream.stream(),
                     (const unsigned short*)dout.data_ptr(), (const unsigned short*)y.data_ptr(),
                     inv.data_ptr<int>(), topw.data_ptr<float>(),
                     (unsigned short*)dy.data_ptr(), dw.data_ptr<float>(), T, H, k);
  HIP_CHECK_LAST();
  return {dy, dw};
}

}  // namespace cai
