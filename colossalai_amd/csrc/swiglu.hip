// Fused SwiGLU (SiLU(gate) * up) forward/backward for gfx950.
//
// The model computes gate and up with ONE hipBLASLt GEMM producing
// [tokens, 2I] (gate = [:, :I], up = [:, I:]); this kernel fuses the
// activation so the intermediate is read once. Backward writes dgate/dup
// into one [tokens, 2I] buffer that directly feeds the backward GEMM.
// Memory-bound: 16 B/lane vectorized bf16 (guide §6 G13).
//
// Reference op: extensions/csrc/kernel/cuda/activation_kernel.cu
// (silu_and_mul; ours adds the fused backward for training).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

constexpr int SWIGLU_BLOCK = 256;

__global__ __launch_bounds__(SWIGLU_BLOCK) void swiglu_fwd_kernel(
    unsigned short* __restrict__ out,           // [tokens, I]
    const unsigned short* __restrict__ gate_up, // [tokens, 2I]
    long tokens,
    int I) {
  const int nvec = I / 8;
  const long total = tokens * (long)nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long t = idx / nvec;
    const int i = (int)(idx % nvec);
    const unsigned short* row = gate_up + t * (2L * I);
    short8 g = *reinterpret_cast<const short8*>(row + i * 8);
    short8 u = *reinterpret_cast<const short8*>(row + I + i * 8);
    short8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const float gf = bf2f((unsigned short)g[k]);
      const float uf = bf2f((unsigned short)u[k]);
      const float sig = 1.0f / (1.0f + __expf(-gf));
      o[k] = (short)f2bf(gf * sig * uf);
    }
    *reinterpret_cast<short8*>(out + t * I + i * 8) = o;
  }
}

__global__ __launch_bounds__(SWIGLU_BLOCK) void swiglu_bwd_kernel(
    unsigned short* __restrict__ dgate_up,      // [tokens, 2I] out
    const unsigned short* __restrict__ dout,    // [tokens, I]
    const unsigned short* __restrict__ gate_up, // [tokens, 2I]
    long tokens,
    int I) {
  const int nvec = I / 8;
  const long total = tokens * (long)nvec;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long t = idx / nvec;
    const int i = (int)(idx % nvec);
    const unsigned short* row = gate_up + t * (2L * I);
    unsigned short* drow = dgate_up + t * (2L * I);
    short8 g = *reinterpret_cast<const short8*>(row + i * 8);
    short8 u = *reinterpret_cast<const short8*>(row + I + i * 8);
    short8 do_ = *reinterpret_cast<const short8*>(dout + t * I + i * 8);
    short8 dg, du;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const float gf = bf2f((unsigned short)g[k]);
      const float uf = bf2f((unsigned short)u[k]);
      const float df = bf2f((unsigned short)do_[k]);
      const float sig = 1.0f / (1.0f + __expf(-gf));
      const float silu = gf * sig;
      // d silu(g)/dg = sig * (1 + g * (1 - sig))
      dg[k] = (short)f2bf(df * uf * sig * (1.0f + gf * (1.0f - sig)));
      du[k] = (short)f2bf(df * silu);
    }
    *reinterpret_cast<short8*>(drow + i * 8) = dg;
    *reinterpret_cast<short8*>(drow + I + i * 8) = du;
  }
}

at::Tensor swiglu_fwd(at::Tensor gate_up) {
  TORCH_CHECK(gate_up.is_contiguous() && gate_up.scalar_type() == at::kBFloat16, "swiglu: bf16 contiguous only");
  const long twoI = gate_up.size(-1);
  TORCH_CHECK(twoI % 16 == 0, "swiglu: intermediate dim must be divisible by 8");
  const int I = (int)(twoI / 2);
  const long tokens = gate_up.numel() / twoI;
  auto sizes = gate_up.sizes().vec();
  sizes.back() = I;
  auto out = at::empty(sizes, gate_up.options());
  auto stream = at::hip::getCurrentHIPStream();
  const long total = tokens * (I / 8);
  const int grid = capped_grid((total + SWIGLU_BLOCK - 1) / SWIGLU_BLOCK, 4096);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(SWIGLU_BLOCK), 0, stream.stream(),
                     (unsigned short*)out.data_ptr(), (const unsigned short*)gate_up.data_ptr(), tokens, I);
  HIP_CHECK_LAST();
  return out;
}

at::Tensor swiglu_bwd(at::Tensor dout, at::Tensor gate_up) {
  TORCH_CHECK(gate_up.is_contiguous() && dout.is_contiguous(), "swiglu_bwd: contiguous only");
  const long twoI = gate_up.size(-1);
  const int I = (int)(twoI / 2);
  const long tokens = gate_up.numel() / twoI;
  auto dgate_up = at::empty_like(gate_up);
  auto stream = at::hip::getCurrentHIPStream();
  const long total = tokens * (I / 8);
  const int grid = capped_grid((total + SWIGLU_BLOCK - 1) / SWIGLU_BLOCK, 4096);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(SWIGLU_BLOCK), 0, stream.stream(),
                     (unsigned short*)dgate_up.data_ptr(), (const unsigned short*)dout.data_ptr(),
                     (const unsigned short*)gate_up.data_ptr(), tokens, I);
  HIP_CHECK_LAST();
  return dgate_up;
}

}  // namespace cai
