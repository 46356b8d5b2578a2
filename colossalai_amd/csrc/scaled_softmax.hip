// Standalone fused scale+mask+softmax fwd/bwd for gfx950.
//
// Used by models that keep explicit attention scores (the flash kernels
// cover the fused path). One 64-wide wave per row; streaming three-pass
// (max, sum-exp, write) with short8-vectorized bf16 — rows are L1/L2
// resident between passes at these sizes, so the op stays BW-bound.
//
// Reference equivalents: scaled_masked_softmax_kernel.cu and
// scaled_upper_triang_masked_softmax_kernel.cu (warp-32 template scheme
// re-derived for wave64; mask is additive bf16 or implicit causal).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

constexpr int SSM_BLOCK = 256;  // 4 waves -> 4 rows per block iteration

template <bool CAUSAL, bool HAS_MASK>
__global__ __launch_bounds__(SSM_BLOCK) void scaled_softmax_fwd_kernel(
    unsigned short* __restrict__ out,
    const unsigned short* __restrict__ in,    // [rows, Sk]
    const unsigned short* __restrict__ mask,  // [maskB, 1, Sq, Sk] additive bf16
    float scale,
    long rows,
    int Sk,
    int Sq,    // for causal: column limit = row % Sq
    int H,     // heads (mask broadcasts over this dim)
    int maskB) {
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const long wave_rows = (long)gridDim.x * (SSM_BLOCK / WAVE);

  for (long row = blockIdx.x * (SSM_BLOCK / WAVE) + w; row < rows; row += wave_rows) {
    const unsigned short* x = in + row * Sk;
    unsigned short* y = out + row * Sk;
    const unsigned short* m = nullptr;
    if (HAS_MASK) {
      const long b = row / ((long)H * Sq);
      const long sq = row % Sq;
      m = mask + (((maskB > 1 ? b : 0) * Sq) + sq) * Sk;
    }
    const int limit = CAUSAL ? (int)(row % Sq) + 1 : Sk;

    float vmax = -INFINITY;
    for (int i = lane * 8; i < limit; i += WAVE * 8) {
      short8 xv = *reinterpret_cast<const short8*>(x + i);
      short8 mv;
      if (HAS_MASK) mv = *reinterpret_cast<const short8*>(m + i);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        if (i + k < limit) {
          float f = bf2f((unsigned short)xv[k]) * scale;
          if (HAS_MASK) f += bf2f((unsigned short)mv[k]);
          vmax = fmaxf(vmax, f);
        }
      }
    }
    vmax = wave_reduce_max(vmax);
    const float msafe = (vmax == -INFINITY) ? 0.0f : vmax;

    float vsum = 0.0f;
    for (int i = lane * 8; i < limit; i += WAVE * 8) {
      short8 xv = *reinterpret_cast<const short8*>(x + i);
      short8 mv;
      if (HAS_MASK) mv = *reinterpret_cast<const short8*>(m + i);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        if (i + k < limit) {
          float f = bf2f((unsigned short)xv[k]) * scale;
          if (HAS_MASK) f += bf2f((unsigned short)mv[k]);
          vsum += __expf(f - msafe);
        }
      }
    }
    vsum = wave_reduce_sum(vsum);
    const float inv = vsum > 0.0f ? 1.0f / vsum : 0.0f;

    for (int i = lane * 8; i < Sk; i += WAVE * 8) {
      short8 xv = *reinterpret_cast<const short8*>(x + i);
      short8 mv;
      if (HAS_MASK) mv = *reinterpret_cast<const short8*>(m + i);
      short8 yv;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float r = 0.0f;
        if (i + k < limit) {
          float f = bf2f((unsigned short)xv[k]) * scale;
          if (HAS_MASK) f += bf2f((unsigned short)mv[k]);
          r = __expf(f - msafe) * inv;
        }
        yv[k] = (short)f2bf(r);
      }
      *reinterpret_cast<short8*>(y + i) = yv;
    }
  }
}

// dx = scale * y * (dy - sum(dy*y))
__global__ __launch_bounds__(SSM_BLOCK) void scaled_softmax_bwd_kernel(
    unsigned short* __restrict__ dx,
    const unsigned short* __restrict__ dy,
    const unsigned short* __restrict__ y,
    float scale,
    long rows,
    int Sk) {
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const long wave_rows = (long)gridDim.x * (SSM_BLOCK / WAVE);

  for (long row = blockIdx.x * (SSM_BLOCK / WAVE) + w; row < rows; row += wave_rows) {
    const unsigned short* dyr = dy + row * Sk;
    const unsigned short* yr = y + row * Sk;
    unsigned short* dxr = dx + row * Sk;

    float dot = 0.0f;
    for (int i = lane * 8; i < Sk; i += WAVE * 8) {
      short8 dv = *reinterpret_cast<const short8*>(dyr + i);
      short8 yv = *reinterpret_cast<const short8*>(yr + i);
#pragma unroll
      for (int k = 0; k < 8; ++k) dot += bf2f((unsigned short)dv[k]) * bf2f((unsigned short)yv[k]);
    }
    dot = wave_reduce_sum(dot);

    for (int i = lane * 8; i < Sk; i += WAVE * 8) {
      short8 dv = *reinterpret_cast<const short8*>(dyr + i);
      short8 yv = *reinterpret_cast<const short8*>(yr + i);
      short8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float yk = bf2f((unsigned short)yv[k]);
        ov[k] = (short)f2bf(scale * yk * (bf2f((unsigned short)dv[k]) - dot));
      }
      *reinterpret_cast<short8*>(dxr + i) = ov;
    }
  }
}

at::Tensor scaled_masked_softmax_fwd(at::Tensor x, c10::optional<at::Tensor> mask, double scale, bool causal) {
  TORCH_CHECK(x.is_contiguous() && x.scalar_type() == at::kBFloat16, "scaled_softmax: bf16 contiguous");
  const int Sk = (int)x.size(-1);
  const int Sq = (int)x.size(-2);
  const long rows = x.numel() / Sk;
  TORCH_CHECK(Sk % 8 == 0, "scaled_softmax: last dim %8==0");
  auto out = at::empty_like(x);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid((rows + 3) / 4, 4096);
  const int H = x.dim() >= 3 ? (int)x.size(-3) : 1;
  int maskB = 1;
  const unsigned short* mptr = nullptr;
  if (mask.has_value()) {
    TORCH_CHECK(mask->is_contiguous() && mask->scalar_type() == at::kBFloat16 && mask->size(-1) == Sk);
    TORCH_CHECK(mask->size(-2) == Sq, "mask must be [B,1,Sq,Sk]-broadcastable");
    maskB = (int)(mask->numel() / ((long)Sq * Sk));
    mptr = (const unsigned short*)mask->data_ptr();
  }
#define LAUNCH_SSM(C, M)                                                                           \
  hipLaunchKernelGGL((scaled_softmax_fwd_kernel<C, M>), dim3(grid), dim3(SSM_BLOCK), 0,            \
                     stream.stream(), (unsigned short*)out.data_ptr(),                             \
                     (const unsigned short*)x.data_ptr(), mptr, (float)scale, rows, Sk, Sq, H,     \
                     maskB)
  if (causal && mptr) LAUNCH_SSM(true, true);
  else if (causal) LAUNCH_SSM(true, false);
  else if (mptr) LAUNCH_SSM(false, true);
  else LAUNCH_SSM(false, false);
#undef LAUNCH_SSM
  HIP_CHECK_LAST();
  return out;
}

at::Tensor scaled_masked_softmax_bwd(at::Tensor dy, at::Tensor y, double scale) {
  const int Sk = (int)y.size(-1);
  const long rows = y.numel() / Sk;
  auto dx = at::empty_like(y);
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid((rows + 3) / 4, 4096);
  hipLaunchKernelGGL(scaled_softmax_bwd_kernel, dim3(grid), dim3(SSM_BLOCK), 0, stream.stream(),
                     (unsigned short*)dx.data_ptr(), (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)y.data_ptr(), (float)scale, rows, Sk);
  HIP_CHECK_LAST();
  return dx;
}

}  // namespace cai
