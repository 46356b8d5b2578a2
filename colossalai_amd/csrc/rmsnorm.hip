// Fused RMSNorm forward/backward for gfx950 (bf16 in/out, fp32 accumulate).
//
// MI355X design notes:
//  - memory-bound (HBM ≈8 TB/s): all bf16 traffic is vectorized short8
//    (16 B/lane) — scalar bf16 loads cost ~2× (guide §6 G13).
//  - one 256-thread block per row (hidden ≤ 8192 handled with per-thread
//    loops); rows are grid-strided with a capped grid.
//  - forward saves inv_rms (fp32 per row) so backward skips the re-reduce.
//  - fused residual-add variant: h = x + residual is computed once and
//    written back to the residual stream (pre-norm transformer block),
//    saving one full read+write per layer vs separate add.
//  - backward accumulates dweight per block in LDS (thread-owned columns,
//    no atomics), then one global atomicAdd pass per block.
//
// Equivalent reference op: extensions/csrc/kernel/cuda/rms_layernorm_kernel.cu
// (re-designed; warp-size-32 assumptions do not carry to CDNA4).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

constexpr int RMS_BLOCK = 256;
constexpr int MAX_H_PER_THREAD = 32;  // supports hidden up to 8192

template <bool FUSED_ADD>
__global__ __launch_bounds__(RMS_BLOCK) void rmsnorm_fwd_kernel(
    unsigned short* __restrict__ out,          // [rows, H] bf16
    unsigned short* __restrict__ residual,     // [rows, H] bf16 (FUSED_ADD: read+write)
    const unsigned short* __restrict__ input,  // [rows, H] bf16
    const unsigned short* __restrict__ weight, // [H] bf16
    float* __restrict__ inv_rms_out,           // [rows] fp32
    float eps,
    long rows,
    int H) {
  __shared__ float red_smem[RMS_BLOCK / WAVE];
  const int nvec = H / 8;  // H % 8 == 0 enforced on host

  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* x = input + row * H;
    unsigned short* res = FUSED_ADD ? residual + row * H : nullptr;
    unsigned short* y = out + row * H;

    float vals[MAX_H_PER_THREAD];
    float ssq = 0.0f;
    int slot = 0;
    for (int i = threadIdx.x; i < nvec; i += RMS_BLOCK, ++slot) {
      short8 xv = *reinterpret_cast<const short8*>(x + i * 8);
      short8 rv;
      if (FUSED_ADD) rv = *reinterpret_cast<const short8*>(res + i * 8);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float f = bf2f((unsigned short)xv[k]);
        if (FUSED_ADD) f += bf2f((unsigned short)rv[k]);
        vals[slot * 8 + k] = f;
        ssq += f * f;
      }
      if (FUSED_ADD) {
        short8 hv;
#pragma unroll
        for (int k = 0; k < 8; ++k) hv[k] = (short)f2bf(vals[slot * 8 + k]);
        *reinterpret_cast<short8*>(res + i * 8) = hv;
      }
    }
    ssq = block_reduce_sum(ssq, red_smem);
    const float inv_rms = rsqrtf(ssq / (float)H + eps);
    if (threadIdx.x == 0 && inv_rms_out != nullptr) inv_rms_out[row] = inv_rms;

    slot = 0;
    for (int i = threadIdx.x; i < nvec; i += RMS_BLOCK, ++slot) {
      short8 wv = *reinterpret_cast<const short8*>(weight + i * 8);
      short8 yv;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        yv[k] = (short)f2bf(vals[slot * 8 + k] * inv_rms * bf2f((unsigned short)wv[k]));
      }
      *reinterpret_cast<short8*>(y + i * 8) = yv;
    }
  }
}

// backward: x here is the NORMALIZATION INPUT (post residual-add when fused).
//   dx_i = inv_rms * (w_i*dy_i - x_i * inv_rms^2/H * sum_j(w_j*dy_j*x_j))
//   dw_j += dy_j * x_j * inv_rms        (accumulated over rows)
__global__ __launch_bounds__(RMS_BLOCK) void rmsnorm_bwd_kernel(
    unsigned short* __restrict__ dx,            // [rows, H] bf16 out
    float* __restrict__ dweight,                // [H] fp32 out (pre-zeroed)
    const unsigned short* __restrict__ dy,      // [rows, H] bf16
    const unsigned short* __restrict__ x,       // [rows, H] bf16
    const unsigned short* __restrict__ weight,  // [H] bf16
    const float* __restrict__ inv_rms,          // [rows]
    long rows,
    int H) {
  __shared__ float red_smem[RMS_BLOCK / WAVE];
  extern __shared__ float dw_local[];  // [H] fp32, thread t owns columns t::RMS_BLOCK
  const int nvec = H / 8;

  for (int i = threadIdx.x; i < nvec; i += RMS_BLOCK) {
#pragma unroll
    for (int k = 0; k < 8; ++k) dw_local[i * 8 + k] = 0.0f;
  }
  __syncthreads();

  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + row * H;
    const unsigned short* dyr = dy + row * H;
    unsigned short* dxr = dx + row * H;
    const float r = inv_rms[row];

    float xv_f[MAX_H_PER_THREAD], dyw_f[MAX_H_PER_THREAD];
    float dot = 0.0f;
    int slot = 0;
    for (int i = threadIdx.x; i < nvec; i += RMS_BLOCK, ++slot) {
      short8 xv = *reinterpret_cast<const short8*>(xr + i * 8);
      short8 dv = *reinterpret_cast<const short8*>(dyr + i * 8);
      short8 wv = *reinterpret_cast<const short8*>(weight + i * 8);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float xf = bf2f((unsigned short)xv[k]);
        const float df = bf2f((unsigned short)dv[k]);
        const float wf = bf2f((unsigned short)wv[k]);
        xv_f[slot * 8 + k] = xf;
        dyw_f[slot * 8 + k] = df * wf;
        dot += df * wf * xf;
        dw_local[i * 8 + k] += df * xf * r;
      }
    }
    dot = block_reduce_sum(dot, red_smem);
    const float c = dot * r * r / (float)H;

    slot = 0;
    for (int i = threadIdx.x; i < nvec; i += RMS_BLOCK, ++slot) {
      short8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        ov[k] = (short)f2bf(r * (dyw_f[slot * 8 + k] - xv_f[slot * 8 + k] * c));
      }
      *reinterpret_cast<short8*>(dxr + i * 8) = ov;
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < H; i += RMS_BLOCK) {
    atomicAdd(&dweight[i], dw_local[i]);
  }
}

// =============================================================== host side

static void check_rms_args(const at::Tensor& t, int H) {
  TORCH_CHECK(t.is_contiguous(), "rmsnorm: tensors must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, "rmsnorm: bf16 only");
  TORCH_CHECK(H % 8 == 0 && H <= MAX_H_PER_THREAD * RMS_BLOCK, "rmsnorm: hidden must be %8==0 and <= 8192");
}

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor input, at::Tensor weight, double eps, bool save_inv_rms) {
  const int H = (int)input.size(-1);
  const long rows = input.numel() / H;
  check_rms_args(input, H);
  auto out = at::empty_like(input);
  auto inv_rms = save_inv_rms ? at::empty({rows}, input.options().dtype(at::kFloat)) : at::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid(rows, 8192);
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<false>), dim3(grid), dim3(RMS_BLOCK), 0, stream.stream(),
                     (unsigned short*)out.data_ptr(), nullptr, (const unsigned short*)input.data_ptr(),
                     (const unsigned short*)weight.data_ptr(),
                     save_inv_rms ? inv_rms.data_ptr<float>() : nullptr, (float)eps, rows, H);
  HIP_CHECK_LAST();
  return {out, inv_rms};
}

// h = input + residual (written into residual); out = rmsnorm(h)
std::vector<at::Tensor> rmsnorm_fused_add_fwd(at::Tensor input, at::Tensor residual, at::Tensor weight,
                                              double eps, bool save_inv_rms) {
  const int H = (int)input.size(-1);
  const long rows = input.numel() / H;
  check_rms_args(input, H);
  auto out = at::empty_like(input);
  auto inv_rms = save_inv_rms ? at::empty({rows}, input.options().dtype(at::kFloat)) : at::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid(rows, 8192);
  hipLaunchKernelGGL((rmsnorm_fwd_kernel<true>), dim3(grid), dim3(RMS_BLOCK), 0, stream.stream(),
                     (unsigned short*)out.data_ptr(), (unsigned short*)residual.data_ptr(),
                     (const unsigned short*)input.data_ptr(), (const unsigned short*)weight.data_ptr(),
                     save_inv_rms ? inv_rms.data_ptr<float>() : nullptr, (float)eps, rows, H);
  HIP_CHECK_LAST();
  return {out, inv_rms};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor inv_rms) {
  const int H = (int)x.size(-1);
  const long rows = x.numel() / H;
  check_rms_args(x, H);
  auto dx = at::empty_like(x);
  auto dweight = at::zeros({H}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const int grid = capped_grid(rows, 2048);
  const size_t lds = (size_t)H * sizeof(float);
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(grid), dim3(RMS_BLOCK), lds, stream.stream(),
                     (unsigned short*)dx.data_ptr(), dweight.data_ptr<float>(),
                     (const unsigned short*)dy.data_ptr(), (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)weight.data_ptr(), inv_rms.data_ptr<float>(), rows, H);
  HIP_CHECK_LAST();
  return {dx, dweight};
}

}  // namespace cai
