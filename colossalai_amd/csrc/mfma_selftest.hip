// MFMA fragment-layout self-test for gfx950.
//
// The flash-attention kernels depend on the exact lane→element mappings of
// mfma_f32_16x16x32_bf16 / mfma_f32_32x32x16_bf16 A/B/C fragments and on
// v_permlane32_swap semantics. C/D layouts are HW-verified in the CDNA4
// guide; A/B layouts are derived from CDNA3 docs with K doubled. This file
// computes a single MFMA tile with those assumed layouts so a GPU test can
// compare against a torch fp32 matmul with RANDOM ASYMMETRIC inputs (which
// detects operand/output transposes — symmetric tests do not).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));

DEV_INLINE __bf16 f2bf16t(float f) {
  union { unsigned short u; __bf16 b; } cvt;
  cvt.u = f2bf(f);
  return cvt.b;
}

// C = A(16x32) @ B(32x16), all fp32 in/out (converted to bf16 inside).
__global__ void mfma_16x16x32_probe(const float* A, const float* B, float* C) {
  const int lane = threadIdx.x & 63;
  bf16x8_t a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // A: lane holds row (lane&15), cols (lane>>4)*8 + j   [assumed]
    a[j] = f2bf16t(A[(lane & 15) * 32 + (lane >> 4) * 8 + j]);
    // B: lane holds col (lane&15), rows (lane>>4)*8 + j   [assumed]
    b[j] = f2bf16t(B[((lane >> 4) * 8 + j) * 16 + (lane & 15)]);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    // C: col = lane&15, row = (lane>>4)*4 + j   [verified, guide §3]
    C[((lane >> 4) * 4 + j) * 16 + (lane & 15)] = acc[j];
  }
}

// C = A(32x16) @ B(16x32)
__global__ void mfma_32x32x16_probe(const float* A, const float* B, float* C) {
  const int lane = threadIdx.x & 63;
  bf16x8_t a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // A: lane holds row (lane&31), cols (lane>>5)*8 + j   [assumed]
    a[j] = f2bf16t(A[(lane & 31) * 16 + (lane >> 5) * 8 + j]);
    // B: lane holds col (lane&31), rows (lane>>5)*8 + j   [assumed]
    b[j] = f2bf16t(B[((lane >> 5) * 8 + j) * 32 + (lane & 31)]);
  }
  f32x16 acc;
#pragma unroll
  for (int j = 0; j < 16; ++j) acc[j] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    // C: col = lane&31, row = (j&3) + 8*(j>>2) + 4*(lane>>5)   [verified, guide §3]
    C[((j & 3) + 8 * (j >> 2) + 4 * (lane >> 5)) * 32 + (lane & 31)] = acc[j];
  }
}

// permlane32_swap probe: out0/out1 record what each lane's two registers
// hold after the swap when inputs are (x = lane, y = 1000 + lane).
__global__ void permlane32_probe(int* out0, int* out1) {
  const int lane = threadIdx.x & 63;
  int x = lane;
  int y = 1000 + lane;
  auto pair = __builtin_amdgcn_permlane32_swap(x, y, false, false);
  out0[lane] = pair[0];
  out1[lane] = pair[1];
}

// ds_read_b64_tr_b16 semantics probe. LDS holds u16 value == element index.
// Each lane supplies addr = base + 8*lane bytes (element 4*lane); the output
// records which elements land in lane l's 4 result slots, distinguishing
//  (a) per-lane strided gather: lane l gets {4l, 4l+16, 4l+32, 4l+48}
//  (b) 16-lane weave:           lane l gets {(l&15)+16j+(l>>4)*64}
typedef __bf16 bf16x4v_t __attribute__((ext_vector_type(4)));

__global__ void tr16_probe(int* out) {  // out[64*4]
  __shared__ unsigned short buf[256];
  const int lane = threadIdx.x & 63;
  for (int i = lane; i < 256; i += 64) buf[i] = (unsigned short)i;
  __syncthreads();
  typedef __attribute__((address_space(3))) bf16x4v_t* lds_v4p;
  bf16x4v_t v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_v4p)&buf[lane * 4]);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    union { __bf16 b; unsigned short u; } c;
    c.b = v[j];
    out[lane * 4 + j] = (int)c.u;
  }
}

std::vector<at::Tensor> mfma_selftest(at::Tensor A16, at::Tensor B16, at::Tensor A32, at::Tensor B32) {
  TORCH_CHECK(A16.sizes() == at::IntArrayRef({16, 32}) && B16.sizes() == at::IntArrayRef({32, 16}));
  TORCH_CHECK(A32.sizes() == at::IntArrayRef({32, 16}) && B32.sizes() == at::IntArrayRef({16, 32}));
  auto opts = A16.options().dtype(at::kFloat);
  auto C16 = at::zeros({16, 16}, opts);
  auto C32 = at::zeros({32, 32}, opts);
  auto p0 = at::zeros({64}, opts.dtype(at::kInt));
  auto p1 = at::zeros({64}, opts.dtype(at::kInt));
  auto tr = at::zeros({64, 4}, opts.dtype(at::kInt));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_16x16x32_probe, dim3(1), dim3(64), 0, stream.stream(),
                     A16.data_ptr<float>(), B16.data_ptr<float>(), C16.data_ptr<float>());
  hipLaunchKernelGGL(mfma_32x32x16_probe, dim3(1), dim3(64), 0, stream.stream(),
                     A32.data_ptr<float>(), B32.data_ptr<float>(), C32.data_ptr<float>());
  hipLaunchKernelGGL(permlane32_probe, dim3(1), dim3(64), 0, stream.stream(),
                     p0.data_ptr<int>(), p1.data_ptr<int>());
  hipLaunchKernelGGL(tr16_probe, dim3(1), dim3(64), 0, stream.stream(), tr.data_ptr<int>());
  HIP_CHECK_LAST();
  return {C16, C32, p0, p1, tr};
}

}  // namespace cai
