// Vectorized CPU Adam for offloaded parameter shards
// (reference equivalent: extensions/csrc/kernel/x86/cpu_adam.cpp — AVX
// intrinsics there; here `#pragma omp parallel for simd` lets the
// compiler emit AVX2/AVX-512 for the EPYC host cores, with OpenMP
// spreading the flat shard across them).
//
// Operates on fp32 master/momentum buffers and an fp32 grad staged from
// the device (D2H), with an optional bf16 working-copy write-back that
// the caller H2D-copies — the same contract as the HIP multi-tensor
// kernel, so LowLevelZeroOptimizer's offloaded buckets are a drop-in.

#include <torch/extension.h>

#include <cmath>
#include <cstdint>

namespace cai_cpu {

void cpu_adam_step(
    at::Tensor param,    // fp32 [n] (master)
    at::Tensor grad,     // fp32 [n]
    at::Tensor exp_avg,  // fp32 [n]
    at::Tensor exp_avg_sq,
    at::Tensor param_out,  // bf16 [n] working copy, or empty
    double lr, double beta1, double beta2, double eps, int64_t step,
    bool adamw, bool bias_correction, double weight_decay, double div_scale) {
  TORCH_CHECK(param.is_contiguous() && grad.is_contiguous() && exp_avg.is_contiguous()
              && exp_avg_sq.is_contiguous(), "cpu_adam: tensors must be contiguous");
  TORCH_CHECK(param.scalar_type() == at::kFloat && grad.scalar_type() == at::kFloat,
              "cpu_adam: fp32 param/grad");
  const int64_t n = param.numel();
  const bool has_out = param_out.numel() > 0;
  if (has_out) TORCH_CHECK(param_out.scalar_type() == at::kBFloat16 && param_out.numel() == n);

  float* p = param.data_ptr<float>();
  const float* g = grad.data_ptr<float>();
  float* m = exp_avg.data_ptr<float>();
  float* v = exp_avg_sq.data_ptr<float>();
  uint16_t* out = has_out ? reinterpret_cast<uint16_t*>(param_out.data_ptr()) : nullptr;

  const float b1 = (float)beta1, b2 = (float)beta2;
  const float inv_div = (float)(1.0 / div_scale);
  const float wd = (float)weight_decay;
  const float flr = (float)lr;
  const float feps = (float)eps;
  float bc1 = 1.0f, bc2_sqrt = 1.0f;
  if (bias_correction) {
    bc1 = 1.0f / (1.0f - std::pow(b1, (float)step));
    bc2_sqrt = 1.0f / std::sqrt(1.0f - std::pow(b2, (float)step));
  }

#pragma omp parallel for simd schedule(static)
  for (int64_t i = 0; i < n; ++i) {
    float gi = g[i] * inv_div;
    float pi = p[i];
    if (!adamw) gi += wd * pi;  // L2 mode
    float mi = b1 * m[i] + (1.0f - b1) * gi;
    float vi = b2 * v[i] + (1.0f - b2) * gi * gi;
    float update = (mi * bc1) / (std::sqrt(vi) * bc2_sqrt + feps);
    if (adamw) update += wd * pi;
    pi -= flr * update;
    p[i] = pi;
    m[i] = mi;
    v[i] = vi;
    if (out) {
      // round-to-nearest-even fp32 -> bf16
      uint32_t bits;
      __builtin_memcpy(&bits, &pi, 4);
      bits += 0x7FFF + ((bits >> 16) & 1);
      out[i] = (uint16_t)(bits >> 16);
    }
  }
}

}  // namespace cai_cpu

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cpu_adam_step", &cai_cpu::cpu_adam_step,
        "vectorized (omp simd) Adam/AdamW step on CPU-resident fp32 buffers");
}
