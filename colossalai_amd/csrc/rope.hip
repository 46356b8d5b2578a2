// Rotary position embedding (RoPE) for gfx950 — NeoX/Llama pair layout:
// rotate (d, d + D/2) pairs.
//
// MI355X design notes:
//  - cos/sin come from a HOST-precomputed fp32 table [max_pos, D/2]
//    (on-device sinf/cosf turns this memory-bound op VALU-bound —
//    guide Appendix B "trig-heavy ops").
//  - q and k are transformed in ONE launch (both read the same cos/sin rows
//    → table hits L2/L3 once).
//  - bf16 loads/stores vectorized 8-wide; one thread handles 8 rotation
//    pairs (reads 8 lo + 8 hi elements).
//  - backward is the transpose rotation (sin sign flip), same kernel.
//
// Layout: q [tokens, Hq, D], k [tokens, Hkv, D] (bshd flattened), positions
// int32 [tokens]. Reference op: fused_rotary_emb_and_cache_kernel.cu
// (training-side subset; inference cache-append variant lives elsewhere).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <algorithm>

#include "common.h"

namespace cai {

constexpr int ROPE_BLOCK = 256;

template <bool BACKWARD>
__global__ __launch_bounds__(ROPE_BLOCK) void rope_kernel(
    unsigned short* __restrict__ q,      // [B, S, Hq, D] view (strided B/S)
    unsigned short* __restrict__ k,      // [B, S, Hkv, D] view (may be null)
    const float* __restrict__ table,     // [max_pos, D] : row = [cos(D/2) | sin(D/2)]
    const int* __restrict__ positions,   // [B*S] (may be null -> pos = s)
    long qbs, long qts, long kbs, long kts,
    long tokens,                         // B*S
    int seq_len,                         // S
    int Hq,
    int Hkv,
    int D) {
  const int half = D / 2;
  const int vec_per_head = half / 8;                // rotation pairs, 8 at a time
  const long q_units = tokens * (long)Hq * vec_per_head;
  const long k_units = k ? tokens * (long)Hkv * vec_per_head : 0;
  const long total = q_units + k_units;

  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const bool is_q = idx < q_units;
    const long u = is_q ? idx : idx - q_units;
    const int H = is_q ? Hq : Hkv;
    const int vec = (int)(u % vec_per_head);
    const long th = u / vec_per_head;  // token*H + head
    const long token = th / H;
    const int head = (int)(th % H);
    const long bb = token / seq_len;
    const long ss = token % seq_len;
    const int pos = positions ? positions[token] : (int)ss;

    unsigned short* base = (is_q ? q : k) + bb * (is_q ? qbs : kbs) + ss * (is_q ? qts : kts) + (long)head * D;
    const float* crow = table + (long)pos * D + vec * 8;
    const float* srow = crow + half;

    short8 lo = *reinterpret_cast<const short8*>(base + vec * 8);
    short8 hi = *reinterpret_cast<const short8*>(base + half + vec * 8);
    float4v c0 = *reinterpret_cast<const float4v*>(crow);
    float4v c1 = *reinterpret_cast<const float4v*>(crow + 4);
    float4v s0 = *reinterpret_cast<const float4v*>(srow);
    float4v s1 = *reinterpret_cast<const float4v*>(srow + 4);

    short8 olo, ohi;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float c = j < 4 ? c0[j] : c1[j - 4];
      float s = j < 4 ? s0[j] : s1[j - 4];
      if (BACKWARD) s = -s;
      const float x = bf2f((unsigned short)lo[j]);
      const float y = bf2f((unsigned short)hi[j]);
      olo[j] = (short)f2bf(x * c - y * s);
      ohi[j] = (short)f2bf(y * c + x * s);
    }
    *reinterpret_cast<short8*>(base + vec * 8) = olo;
    *reinterpret_cast<short8*>(base + half + vec * 8) = ohi;
  }
}

// Decode-step KV append: k/v [B, Hkv, D] current-token states scatter into
// the paged pools at per-sequence physical rows — both tensors in ONE
// launch (reference: decode_kv_cache_memcpy_kernel.cu). 16 B/lane copies.
__global__ __launch_bounds__(ROPE_BLOCK) void kv_cache_append_kernel(
    const unsigned short* __restrict__ k, const unsigned short* __restrict__ v,
    unsigned short* __restrict__ kpool, unsigned short* __restrict__ vpool,
    const int* __restrict__ slot_rows, long units, int hd8) {
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < 2 * units;
       idx += (long)gridDim.x * blockDim.x) {
    const bool is_k = idx < units;
    const long u = is_k ? idx : idx - units;
    const long b = u / hd8;
    const int off = (int)(u % hd8) * 8;
    const long dst = (long)slot_rows[b] * hd8 * 8 + off;
    const short8 val = *reinterpret_cast<const short8*>((is_k ? k : v) + u * 8);
    *reinterpret_cast<short8*>((is_k ? kpool : vpool) + dst) = val;
  }
}

void kv_cache_append(at::Tensor k, at::Tensor v, at::Tensor kpool, at::Tensor vpool,
                     at::Tensor slot_rows) {
  TORCH_CHECK(k.dim() == 3 && k.is_contiguous() && v.is_contiguous(), "kv [B,Hkv,D] contiguous");
  TORCH_CHECK(k.scalar_type() == at::kBFloat16 && kpool.scalar_type() == at::kBFloat16);
  TORCH_CHECK(slot_rows.scalar_type() == at::kInt && slot_rows.is_contiguous());
  const int B = (int)k.size(0), Hkv = (int)k.size(1), D = (int)k.size(2);
  TORCH_CHECK(D % 8 == 0);
  const int hd8 = Hkv * D / 8;
  const long units = (long)B * hd8;
  auto stream = at::hip::getCurrentHIPStream();
  const int blocks = (int)std::min((2 * units + ROPE_BLOCK - 1) / ROPE_BLOCK, (long)1024);
  hipLaunchKernelGGL(kv_cache_append_kernel, dim3(blocks), dim3(ROPE_BLOCK), 0, stream.stream(),
                     (const unsigned short*)k.data_ptr(), (const unsigned short*)v.data_ptr(),
                     (unsigned short*)kpool.data_ptr(), (unsigned short*)vpool.data_ptr(),
                     slot_rows.data_ptr<int>(), units, hd8);
  HIP_CHECK_LAST();
}

// In-place RoPE on q (and optionally k), both [B,S,H,D] views (strided B/S ok).
void rope_inplace(at::Tensor q, c10::optional<at::Tensor> k, at::Tensor table,
                  c10::optional<at::Tensor> positions, bool backward) {
  TORCH_CHECK(q.dim() == 4 && q.scalar_type() == at::kBFloat16, "rope: q must be bf16 [B,S,H,D]");
  TORCH_CHECK(q.stride(3) == 1 && q.stride(2) == q.size(3), "rope: q must be dense in [H,D]");
  const int B = (int)q.size(0);
  const int S = (int)q.size(1);
  const int Hq = (int)q.size(2);
  const int D = (int)q.size(3);
  const long tokens = (long)B * S;
  TORCH_CHECK(D % 16 == 0, "rope: head dim must be divisible by 16");
  TORCH_CHECK(table.scalar_type() == at::kFloat && table.is_contiguous() && table.size(-1) == D,
              "rope: table must be fp32 [max_pos, D] (cos|sin halves)");
  int Hkv = 0;
  unsigned short* kptr = nullptr;
  long kbs = 0, kts = 0;
  if (k.has_value()) {
    TORCH_CHECK(k->dim() == 4 && k->scalar_type() == at::kBFloat16 && k->size(3) == D);
    TORCH_CHECK(k->stride(3) == 1 && k->stride(2) == D, "rope: k must be dense in [H,D]");
    TORCH_CHECK(k->size(0) == B && k->size(1) == S, "rope: q/k B,S mismatch");
    Hkv = (int)k->size(2);
    kptr = (unsigned short*)k->data_ptr();
    kbs = k->stride(0);
    kts = k->stride(1);
  }
  const int* pos_ptr = nullptr;
  if (positions.has_value()) {
    TORCH_CHECK(positions->scalar_type() == at::kInt && positions->is_contiguous());
    TORCH_CHECK(positions->numel() == tokens, "rope: positions must have one entry per token");
    pos_ptr = positions->data_ptr<int>();
  }
  auto stream = at::hip::getCurrentHIPStream();
  const long total = tokens * ((long)Hq + Hkv) * (D / 16);
  const int grid = capped_grid((total + ROPE_BLOCK - 1) / ROPE_BLOCK, 4096);
  if (backward) {
    hipLaunchKernelGGL((rope_kernel<true>), dim3(grid), dim3(ROPE_BLOCK), 0, stream.stream(),
                       (unsigned short*)q.data_ptr(), kptr, table.data_ptr<float>(), pos_ptr,
                       q.stride(0), q.stride(1), kbs, kts, tokens, S, Hq, Hkv, D);
  } else {
    hipLaunchKernelGGL((rope_kernel<false>), dim3(grid), dim3(ROPE_BLOCK), 0, stream.stream(),
                       (unsigned short*)q.data_ptr(), kptr, table.data_ptr<float>(), pos_ptr,
                       q.stride(0), q.stride(1), kbs, kts, tokens, S, Hq, Hkv, D);
  }
  HIP_CHECK_LAST();
}

}  // namespace cai
