// Single-token (decode) attention over a KV cache, gfx950.
//
// Memory-bound: the whole job is streaming K/V rows at HBM speed. One
// 256-thread block per (batch, q-head); each 16-lane group owns a strided
// subset of cache positions with its own online-softmax partial (m, l,
// o[128]) — 16 partials merge via LDS with the standard LSE rescale.
// K/V loads are 16 B/lane vectorized (guide §6 G13); GQA folds kv-head
// selection into pointer math.
//
// Reference equivalent: extensions/csrc/kernel/cuda/flash_decoding_attention
// _kernel.cu. Both contiguous caches and vLLM-style paged pools (block
// tables, power-of-two block size folded into shift/mask addressing).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <algorithm>

#include "common.h"

namespace cai {

constexpr int DEC_BLOCK = 256;
constexpr int GROUPS = DEC_BLOCK / 16;  // 16-lane groups

// PAGED=false: KC/VC are [B, Smax, Hkv, D] contiguous caches (Smax = stride).
// PAGED=true: KC/VC are [num_blocks, block_size, Hkv, D] pools; logical
// position s of sequence b lives at physical row
//   block_tables[b*max_blocks + (s >> lbs)] * block_size + (s & (bs-1)).
// SPLIT=true: flash-decoding v2 — blockIdx.y indexes n_splits chunks of each
// sequence; unnormalized (m, l, o) partials land in part_* and a second
// kernel LSE-merges them. B*Hq blocks alone can't fill 256 CUs at decode
// batch sizes; splitting the sequence restores the parallelism
// (reference: flash_decoding_attention_kernel.cu v2 + reduce at :558).
// GQ = Hq/Hkv: one block serves ALL GQ query heads of its kv head, so the
// K/V stream is read ONCE instead of GQ times (decode is pure KV
// bandwidth; GQA models were paying a GQ-fold read tax).
template <int D, bool PAGED, bool SPLIT, int GQ>
__global__ __launch_bounds__(DEC_BLOCK) void decode_attn_kernel(
    const unsigned short* __restrict__ Q,   // [B, Hq, D]
    const unsigned short* __restrict__ KC,
    const unsigned short* __restrict__ VC,
    unsigned short* __restrict__ O,         // [B, Hq, D]
    const int* __restrict__ seq_lens,       // [B] (length INCLUDING current token)
    const int* __restrict__ block_tables,   // [B, max_blocks] (PAGED only)
    int max_blocks, int lbs,                // log2(block_size)
    int B, int Smax, int Hq, int Hkv, float scale,
    int Hg, int gpk,  // Hg = Hq/GQ head-groups per batch; gpk = Hg/Hkv
    int n_splits, float* __restrict__ part_m,  // [B, Hq, NS]
    float* __restrict__ part_l,                // [B, Hq, NS]
    float* __restrict__ part_o) {              // [B, Hq, NS, D]
  constexpr int EPL = D / 16;  // elements per lane (8 for D=128)
  __shared__ float sm_m[GROUPS];
  __shared__ float sm_l[GROUPS];
  __shared__ float sm_o[GROUPS][D];

  const int bh = blockIdx.x;                // over B * Hg
  const int b = bh / Hg;
  const int gidx = bh % Hg;                 // q-head group
  const int hk = gidx / gpk;                // its kv head
  const int S = seq_lens[b];
  const int g = threadIdx.x / 16;   // group id
  const int e = threadIdx.x % 16;   // lane-in-group: owns d = e*EPL..+EPL

  const long kv_tok = (long)Hkv * D;
  const long seq_off = PAGED ? 0 : (long)b * Smax * kv_tok;
  const unsigned short* kbase = KC + seq_off + (long)hk * D + e * EPL;
  const unsigned short* vbase = VC + seq_off + (long)hk * D + e * EPL;
  const int* bt = PAGED ? block_tables + (long)b * max_blocks : nullptr;

  float qf[GQ][EPL];
#pragma unroll
  for (int qh = 0; qh < GQ; ++qh) {
    const unsigned short* q = Q + ((long)b * Hq + gidx * GQ + qh) * D + e * EPL;
#pragma unroll
    for (int j = 0; j < EPL; ++j) qf[qh][j] = bf2f(q[j]);
  }

  int lo = 0, hi = S;
  if constexpr (SPLIT) {
    const int chunk = (S + n_splits - 1) / n_splits;
    lo = blockIdx.y * chunk;
    hi = min(S, lo + chunk);
  }

  float m[GQ], l[GQ], o[GQ][EPL];
#pragma unroll
  for (int qh = 0; qh < GQ; ++qh) {
    m[qh] = -INFINITY;
    l[qh] = 0.0f;
#pragma unroll
    for (int j = 0; j < EPL; ++j) o[qh][j] = 0.0f;
  }

  for (int s = lo + g; s < hi; s += GROUPS) {
    const long row = PAGED ? (((long)bt[s >> lbs] << lbs) | (s & ((1 << lbs) - 1))) : (long)s;
    const unsigned short* kp = kbase + row * kv_tok;
    float kfv[EPL];
    if constexpr (EPL == 8) {
      short8 kv = *reinterpret_cast<const short8*>(kp);
#pragma unroll
      for (int j = 0; j < 8; ++j) kfv[j] = bf2f((unsigned short)kv[j]);
    } else {
      short4v kv = *reinterpret_cast<const short4v*>(kp);
#pragma unroll
      for (int j = 0; j < EPL; ++j) kfv[j] = bf2f((unsigned short)kv[j]);
    }
    float vfv[EPL];
    const unsigned short* vp = vbase + row * kv_tok;
    if constexpr (EPL == 8) {
      short8 vv = *reinterpret_cast<const short8*>(vp);
#pragma unroll
      for (int j = 0; j < 8; ++j) vfv[j] = bf2f((unsigned short)vv[j]);
    } else {
      short4v vv = *reinterpret_cast<const short4v*>(vp);
#pragma unroll
      for (int j = 0; j < EPL; ++j) vfv[j] = bf2f((unsigned short)vv[j]);
    }
#pragma unroll
    for (int qh = 0; qh < GQ; ++qh) {
      float dot = 0.0f;
#pragma unroll
      for (int j = 0; j < EPL; ++j) dot += qf[qh][j] * kfv[j];
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) dot += __shfl_xor(dot, off);
      const float sc = dot * scale;
      const float m_new = fmaxf(m[qh], sc);
      const float corr = (m[qh] == -INFINITY) ? 0.0f : __expf(m[qh] - m_new);
      const float pp = __expf(sc - m_new);
      l[qh] = l[qh] * corr + pp;
#pragma unroll
      for (int j = 0; j < EPL; ++j) o[qh][j] = o[qh][j] * corr + pp * vfv[j];
      m[qh] = m_new;
    }
  }

  // merge the GROUPS partials, one q-head at a time (shared LDS scratch)
#pragma unroll
  for (int qh = 0; qh < GQ; ++qh) {
    if (qh > 0) __syncthreads();  // previous head's reduction done
    if (e == 0) {
      sm_m[g] = m[qh];
      sm_l[g] = l[qh];
    }
#pragma unroll
    for (int j = 0; j < EPL; ++j) sm_o[g][e * EPL + j] = o[qh][j];
    __syncthreads();

    if (g == 0) {
      float m_all = -INFINITY;
#pragma unroll
      for (int i = 0; i < GROUPS; ++i) m_all = fmaxf(m_all, sm_m[i]);
      float l_all = 0.0f;
      float acc[EPL];
#pragma unroll
      for (int j = 0; j < EPL; ++j) acc[j] = 0.0f;
#pragma unroll
      for (int i = 0; i < GROUPS; ++i) {
        const float c = (sm_m[i] == -INFINITY) ? 0.0f : __expf(sm_m[i] - m_all);
        l_all += sm_l[i] * c;
#pragma unroll
        for (int j = 0; j < EPL; ++j) acc[j] += sm_o[i][e * EPL + j] * c;
      }
      const int h = gidx * GQ + qh;
      if constexpr (SPLIT) {
        const long pslot = ((long)b * Hq + h) * n_splits + blockIdx.y;
        if (e == 0) {
          part_m[pslot] = m_all;
          part_l[pslot] = l_all;
        }
        float* po = part_o + pslot * D + e * EPL;
#pragma unroll
        for (int j = 0; j < EPL; ++j) po[j] = acc[j];
      } else {
        const float inv_l = l_all > 0.0f ? 1.0f / l_all : 0.0f;
        unsigned short* op = O + ((long)b * Hq + h) * D + e * EPL;
#pragma unroll
        for (int j = 0; j < EPL; ++j) op[j] = f2bf(acc[j] * inv_l);
      }
    }
  }
}

// LSE-merge of the split partials: one D-thread block per (b, h).
template <int D>
__global__ __launch_bounds__(D) void decode_reduce_kernel(
    const float* __restrict__ part_m, const float* __restrict__ part_l,
    const float* __restrict__ part_o, unsigned short* __restrict__ O,
    int Hq, int n_splits) {
  const long bh = blockIdx.x;
  const int d = threadIdx.x;
  const long base = bh * n_splits;
  float m_all = -INFINITY;
  for (int i = 0; i < n_splits; ++i) m_all = fmaxf(m_all, part_m[base + i]);
  float l_all = 0.0f, acc = 0.0f;
  for (int i = 0; i < n_splits; ++i) {
    const float pm = part_m[base + i];
    const float c = (pm == -INFINITY) ? 0.0f : __expf(pm - m_all);
    l_all += part_l[base + i] * c;
    acc += part_o[(base + i) * D + d] * c;
  }
  const float inv_l = l_all > 0.0f ? 1.0f / l_all : 0.0f;
  O[bh * D + d] = f2bf(acc * inv_l);
}

// Pick n_splits to fill the chip: B*Hq blocks alone vs 256 CUs (2 blocks/CU
// worth of headroom), bounded so each chunk keeps >= 256 positions of work.
static int pick_splits(int bh_blocks, int s_max) {
  int want = (2 * 256) / std::max(bh_blocks, 1);
  int cap = std::max(s_max / 256, 1);
  return std::max(1, std::min({want, cap, 64}));
}

template <bool PAGED>
static at::Tensor decode_launch(at::Tensor q, at::Tensor kc, at::Tensor vc,
                                const int* bt_ptr, int max_blocks, int lbs, int Smax,
                                at::Tensor seq_lens, double scale, int n_splits) {
  const int B = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
  const int Hkv = (int)kc.size(2);
  const int G = Hq / Hkv;
  // largest supported group factor: those query heads share one block's
  // K/V stream (decode is KV bandwidth; GQA was paying a G-fold read tax)
  const int GQ = (G % 8 == 0) ? 8 : (G % 4 == 0) ? 4 : (G % 2 == 0) ? 2 : 1;
  const int Hg = Hq / GQ;         // head-groups per batch
  const int gpk = Hg / Hkv;       // groups per kv head
  const int n_bh = B * Hg;
  auto out = at::empty_like(q);
  auto stream = at::hip::getCurrentHIPStream();
  if (n_splits <= 0) n_splits = pick_splits(n_bh, Smax > 0 ? Smax : 4096);
  at::Tensor pm, pl, po;
  float *pm_p = nullptr, *pl_p = nullptr, *po_p = nullptr;
  if (n_splits > 1) {
    auto opts = q.options().dtype(at::kFloat);
    pm = at::empty({(long)B * Hq * n_splits}, opts);
    pl = at::empty({(long)B * Hq * n_splits}, opts);
    po = at::empty({(long)B * Hq * n_splits * D}, opts);
    pm_p = pm.data_ptr<float>(); pl_p = pl.data_ptr<float>(); po_p = po.data_ptr<float>();
  }
#define LAUNCH_DEC(DD, SPL, GQC, GRID)                                                             \
  hipLaunchKernelGGL((decode_attn_kernel<DD, PAGED, SPL, GQC>), GRID, dim3(DEC_BLOCK), 0,          \
                     stream.stream(), (const unsigned short*)q.data_ptr(),                         \
                     (const unsigned short*)kc.data_ptr(), (const unsigned short*)vc.data_ptr(),   \
                     (unsigned short*)out.data_ptr(), seq_lens.data_ptr<int>(), bt_ptr,            \
                     max_blocks, lbs, B, Smax, Hq, Hkv, (float)scale, Hg, gpk, n_splits,           \
                     pm_p, pl_p, po_p)
#define DISPATCH_GQ(DD, SPL, GRID)                                                                 \
  do {                                                                                             \
    if (GQ == 8) LAUNCH_DEC(DD, SPL, 8, GRID);                                                     \
    else if (GQ == 4) LAUNCH_DEC(DD, SPL, 4, GRID);                                                \
    else if (GQ == 2) LAUNCH_DEC(DD, SPL, 2, GRID);                                                \
    else LAUNCH_DEC(DD, SPL, 1, GRID);                                                             \
  } while (0)
  if (n_splits > 1) {
    const dim3 grid(n_bh, n_splits);
    if (D == 128) DISPATCH_GQ(128, true, grid); else DISPATCH_GQ(64, true, grid);
    HIP_CHECK_LAST();
    const dim3 rgrid(B * Hq);
    if (D == 128)
      hipLaunchKernelGGL((decode_reduce_kernel<128>), rgrid, dim3(128), 0, stream.stream(),
                         pm_p, pl_p, po_p, (unsigned short*)out.data_ptr(), Hq, n_splits);
    else
      hipLaunchKernelGGL((decode_reduce_kernel<64>), rgrid, dim3(64), 0, stream.stream(),
                         pm_p, pl_p, po_p, (unsigned short*)out.data_ptr(), Hq, n_splits);
  } else {
    const dim3 grid(n_bh);
    if (D == 128) DISPATCH_GQ(128, false, grid); else DISPATCH_GQ(64, false, grid);
  }
#undef DISPATCH_GQ
#undef LAUNCH_DEC
  HIP_CHECK_LAST();
  return out;
}

at::Tensor decode_attention(at::Tensor q, at::Tensor kcache, at::Tensor vcache, at::Tensor seq_lens,
                            double scale, int64_t n_splits) {
  TORCH_CHECK(q.dim() == 3 && q.scalar_type() == at::kBFloat16 && q.is_contiguous(), "q must be [B,Hq,D] bf16");
  TORCH_CHECK(kcache.is_contiguous() && vcache.is_contiguous(), "kv cache must be contiguous");
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int D = (int)q.size(2);
  const int Smax = (int)kcache.size(1);
  TORCH_CHECK(D == 64 || D == 128, "decode_attention: head dim 64/128");
  return decode_launch<false>(q, kcache, vcache, nullptr, 0, 0, Smax, seq_lens, scale,
                              (int)n_splits);
}

// Paged variant: kpool/vpool [num_blocks, block_size, Hkv, D],
// block_tables [B, max_blocks] int32, block_size a power of two.
at::Tensor decode_attention_paged(at::Tensor q, at::Tensor kpool, at::Tensor vpool,
                                  at::Tensor block_tables, at::Tensor seq_lens, double scale,
                                  int64_t n_splits) {
  TORCH_CHECK(q.dim() == 3 && q.scalar_type() == at::kBFloat16 && q.is_contiguous(), "q must be [B,Hq,D] bf16");
  TORCH_CHECK(kpool.dim() == 4 && kpool.is_contiguous() && vpool.is_contiguous(), "kv pool [NB,BS,Hkv,D]");
  TORCH_CHECK(block_tables.dim() == 2 && block_tables.scalar_type() == at::kInt && block_tables.is_contiguous());
  TORCH_CHECK(seq_lens.scalar_type() == at::kInt && seq_lens.is_contiguous());
  const int D = (int)q.size(2);
  const int BS = (int)kpool.size(1);
  const int max_blocks = (int)block_tables.size(1);
  TORCH_CHECK(D == 64 || D == 128, "decode_attention: head dim 64/128");
  TORCH_CHECK((BS & (BS - 1)) == 0, "block_size must be a power of two");
  int lbs = 0;
  while ((1 << lbs) < BS) ++lbs;
  // Smax for split sizing only: the deepest sequence any block table can hold
  return decode_launch<true>(q, kpool, vpool, block_tables.data_ptr<int>(), max_blocks, lbs,
                             max_blocks * BS, seq_lens, scale, (int)n_splits);
}

}  // namespace cai
