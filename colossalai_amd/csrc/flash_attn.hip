// Flash attention forward + backward for gfx950 (CDNA4), bf16, causal, GQA.
//
// MI355X-native design (not a CUDA port):
//  - mfma_f32_32x32x16_bf16 tiles; 64-wide waves; 8 waves per workgroup
//    forward, 4-wave blocks backward.
//  - ALL operand tiles live in LDS in ONE layout: "subtiled row-major"
//    [D/16 col-groups][ROWS][16 elems] with 32 B row pitch (+16 B pad per
//    group to stagger banks). Row-slice fragment reads (8 consecutive d at
//    fixed row) are single 16 B vector loads; COLUMN fragment reads (the
//    transposed consumption that previously needed separate [D][ROWS]
//    tiles built with 64 scalar ds_write_b16 per thread per tile) use
//    gfx950's ds_read_b64_tr_b16 hardware weave (guide T10).
//    HW-verified semantics (tr16_probe, mfma_selftest.hip): each lane reads
//    64 bits at its own address; within a 16-lane group, out[l][j] =
//    pool[(l&15) + 16 j] where pool is the group's reads in lane order.
//    With lane addressing row=(m>>2), byte=(m&3)*8 this delivers
//    tile[row0+j][col m] per lane — a free 4x16 transpose with any row pitch.
//  - forward: each wave owns 32 q rows; K/V tiles of 64 staged subtiled;
//    QK^T operand-swapped (S^T = K.Q^T) so softmax is lane-local; defer-max
//    rescale (T13); async-split staging (T14): next tile's global loads
//    issue before this tile's MFMAs.
//  - backward FA2-split: delta preprocess; dK/dV kernel (per kv-tile, GQA
//    group in registers); dQ kernel (per q-tile). P/dS restage between the
//    MFMA C-layout and A-fragment layout goes through a per-wave 2 KB
//    subtiled tile: packed 8 B stores (4 consecutive rows of the C-layout
//    share a lane) + tr-read A-fragments. Score tiles are processed in
//    32-wide halves, halving live p/dp registers.
//
// Shapes: q [B,S,Hq,D], k/v [B,S,Hkv,D] ("bshd" — no transposes in the
// model), D in {64,128}. lse/delta are [B,Hq,S] fp32.
// Reference behavioral equivalent: flash-attn 2 package as consumed by
// colossalai/shardformer/layer/attn.py (ColoAttention).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4v_t __attribute__((ext_vector_type(4)));

namespace fa {

constexpr int KVB = 64;     // kv rows per tile
constexpr int QW = 32;      // q rows per wave (fwd / dq); kv rows per wave (dkdv)
constexpr int NW = 8;       // waves per block (forward). NW=4 (two independent
                            // blocks/CU for natural stagger) measured 319 vs 519
                            // TF — the doubled K/V staging work loses more than
                            // the barrier decoupling gains; struck.
constexpr int NT = 512;     // threads per block (forward)
// Backward kernels carry dK+dV (or dQ) accumulators plus K/V (or Q/dO)
// fragments in registers (~380-450 VGPR+AGPR): 1 wave/SIMD — 4-wave blocks.
constexpr int NWB = 4;
constexpr int NTB = 256;

DEV_INLINE bf16x8_t ld_g16(const unsigned short* p) {
  return *reinterpret_cast<const bf16x8_t*>(p);
}

DEV_INLINE void st_lds16(char* lds, int byte, bf16x8_t v) {
  *reinterpret_cast<bf16x8_t*>(lds + byte) = v;
}

DEV_INLINE bf16x8_t ld_lds16(const char* lds, int byte) {
  return *reinterpret_cast<const bf16x8_t*>(lds + byte);
}

DEV_INLINE unsigned short bf_raw(__bf16 b) {
  union { __bf16 b; unsigned short u; } c;
  c.b = b;
  return c.u;
}

DEV_INLINE __bf16 f2b(float f) {
  union { unsigned short u; __bf16 b; } c;
  c.u = f2bf(f);
  return c.b;
}

// C-layout row for mfma_f32_32x32x16 register j (guide §3, HW-verified).
DEV_INLINE constexpr int crow(int j, int half) { return (j & 3) + 8 * (j >> 2) + 4 * half; }

// Subtiled row-major LDS tiles + ds_read_b64_tr_b16 column fragments live
// in common.h (shared with grouped_gemm.hip); re-exported into this
// namespace for the kernel bodies below.
using cai::sub_pitch;
using cai::sub_off;
using cai::sub_bytes;
using cai::ld_tr16;
using cai::ld_frag_tr;

// Synchronous staging: load + store back-to-back (transient registers).
template <int D, int ROWS, int NT_>
DEV_INLINE void stage_direct(char* lds, const unsigned short* base, long stride, int row0, int S);

// Register-staged tile for async-split staging (guide T14 / G15): global
// loads for tile t+1 are ISSUED before tile t's compute (results land in
// registers under the MFMAs), LDS writes happen after the barrier. NT_ must
// divide ROWS*CPR.
template <int D, int ROWS, int NT_>
struct RegStage {
  static constexpr int CPR = D / 8;
  static constexpr int NCH = ROWS * CPR / NT_;
  static_assert(ROWS * CPR % NT_ == 0, "RegStage: uneven chunking");
  bf16x8_t r[NCH];

  DEV_INLINE void load(const unsigned short* base, long tok_stride, int row0, int S) {
#pragma unroll
    for (int i = 0; i < NCH; ++i) {
      const int c = (int)threadIdx.x + i * NT_;
      const int row = c / CPR, ch = c % CPR;
      const int grow = max(min(row0 + row, S - 1), 0);
      r[i] = ld_g16(base + (long)grow * tok_stride + ch * 8);
    }
  }
  DEV_INLINE void store_subtiled(char* lds) const {
#pragma unroll
    for (int i = 0; i < NCH; ++i) {
      const int c = (int)threadIdx.x + i * NT_;
      st_lds16(lds, sub_off<ROWS>(c / CPR, (c % CPR) * 8), r[i]);
    }
  }
};

template <int D, int ROWS, int NT_>
DEV_INLINE void stage_direct(char* lds, const unsigned short* base, long stride, int row0, int S) {
  RegStage<D, ROWS, NT_> st;
  st.load(base, stride, row0, S);
  st.store_subtiled(lds);
}

// Swizzled byte offset inside a row-major LDS tile with ROWB bytes per row
// (forward P tile only). XOR spreads a column access across the row's
// 16-byte slots — guide §6 G4.
template <int ROWB>
DEV_INLINE int swz(int row, int byte_in_row) {
  constexpr int CPR_MASK = (ROWB / 16) - 1;
  return row * ROWB + (byte_in_row ^ ((row & CPR_MASK) << 4));
}

}  // namespace fa

// ============================================================ forward kernel
//
// grid: (ceil(S/256), B*Hq). LDS: K_sub + V_sub + P[NW][32][KVB].
//
// Mask generality (reference: colossalai/shardformer/layer/attn.py:139
// prepare_attn_kwargs CAUSAL / PADDED / PADDED_CAUSAL + varlen):
//  - VARLEN=false, CU=nullptr: dense [B,S,H,D].
//  - VARLEN=false, CU=seqlens[B]: right-padded batches; rows/cols beyond
//    seqlens[b] are masked; O=0 and LSE=-inf are written for pad rows so
//    the backward masks them via the existing lse==-inf guard.
//  - VARLEN=true, CU=cu_seqlens[n+1]: packed ragged batch [total,H,D]
//    passed with S=total and grid.y = n_seq*Hq; sequence b spans tokens
//    [CU[b], CU[b+1]); LSE/delta are [Hq, total].
//  - CUK (padded path only): SEPARATE kv-side valid counts — ring-attention
//    pieces where the q chunk and the arriving kv chunk cover different
//    parts of each padded sequence. null -> kv counts = CU.
template <int D, bool VARLEN>
__global__ __launch_bounds__(fa::NT) void fa_fwd_kernel(
    const unsigned short* __restrict__ Q,
    const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V,
    unsigned short* __restrict__ O,
    float* __restrict__ LSE,
    const int* __restrict__ CU,
    const int* __restrict__ CUK,
    long qbs, long qts, long kbs, long kts,
    int B, int S, int Sk, int Hq, int Hkv, float scale, int causal) {
  using namespace fa;
  extern __shared__ char smem[];
  constexpr int KB_BYTES = sub_bytes<KVB, D>();
  // double-buffered K/V: tile t+1 stages into the OTHER buffer while tile
  // t is being consumed — one barrier per tile instead of two
  // (no pointer array: LDS addrspace casts cannot form a static initializer)
  char* Pw = smem + 4 * KB_BYTES;  // + wave*QW*KVB*2

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int ln = lane & 31;

  int swz_x, swz_y;
  xcd_swizzle_xy(swz_x, swz_y);  // same-bh tiles share one XCD's L2 (K/V reuse)
  const int bh = swz_y;
  const int b = bh / Hq, h = bh % Hq;
  const int hk = h / (Hq / Hkv);
  const int q0 = swz_x * (NW * QW);
  const int qw = q0 + w * QW;  // this wave's first q row

  long seq0 = 0;
  int Seff = S;
  if (VARLEN) {
    seq0 = CU[b];
    Seff = CU[b + 1] - (int)seq0;
    if (q0 >= Seff) return;
  } else if (CU != nullptr) {
    Seff = CU[b];
  }
  // kv extent: its own tensor length Sk (ring blocks where kv covers a
  // DIFFERENT — possibly longer — span than q), clipped by per-batch valid
  // counts: CUK when given, else CU only in the symmetric padded case
  int SeffK = VARLEN ? Seff
            : (CUK != nullptr ? CUK[b] : (CU != nullptr ? min(CU[b], Sk) : Sk));

  const unsigned short* q_base = VARLEN ? Q + seq0 * qts + (long)h * D : Q + (long)b * qbs + (long)h * D;
  const unsigned short* k_base = VARLEN ? K + seq0 * kts + (long)hk * D : K + (long)b * kbs + (long)hk * D;
  const unsigned short* v_base = VARLEN ? V + seq0 * kts + (long)hk * D : V + (long)b * kbs + (long)hk * D;
  const long o_stride = (long)Hq * D;  // O/LSE are always packed

  constexpr int DSL = D / 16;  // MFMA K-slices over the head dim
  char* P = Pw + w * (QW * KVB * 2);

  // ---- Q fragments (B-operand of S^T = K·Q^T): lane holds q-col (ln),
  //      d-rows half*8+[0..7] per 16-d slice — one 16 B load per slice.
  const int q_my = min(qw + ln, Seff - 1);
  bf16x8_t qf[DSL];
  {
    const unsigned short* qp = q_base + (long)q_my * qts + half * 8;
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) qf[sl] = ld_g16(qp + sl * 16);
  }

  const float scale2 = scale * 1.44269504f;  // scale * log2(e)
  f32x16 oacc[D / 32];
#pragma unroll
  for (int nb = 0; nb < D / 32; ++nb)
#pragma unroll
    for (int j = 0; j < 16; ++j) oacc[nb][j] = 0.0f;
  float m_run = -INFINITY, l_run = 0.0f;

  const int kv_tiles_all = (SeffK + KVB - 1) / KVB;
  const int kv_tiles = causal ? min(kv_tiles_all, (q0 + NW * QW + KVB - 1) / KVB) : kv_tiles_all;

  // async-split staging (T14): tile 0 staged synchronously, tile t+1's global
  // loads issue before tile t's compute and land in LDS after the barrier.
  RegStage<D, KVB, NT> kstage, vstage;
  // SeffK == 0 blocks skip the tile loop; the clamped loads stay in-bounds
  kstage.load(k_base, kts, 0, SeffK);
  vstage.load(v_base, kts, 0, SeffK);
  kstage.store_subtiled(smem);
  vstage.store_subtiled(smem + KB_BYTES);
  __syncthreads();

  for (int t = 0; t < kv_tiles; ++t) {
    const int k0 = t * KVB;
    const bool has_next = (t + 1 < kv_tiles);
    char* Klds = smem + (t & 1) * 2 * KB_BYTES;
    char* Vlds = Klds + KB_BYTES;
    if (has_next) {
      kstage.load(k_base, kts, k0 + KVB, SeffK);
      vstage.load(v_base, kts, k0 + KVB, SeffK);
    }
    // waves entirely above the diagonal produce nothing (barriers stay uniform)
    const bool active = !(causal && k0 > qw + QW - 1) && (qw < Seff);
    if (active) {
    // ---- S^T = K · Q^T : KVB/32 32-k blocks. C: col = q (ln), row = k (crow).
    // The kb accumulator chains are independent: interleaving them keeps
    // two MFMAs in flight instead of one serial 8-deep chain per block.
    constexpr int KB = KVB / 32;
    float p[KB][16];
    {
      f32x16 acc[KB];
#pragma unroll
      for (int kb = 0; kb < KB; ++kb)
#pragma unroll
        for (int j = 0; j < 16; ++j) acc[kb][j] = 0.0f;
      // software-pipelined A-operand loads: slice sl+1's LDS reads issue
      // UNDER slice sl's MFMAs (the ld->mfma dependency chain was the
      // limiter — each b128 LDS read is ~64 cycles the MFMA waited out)
      bf16x8_t kf_cur[KB], kf_nxt[KB];
#pragma unroll
      for (int kb = 0; kb < KB; ++kb)
        kf_cur[kb] = ld_lds16(Klds, sub_off<KVB>(kb * 32 + ln, half * 8));
#pragma unroll
      for (int sl = 0; sl < DSL; ++sl) {
        if (sl + 1 < DSL) {
#pragma unroll
          for (int kb = 0; kb < KB; ++kb)
            kf_nxt[kb] = ld_lds16(Klds, sub_off<KVB>(kb * 32 + ln, (sl + 1) * 16 + half * 8));
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kb = 0; kb < KB; ++kb)
          acc[kb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf_cur[kb], qf[sl], acc[kb], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int kb = 0; kb < KB; ++kb) kf_cur[kb] = kf_nxt[kb];
      }
      // exp2-domain softmax: v_exp_f32 IS exp2, so fold scale*log2e into
      // the copy-out and every __expf's hidden *log2e mul disappears.
      // m_run / l_run / rowmax live in the log2 domain; LSE converts back
      // at the epilogue.
#pragma unroll
      for (int kb = 0; kb < KB; ++kb)
#pragma unroll
        for (int j = 0; j < 16; ++j) p[kb][j] = acc[kb][j] * scale2;
    }

    // ---- online softmax over this tile's 64 scores of q-row (qw+ln)
    const int q_abs = qw + ln;
    float rowmax = -INFINITY;
    // interior tiles (every row/col in range, strictly below the causal
    // diagonal) skip the per-element mask math — at S=4096 that is ~94%
    // of tiles, and the mask chain was ~4 VALU ops per score
    const bool full = (k0 + KVB <= SeffK) && (qw + QW <= Seff) && (!causal || k0 + KVB - 1 <= qw);
    if (full) {
#pragma unroll
      for (int kb = 0; kb < KB; ++kb)
#pragma unroll
        for (int j = 0; j < 16; ++j) rowmax = fmaxf(rowmax, p[kb][j]);
    } else {
#pragma unroll
    for (int kb = 0; kb < KB; ++kb)
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int k_abs = k0 + kb * 32 + crow(j, half);
        const bool valid = (q_abs < Seff) && (k_abs < SeffK) && (!causal || k_abs <= q_abs);
        if (!valid) p[kb][j] = -INFINITY;
        rowmax = fmaxf(rowmax, p[kb][j]);
      }
    }
    rowmax = fmaxf(rowmax, __shfl_xor(rowmax, 32));
    // defer-max (guide T13): if this tile's max is within 8 of the running
    // max, keep the old max (P bounded by e^8, fine in fp32/bf16) and skip
    // the O-rescale pass entirely when every row defers.
    const bool defer = (m_run != -INFINITY) && (rowmax - m_run <= 11.5416f);  // 8 nats in log2
    const float m_new = defer ? m_run : fmaxf(m_run, rowmax);
    const float msafe = (m_new == -INFINITY) ? 0.0f : m_new;
    const float corr =
        defer ? 1.0f : ((m_run == -INFINITY) ? ((m_new == -INFINITY) ? 1.0f : 0.0f) : __builtin_amdgcn_exp2f(m_run - m_new));
    float rowsum = 0.0f;
#pragma unroll
    for (int kb = 0; kb < KB; ++kb)
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        p[kb][j] = __builtin_amdgcn_exp2f(p[kb][j] - msafe);
        rowsum += p[kb][j];
      }
    rowsum += __shfl_xor(rowsum, 32);
    m_run = m_new;
    l_run = l_run * corr + rowsum;

    // ---- stage P (bf16) into swizzled per-wave LDS: P[q=ln][k]
    // C-regs j=0..3 within a group are 4 consecutive k values → 8 B packed.
#pragma unroll
    for (int kb = 0; kb < KB; ++kb)
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const int kcol = kb * 32 + 8 * jj + 4 * half;
        short4v pk;
#pragma unroll
        for (int e = 0; e < 4; ++e) pk[e] = (short)f2bf(p[kb][jj * 4 + e]);
        *reinterpret_cast<short4v*>(P + swz<KVB * 2>(ln, kcol * 2)) = pk;
      }

    // ---- rescale O accumulator by corr (per q-row, via shfl broadcast);
    // skipped entirely when every row of the wave deferred (corr == 1)
    if (!__all(corr == 1.0f)) {
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const int r = crow(j, half);
        const float c = __shfl(corr, r);  // corr for q-row r lives on lanes r, r+32
#pragma unroll
        for (int nb = 0; nb < D / 32; ++nb) oacc[nb][j] *= c;
      }
    }

    // ---- O += P · V  (A = P from LDS, B = V column-fragments via tr-read)
    // software-pipelined: the NEXT (pa, vb) pair's LDS reads issue under
    // the current MFMA so the ld->mfma chain never serializes
    {
      constexpr int KS = KVB / 16, NB = D / 32;
      bf16x8_t pa_cur = ld_lds16(P, swz<KVB * 2>(ln, (half * 8) * 2));
      bf16x8_t vb_cur = ld_frag_tr<KVB>(Vlds, lane, 0, 0);
      bf16x8_t pa_nxt, vb_nxt;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          if (nb + 1 < NB) {
            vb_nxt = ld_frag_tr<KVB>(Vlds, lane, ks * 16, (nb + 1) * 2);
          } else if (ks + 1 < KS) {
            pa_nxt = ld_lds16(P, swz<KVB * 2>(ln, ((ks + 1) * 16 + half * 8) * 2));
            vb_nxt = ld_frag_tr<KVB>(Vlds, lane, (ks + 1) * 16, 0);
          }
          __builtin_amdgcn_s_setprio(1);
          oacc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa_cur, vb_cur, oacc[nb], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
          vb_cur = vb_nxt;
          if (nb + 1 == NB) pa_cur = pa_nxt;
        }
      }
    }
    }  // active

    // stores target the OTHER buffer: no barrier needed before them, and
    // the single barrier below covers both "stores visible" and "everyone
    // done with this tile's buffer before it is reused two tiles on"
    if (has_next) {
      char* Kn = smem + ((t + 1) & 1) * 2 * KB_BYTES;
      kstage.store_subtiled(Kn);
      vstage.store_subtiled(Kn + KB_BYTES);
    }
    __syncthreads();
  }

  // ---- epilogue: O /= l, store O and LSE. Rows beyond Seff (right-padded
  // batches) have l_run == 0 -> O = 0, LSE = -inf, which the backward uses
  // to mask them; the buffer extent S is still fully written.
  const int row_lim = VARLEN ? Seff : S;
  if (qw >= row_lim) return;
  unsigned short* o_base = VARLEN ? O + (seq0 * Hq + h) * D : O + ((long)b * S * Hq + h) * D;
  float* lse_base = VARLEN ? LSE + (long)h * S + seq0 : LSE + ((long)b * Hq + h) * S;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int r = crow(j, half);
    const float l_r = __shfl(l_run, r);
    const float inv_l = (l_r > 0.0f) ? 1.0f / l_r : 0.0f;
    const int q_abs = qw + r;
    if (q_abs < row_lim) {
#pragma unroll
      for (int nb = 0; nb < D / 32; ++nb) {
        o_base[(long)q_abs * o_stride + nb * 32 + ln] = f2bf(oacc[nb][j] * inv_l);
      }
    }
  }
  if (lane < QW && qw + lane < row_lim) {
    // back to nats: m_run is log2-domain
    const float lse = (l_run > 0.0f) ? m_run * 0.6931472f + __logf(l_run) : -INFINITY;
    lse_base[qw + lane] = lse;
  }
}

// ==================================================== delta = rowsum(dO ⊙ O)
__global__ __launch_bounds__(256) void fa_delta_kernel(
    const unsigned short* __restrict__ dO,
    const unsigned short* __restrict__ Oin,
    float* __restrict__ delta,  // [B, H, S]
    long rows,  // B*S*H
    int S, int H, int D) {
  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int sub = lane >> 4;       // 4 rows per wave
  const int e16 = lane & 15;       // 16 lanes per row
  const int epl = D / 16;          // elems per lane (8 for D=128)
  const long waves_total = (long)gridDim.x * 4;  // 4 waves per block
  for (long r4 = blockIdx.x * 4L + w; r4 * 4 < rows; r4 += waves_total) {
    const long row = r4 * 4 + sub;
    if (row >= rows) continue;
    float acc = 0.0f;
    const unsigned short* dop = dO + row * D + e16 * epl;
    const unsigned short* op = Oin + row * D + e16 * epl;
    if (epl >= 8) {
      for (int k = 0; k < epl; k += 8) {
        short8 a = *reinterpret_cast<const short8*>(dop + k);
        short8 b = *reinterpret_cast<const short8*>(op + k);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += bf2f((unsigned short)a[j]) * bf2f((unsigned short)b[j]);
      }
    } else {  // D=64: 4 elems per lane — load width must match the slice
      short4v a = *reinterpret_cast<const short4v*>(dop);
      short4v b = *reinterpret_cast<const short4v*>(op);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc += bf2f((unsigned short)a[j]) * bf2f((unsigned short)b[j]);
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) acc += __shfl_xor(acc, off);
    if (e16 == 0) {
      const long bidx = row / ((long)S * H);
      const long rem = row % ((long)S * H);
      const long s = rem / H;
      const long hh = rem % H;
      delta[(bidx * H + hh) * S + s] = acc;
    }
  }
}

// ================================================= backward dK/dV kernel
//
// grid: (ceil(S/128) kv-tiles, B*Hkv). Each wave owns 32 kv rows; loops over
// the GQA group's q-heads and that head's q-tiles (64 q rows staged subtiled
// in LDS). dK/dV accumulate in registers across the whole loop. Score tiles
// are processed in 32-q halves through a per-wave 2 KB P tile.
// VAR experiment ladder (select at launch via CAI_FA_DKDV_VAR):
//  0: async-split RegStage prefetch, K+V fragments resident, no occupancy
//     bound (352 regs -> 1 wave/SIMD)
//  1: same body, forced 2 waves/SIMD (256-reg cap -> ~75 spills)
//  2: register diet for a TRUE 2 waves/SIMD: synchronous staging (no
//     prefetch registers) + V fragments re-loaded from L1 each tile
//  3: VAR 2 + K fragments also re-loaded (zero spill target)
template <int D, int VAR, bool VARLEN = false>
__global__ __launch_bounds__(fa::NTB, VAR >= 1 ? 2 : 1) void fa_bwd_dkdv_kernel(
    const unsigned short* __restrict__ Q,
    const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V,
    const unsigned short* __restrict__ dO,
    const float* __restrict__ LSE,
    const float* __restrict__ DELTA,
    unsigned short* __restrict__ dK,
    unsigned short* __restrict__ dV,
    const int* __restrict__ CU,
    const int* __restrict__ CUK,  // padded ring pieces: kv-side valid counts
    long qbs, long qts, long kbs, long kts, long dkbs, long dkts,
    int B, int S, int Sk, int Hq, int Hkv, float scale, int causal) {
  using namespace fa;
  constexpr int QB = 64;  // q rows per staged tile
  extern __shared__ char smem[];
  constexpr int TB = sub_bytes<QB, D>();
  constexpr int PB = sub_bytes<32, 32>();  // per-wave P tile: [q 32][kv 32]
  char* Qlds = smem;
  char* dOlds = smem + TB;
  char* Pw = smem + 2 * TB;  // + wave*PB
  float* lse_lds = reinterpret_cast<float*>(smem + 2 * TB + NWB * PB);
  float* dta_lds = lse_lds + QB;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int ln = lane & 31;

  int swz_x, swz_y;
  xcd_swizzle_xy(swz_x, swz_y);  // same-bh kv tiles share an XCD's L2 (Q/dO stream reuse)
  const int bh = swz_y;
  const int b = bh / Hkv, hk = bh % Hkv;
  const int G = Hq / Hkv;
  const int kv0 = swz_x * (NWB * QW);
  const int kvw = kv0 + w * QW;  // this wave's first kv row

  long seq0 = 0;
  int Seff = S;
  if (VARLEN) {
    seq0 = CU[b];
    Seff = CU[b + 1] - (int)seq0;
    if (kv0 >= Seff) return;
  } else if (CU != nullptr) {
    Seff = CU[b];
  }
  // this kernel's OWN rows are kv (extent Sk); q streams over Seff
  int SeffK = VARLEN ? Seff
            : (CUK != nullptr ? CUK[b] : (CU != nullptr ? min(CU[b], Sk) : Sk));

  const unsigned short* k_base = VARLEN ? K + seq0 * kts + (long)hk * D : K + (long)b * kbs + (long)hk * D;
  const unsigned short* v_base = VARLEN ? V + seq0 * kts + (long)hk * D : V + (long)b * kbs + (long)hk * D;

  constexpr int DSL = D / 16;
  char* P = Pw + w * PB;

  // ---- K, V fragments (A-operands): lane row = kv (ln), d = sl*16+half*8+[0..7]
  const int kv_my = max(min(kvw + ln, SeffK - 1), 0);
  const unsigned short* vp = v_base + (long)kv_my * kts + half * 8;
  const unsigned short* kp = k_base + (long)kv_my * kts + half * 8;
  bf16x8_t kf[VAR == 3 ? 1 : DSL], vf[VAR >= 2 ? 1 : DSL];
  {
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      if (VAR != 3) kf[sl] = ld_g16(kp + sl * 16);
      if (VAR < 2) vf[sl] = ld_g16(vp + sl * 16);
    }
  }

  f32x16 dk_acc[D / 32], dv_acc[D / 32];
#pragma unroll
  for (int nb = 0; nb < D / 32; ++nb)
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      dk_acc[nb][j] = 0.0f;
      dv_acc[nb][j] = 0.0f;
    }

  const int qt_first = causal ? (kv0 / QB) : 0;
  const int qt_last = (Seff + QB - 1) / QB;

  RegStage<D, QB, NTB> qstage, dostage;
  for (int g = 0; g < G; ++g) {
    const int h = hk * G + g;
    const unsigned short* q_base = VARLEN ? Q + seq0 * qts + (long)h * D : Q + (long)b * qbs + (long)h * D;
    const unsigned short* do_base =
        VARLEN ? dO + (seq0 * Hq + h) * D : dO + ((long)b * S * Hq + h) * D;  // dO packed
    const long do_stride = (long)Hq * D;
    const float* lse_base = VARLEN ? LSE + (long)h * S + seq0 : LSE + ((long)b * Hq + h) * S;
    const float* dta_base = VARLEN ? DELTA + (long)h * S + seq0 : DELTA + ((long)b * Hq + h) * S;

    // prologue: stage the first q-tile of this head synchronously
    if (VAR != 2) {
      qstage.load(q_base, qts, qt_first * QB, Seff);
      dostage.load(do_base, do_stride, qt_first * QB, Seff);
    }
    __syncthreads();  // previous head's compute done before overwriting LDS
    if (VAR != 2) {
      qstage.store_subtiled(Qlds);
      dostage.store_subtiled(dOlds);
    } else {
      stage_direct<D, QB, NTB>(Qlds, q_base, qts, qt_first * QB, Seff);
      stage_direct<D, QB, NTB>(dOlds, do_base, do_stride, qt_first * QB, Seff);
    }
    if (threadIdx.x < QB) {
      const int qr = max(min(qt_first * QB + (int)threadIdx.x, Seff - 1), 0);
      lse_lds[threadIdx.x] = lse_base[qr];
      dta_lds[threadIdx.x] = dta_base[qr];
    }
    __syncthreads();

    for (int qt = qt_first; qt < qt_last; ++qt) {
      const int qt0 = qt * QB;
      const bool has_next = (qt + 1 < qt_last);
      if (VAR != 2 && has_next) {
        qstage.load(q_base, qts, qt0 + QB, Seff);
        dostage.load(do_base, do_stride, qt0 + QB, Seff);
      }
      const bool active = !(causal && qt0 + QB - 1 < kvw) && (kvw < SeffK);
      if (active) {

      // 32-q halves: live score state is p[16]+dp[16] instead of [2][16].
#pragma unroll
      for (int qb = 0; qb < 2; ++qb) {
        // ---- S'^T[kv][q] = K · Q^T : C col = q (ln within half), row = kv.
        f32x16 acc;
#pragma unroll
        for (int j = 0; j < 16; ++j) acc[j] = 0.0f;
        // pipelined: slice sl+1's Q row-slice read issues under slice sl's MFMA
        bf16x8_t qb_cur = ld_lds16(Qlds, sub_off<QB>(qb * 32 + ln, half * 8));
        bf16x8_t qb_nxt;
#pragma unroll
        for (int sl = 0; sl < DSL; ++sl) {
          if (sl + 1 < DSL)
            qb_nxt = ld_lds16(Qlds, sub_off<QB>(qb * 32 + ln, (sl + 1) * 16 + half * 8));
          bf16x8_t kfr = (VAR == 3) ? ld_g16(kp + sl * 16) : kf[VAR == 3 ? 0 : sl];
          __builtin_amdgcn_s_setprio(1);
          acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qb_cur, acc, 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
          qb_cur = qb_nxt;
        }

        // ---- P' = exp(scale*s - lse[q]), masked (interior blocks skip
        // the per-element mask chain — most blocks at training seqlens)
        float p[16];
        {
          const int q_abs = qt0 + qb * 32 + ln;
          const float lse = lse_lds[qb * 32 + ln];
          // exp2 domain: exp(a*s - l) = exp2(fma(a, s*log2e, -l*log2e))
          const float scale2 = scale * 1.44269504f;
          const float lse2 = lse * 1.44269504f;
          const bool blk_full = (qt0 + qb * 32 + 32 <= Seff) && (kvw + QW <= SeffK) &&
                                (!causal || kvw + QW - 1 <= qt0 + qb * 32);
          if (blk_full) {
#pragma unroll
            for (int j = 0; j < 16; ++j)
              p[j] = __builtin_amdgcn_exp2f(fmaf(acc[j], scale2, -lse2));
          } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) {
            const int k_abs = kvw + crow(j, half);
            const bool valid = (q_abs < Seff) && (k_abs < SeffK) && (!causal || k_abs <= q_abs) && (lse != -INFINITY);
            p[j] = valid ? __builtin_amdgcn_exp2f(fmaf(acc[j], scale2, -lse2)) : 0.0f;
          }
          }
        }

        // ---- stage P' into the per-wave subtiled [q 32][kv 32] tile:
        // C-regs j in a 4-group are 4 consecutive kv rows → packed 8 B store
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
          const int kv_loc = 8 * jj + 4 * half;
          short4v pk;
#pragma unroll
          for (int e = 0; e < 4; ++e) pk[e] = (short)f2bf(p[jj * 4 + e]);
          *reinterpret_cast<short4v*>(P + sub_off<32>(ln, kv_loc)) = pk;
        }

        // ---- dV += P' · dO  (A = P' via tr-read: lane row = kv, k = q-local;
        //      B = dO column-fragments, k = q-global)
        {
          constexpr int NB = D / 32;
          bf16x8_t pa_cur = ld_frag_tr<32>(P, lane, 0, 0);
          bf16x8_t dob_cur = ld_frag_tr<QB>(dOlds, lane, qb * 32, 0);
          bf16x8_t pa_nxt, dob_nxt;
#pragma unroll
          for (int ks2 = 0; ks2 < 2; ++ks2) {
#pragma unroll
            for (int nb = 0; nb < NB; ++nb) {
              if (nb + 1 < NB) {
                dob_nxt = ld_frag_tr<QB>(dOlds, lane, qb * 32 + ks2 * 16, (nb + 1) * 2);
              } else if (ks2 == 0) {
                pa_nxt = ld_frag_tr<32>(P, lane, 16, 0);
                dob_nxt = ld_frag_tr<QB>(dOlds, lane, qb * 32 + 16, 0);
              }
              __builtin_amdgcn_s_setprio(1);
              dv_acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa_cur, dob_cur, dv_acc[nb], 0, 0, 0);
              __builtin_amdgcn_s_setprio(0);
              dob_cur = dob_nxt;
              if (nb + 1 == NB) pa_cur = pa_nxt;
            }
          }
        }

        // ---- dP'[kv][q] = V · dO^T  (A = V regs — VAR 2 re-loads each V
        //      fragment from L1; the row is hot — B = dO row-slice reads)
        f32x16 acc2;
#pragma unroll
        for (int j = 0; j < 16; ++j) acc2[j] = 0.0f;
        bf16x8_t do_cur = ld_lds16(dOlds, sub_off<QB>(qb * 32 + ln, half * 8));
        bf16x8_t do_nxt;
        // VAR>=2 reloads V fragments from L1 — pipeline those too, the
        // global-load latency was the serial chain here (PMC: both pipes idle)
        bf16x8_t vf_cur = (VAR >= 2) ? ld_g16(vp) : vf[0];
        bf16x8_t vf_nxt;
#pragma unroll
        for (int sl = 0; sl < DSL; ++sl) {
          if (sl + 1 < DSL) {
            do_nxt = ld_lds16(dOlds, sub_off<QB>(qb * 32 + ln, (sl + 1) * 16 + half * 8));
            if (VAR >= 2) vf_nxt = ld_g16(vp + (sl + 1) * 16);
          }
          bf16x8_t vfr = (VAR >= 2) ? vf_cur : vf[VAR >= 2 ? 0 : sl];
          acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, do_cur, acc2, 0, 0, 0);
          do_cur = do_nxt;
          vf_cur = vf_nxt;
        }

        // ---- dS' = scale * P' ⊙ (dP' - delta[q]) → overwrite P tile
        {
          const float dta = dta_lds[qb * 32 + ln];
#pragma unroll
          for (int jj = 0; jj < 4; ++jj) {
            const int kv_loc = 8 * jj + 4 * half;
            short4v dk4;
#pragma unroll
            for (int e = 0; e < 4; ++e) {
              const float ds = scale * p[jj * 4 + e] * (acc2[jj * 4 + e] - dta);
              dk4[e] = (short)f2bf(ds);
            }
            *reinterpret_cast<short4v*>(P + sub_off<32>(ln, kv_loc)) = dk4;
          }
        }

        // ---- dK += dS' · Q   (A = dS' tr-read, B = Q column-fragments)
        {
          constexpr int NB = D / 32;
          bf16x8_t dsa_cur = ld_frag_tr<32>(P, lane, 0, 0);
          bf16x8_t qtb_cur = ld_frag_tr<QB>(Qlds, lane, qb * 32, 0);
          bf16x8_t dsa_nxt, qtb_nxt;
#pragma unroll
        for (int ks2 = 0; ks2 < 2; ++ks2) {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) {
            if (nb + 1 < NB) {
              qtb_nxt = ld_frag_tr<QB>(Qlds, lane, qb * 32 + ks2 * 16, (nb + 1) * 2);
            } else if (ks2 == 0) {
              dsa_nxt = ld_frag_tr<32>(P, lane, 16, 0);
              qtb_nxt = ld_frag_tr<QB>(Qlds, lane, qb * 32 + 16, 0);
            }
            __builtin_amdgcn_s_setprio(1);
            dk_acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa_cur, qtb_cur, dk_acc[nb], 0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
            qtb_cur = qtb_nxt;
            if (nb + 1 == NB) dsa_cur = dsa_nxt;
          }
        }
        }
      }  // qb halves
      }  // active

      __syncthreads();  // all waves done with this q-tile's LDS
      if (has_next) {
        if (VAR != 2) {
          qstage.store_subtiled(Qlds);
          dostage.store_subtiled(dOlds);
        } else {
          stage_direct<D, QB, NTB>(Qlds, q_base, qts, qt0 + QB, Seff);
          stage_direct<D, QB, NTB>(dOlds, do_base, do_stride, qt0 + QB, Seff);
        }
        if (threadIdx.x < QB) {
          const int qr = max(min(qt0 + QB + (int)threadIdx.x, Seff - 1), 0);
          lse_lds[threadIdx.x] = lse_base[qr];
          dta_lds[threadIdx.x] = dta_base[qr];
        }
      }
      __syncthreads();
    }
  }

  // ---- epilogue: write dK, dV (each kv row owned by exactly one block).
  // Padded rows accumulate nothing and are written as zeros.
  const int kv_lim = VARLEN ? Seff : Sk;
  if (kvw >= kv_lim) return;
  unsigned short* dk_base = VARLEN ? dK + seq0 * dkts + (long)hk * D : dK + (long)b * dkbs + (long)hk * D;
  unsigned short* dv_base = VARLEN ? dV + seq0 * dkts + (long)hk * D : dV + (long)b * dkbs + (long)hk * D;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int r = crow(j, half);
    const int kv_abs = kvw + r;
    if (kv_abs < kv_lim) {
#pragma unroll
      for (int nb = 0; nb < D / 32; ++nb) {
        dk_base[(long)kv_abs * dkts + nb * 32 + ln] = f2bf(dk_acc[nb][j]);
        dv_base[(long)kv_abs * dkts + nb * 32 + ln] = f2bf(dv_acc[nb][j]);
      }
    }
  }
}

// ===================================================== backward dQ kernel
//
// grid: (ceil(S/128) q-tiles, B*Hq). Each wave owns 32 q rows; kv tiles of
// 64 staged subtiled in LDS (K serves both row-slice and column fragments,
// V row-slice only). dS restages through a per-wave [kv 32][q 32] tile.
// __launch_bounds__ min-waves/SIMD = 2 caps the allocation at 256 registers:
// with the tr-read design the kernel fits (vs 441 before), doubling occupancy.
template <int D, bool VARLEN = false>
__global__ __launch_bounds__(fa::NTB, 2) void fa_bwd_dq_kernel(
    const unsigned short* __restrict__ Q,
    const unsigned short* __restrict__ K,
    const unsigned short* __restrict__ V,
    const unsigned short* __restrict__ dO,
    const float* __restrict__ LSE,
    const float* __restrict__ DELTA,
    unsigned short* __restrict__ dQ,
    const int* __restrict__ CU,
    const int* __restrict__ CUK,  // padded ring pieces: kv-side valid counts
    long qbs, long qts, long kbs, long kts, long dqbs, long dqts,
    int B, int S, int Sk, int Hq, int Hkv, float scale, int causal) {
  using namespace fa;
  extern __shared__ char smem[];
  constexpr int KB_BYTES = sub_bytes<KVB, D>();
  constexpr int PB = sub_bytes<32, 32>();  // per-wave dS tile: [kv 32][q 32]
  // single-buffered here: double-buffering (as in the forward) pushes the
  // block past the 80 KB needed for 2 blocks/CU and measured -13%
  char* Klds = smem;
  char* Vlds = smem + KB_BYTES;
  char* Pw = smem + 2 * KB_BYTES;  // + wave*PB
  float* lse_lds = reinterpret_cast<float*>(smem + 2 * KB_BYTES + NWB * PB);
  float* dta_lds = lse_lds + NWB * QW;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int ln = lane & 31;

  int swz_x, swz_y;
  xcd_swizzle_xy(swz_x, swz_y);  // same-bh q tiles share an XCD's L2 (K/V stream reuse)
  const int bh = swz_y;
  const int b = bh / Hq, h = bh % Hq;
  const int hk = h / (Hq / Hkv);
  const int q0 = swz_x * (NWB * QW);
  const int qw = q0 + w * QW;

  long seq0 = 0;
  int Seff = S;
  if (VARLEN) {
    seq0 = CU[b];
    Seff = CU[b + 1] - (int)seq0;
    if (q0 >= Seff) return;
  } else if (CU != nullptr) {
    Seff = CU[b];
  }
  const int SeffK = VARLEN ? Seff
            : (CUK != nullptr ? CUK[b] : (CU != nullptr ? min(CU[b], Sk) : Sk));

  const unsigned short* q_base = VARLEN ? Q + seq0 * qts + (long)h * D : Q + (long)b * qbs + (long)h * D;
  const unsigned short* do_base =
      VARLEN ? dO + (seq0 * Hq + h) * D : dO + ((long)b * S * Hq + h) * D;  // dO packed
  const long do_stride = (long)Hq * D;
  const unsigned short* k_base = VARLEN ? K + seq0 * kts + (long)hk * D : K + (long)b * kbs + (long)hk * D;
  const unsigned short* v_base = VARLEN ? V + seq0 * kts + (long)hk * D : V + (long)b * kbs + (long)hk * D;

  constexpr int DSL = D / 16;
  char* P = Pw + w * PB;

  // lse/delta for the block's q rows
  {
    const float* lse_base = VARLEN ? LSE + (long)h * S + seq0 : LSE + ((long)b * Hq + h) * S;
    const float* dta_base = VARLEN ? DELTA + (long)h * S + seq0 : DELTA + ((long)b * Hq + h) * S;
    for (int i = threadIdx.x; i < NWB * QW; i += NTB) {
      const int qr = max(min(q0 + i, Seff - 1), 0);
      lse_lds[i] = lse_base[qr];
      dta_lds[i] = dta_base[qr];
    }
  }

  // ---- Q, dO fragments (A-operands): lane row = q (ln), d cols
  const int q_my = max(min(qw + ln, Seff - 1), 0);
  bf16x8_t qa[DSL], doa[DSL];
  {
    const unsigned short* qp = q_base + (long)q_my * qts + half * 8;
    const unsigned short* dop = do_base + (long)q_my * do_stride + half * 8;
#pragma unroll
    for (int sl = 0; sl < DSL; ++sl) {
      qa[sl] = ld_g16(qp + sl * 16);
      doa[sl] = ld_g16(dop + sl * 16);
    }
  }

  f32x16 dq_acc[D / 32];
#pragma unroll
  for (int nb = 0; nb < D / 32; ++nb)
#pragma unroll
    for (int j = 0; j < 16; ++j) dq_acc[nb][j] = 0.0f;

  const int kv_tiles_all = (SeffK + KVB - 1) / KVB;
  const int kv_tiles = causal ? min(kv_tiles_all, (q0 + NWB * QW + KVB - 1) / KVB) : kv_tiles_all;

  RegStage<D, KVB, NTB> kstage, vstage;
  // SeffK == 0 blocks skip the tile loop; the clamped loads stay in-bounds
  kstage.load(k_base, kts, 0, SeffK);
  vstage.load(v_base, kts, 0, SeffK);
  kstage.store_subtiled(Klds);
  vstage.store_subtiled(Vlds);
  __syncthreads();

  for (int t = 0; t < kv_tiles; ++t) {
    const int k0 = t * KVB;
    const bool has_next = (t + 1 < kv_tiles);
    if (has_next) {
      kstage.load(k_base, kts, k0 + KVB, SeffK);
      vstage.load(v_base, kts, k0 + KVB, SeffK);
    }
    const bool active = !(causal && k0 > qw + QW - 1) && (qw < Seff);
    if (active) {

    // 32-kv halves: live score state is p[16]+dp[16] instead of [2][16].
#pragma unroll
    for (int kb = 0; kb < 2; ++kb) {
      // ---- S[q][kv] = Q · K^T and dP[q][kv] = dO · V^T (shared loop)
      f32x16 acc, acc2;
#pragma unroll
      for (int j = 0; j < 16; ++j) { acc[j] = 0.0f; acc2[j] = 0.0f; }
      // software-pipelined: slice sl+1's K/V row-slice reads issue under
      // slice sl's MFMAs (same restructure as the forward, +10% there)
      bf16x8_t kb_cur = ld_lds16(Klds, sub_off<KVB>(kb * 32 + ln, half * 8));
      bf16x8_t vb_cur = ld_lds16(Vlds, sub_off<KVB>(kb * 32 + ln, half * 8));
      bf16x8_t kb_nxt, vb_nxt;
#pragma unroll
      for (int sl = 0; sl < DSL; ++sl) {
        if (sl + 1 < DSL) {
          kb_nxt = ld_lds16(Klds, sub_off<KVB>(kb * 32 + ln, (sl + 1) * 16 + half * 8));
          vb_nxt = ld_lds16(Vlds, sub_off<KVB>(kb * 32 + ln, (sl + 1) * 16 + half * 8));
        }
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa[sl], kb_cur, acc, 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(doa[sl], vb_cur, acc2, 0, 0, 0);
        kb_cur = kb_nxt;
        vb_cur = vb_nxt;
      }

      // ---- dS = scale * P ⊙ (dP - delta[q]) with P = exp(scale*s - lse[q]);
      // C-regs j in a 4-group are 4 consecutive q rows → packed 8 B store
      // into the per-wave [kv 32][q 32] subtiled tile.
      // interior kb-halves (all q/k in range, below the diagonal) take a
      // maskless dS loop — branching at loop level keeps each path's
      // register footprint tight
      const bool blk_full = (k0 + kb * 32 + 32 <= SeffK) && (qw + QW <= Seff) &&
                            (!causal || k0 + kb * 32 + 31 <= qw);
      if (blk_full) {
#pragma unroll
        for (int jj = 0; jj < 4; ++jj) {
          const int q_loc0 = 8 * jj + 4 * half;
          short4v dk4;
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const int r = q_loc0 + e;
            const float pv = __builtin_amdgcn_exp2f(
                fmaf(acc[jj * 4 + e], scale * 1.44269504f, -lse_lds[w * QW + r] * 1.44269504f));
            const float ds = scale * pv * (acc2[jj * 4 + e] - dta_lds[w * QW + r]);
            dk4[e] = (short)f2bf(ds);
          }
          *reinterpret_cast<short4v*>(P + sub_off<32>(ln, q_loc0)) = dk4;
        }
      } else {
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const int q_loc0 = 8 * jj + 4 * half;
        short4v dk4;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const int r = q_loc0 + e;
          const int q_abs = qw + r;
          const int k_abs = k0 + kb * 32 + ln;
          const float lse = lse_lds[w * QW + r];
          const float dta = dta_lds[w * QW + r];
          const bool valid = (q_abs < Seff) && (k_abs < SeffK) && (!causal || k_abs <= q_abs) && (lse != -INFINITY);
          const float pv = valid
              ? __builtin_amdgcn_exp2f(fmaf(acc[jj * 4 + e], scale * 1.44269504f, -lse * 1.44269504f))
              : 0.0f;
          const float ds = scale * pv * (acc2[jj * 4 + e] - dta);
          dk4[e] = (short)f2bf(ds);
        }
        *reinterpret_cast<short4v*>(P + sub_off<32>(ln, q_loc0)) = dk4;
      }
      }

      // ---- dQ += dS · K   (A = dS tr-read: lane row = q, k = kv-local;
      //      B = K column-fragments, k = kv-global)
      {
        constexpr int NB = D / 32;
        bf16x8_t dsa_cur = ld_frag_tr<32>(P, lane, 0, 0);
        bf16x8_t ktb_cur = ld_frag_tr<KVB>(Klds, lane, kb * 32, 0);
        bf16x8_t dsa_nxt, ktb_nxt;
#pragma unroll
        for (int ks2 = 0; ks2 < 2; ++ks2) {
#pragma unroll
          for (int nb = 0; nb < NB; ++nb) {
            if (nb + 1 < NB) {
              ktb_nxt = ld_frag_tr<KVB>(Klds, lane, kb * 32 + ks2 * 16, (nb + 1) * 2);
            } else if (ks2 == 0) {
              dsa_nxt = ld_frag_tr<32>(P, lane, 16, 0);
              ktb_nxt = ld_frag_tr<KVB>(Klds, lane, kb * 32 + 16, 0);
            }
            __builtin_amdgcn_s_setprio(1);
            dq_acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsa_cur, ktb_cur, dq_acc[nb], 0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
            ktb_cur = ktb_nxt;
            if (nb + 1 == NB) dsa_cur = dsa_nxt;
          }
        }
      }
    }  // kb halves
    }  // active

    __syncthreads();
    if (has_next) {
      kstage.store_subtiled(Klds);
      vstage.store_subtiled(Vlds);
    }
    __syncthreads();
  }

  // ---- epilogue: write dQ (pad rows accumulate nothing -> zeros)
  const int q_lim = VARLEN ? Seff : S;
  if (qw >= q_lim) return;
  unsigned short* dq_base = VARLEN ? dQ + seq0 * dqts + (long)h * D : dQ + (long)b * dqbs + (long)h * D;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int r = crow(j, half);
    const int q_abs = qw + r;
    if (q_abs < q_lim) {
#pragma unroll
      for (int nb = 0; nb < D / 32; ++nb) {
        dq_base[(long)q_abs * dqts + nb * 32 + ln] = f2bf(dq_acc[nb][j]);
      }
    }
  }
}

// =============================================================== host side

namespace {

void set_lds_limit(const void* kernel, size_t bytes) {
  if (bytes > 65536) {
    (void)hipFuncSetAttribute(kernel, hipFuncAttributeMaxDynamicSharedMemorySize, (int)bytes);
  }
}

// q/k/v are [B,S,H,D] views that may be strided in dims 0/1 (e.g. views into
// a packed QKV GEMM output); dims 2/3 must be dense.
void check_fa_view(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.dim() == 4, "flash_attn: ", name, " must be [B,S,H,D]");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, "flash_attn: bf16 only");
  TORCH_CHECK(t.stride(3) == 1 && t.stride(2) == t.size(3), "flash_attn: ", name,
              " must be dense in [H,D] (strided B/S allowed)");
}

void check_fa_inputs(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v) {
  check_fa_view(q, "q");
  check_fa_view(k, "k");
  check_fa_view(v, "v");
  const int D = (int)q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn: head dim must be 64 or 128");
  TORCH_CHECK(k.size(3) == D && v.size(3) == D);
  TORCH_CHECK(k.sizes() == v.sizes() && k.strides() == v.strides(), "flash_attn: k/v must match");
  TORCH_CHECK((int)q.size(2) % (int)k.size(2) == 0, "flash_attn: Hq must be a multiple of Hkv (GQA)");
}

}  // namespace

std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, bool causal, double scale,
                                       c10::optional<at::Tensor> seqlens_opt,
                                       c10::optional<at::Tensor> seqlens_k_opt) {
  at::Tensor seqlens = seqlens_opt.has_value() ? *seqlens_opt : at::Tensor();
  at::Tensor seqlens_k = seqlens_k_opt.has_value() ? *seqlens_k_opt : at::Tensor();
  using namespace fa;
  check_fa_inputs(q, k, v);
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2), D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  const int Sk = (int)k.size(1);  // kv may be a shorter ring piece
  const int* cu = nullptr;
  const int* cuk = nullptr;
  if (seqlens.defined() && seqlens.numel() > 0) {
    TORCH_CHECK(seqlens.scalar_type() == at::kInt && seqlens.numel() == B && seqlens.is_cuda(),
                "flash_attn: seqlens must be int32 [B] on device");
    cu = seqlens.data_ptr<int>();
  }
  if (seqlens_k.defined() && seqlens_k.numel() > 0) {
    TORCH_CHECK(cu != nullptr, "flash_attn: seqlens_k requires seqlens");
    TORCH_CHECK(seqlens_k.scalar_type() == at::kInt && seqlens_k.numel() == B && seqlens_k.is_cuda());
    cuk = seqlens_k.data_ptr<int>();
  }
  auto o = at::empty({B, S, Hq, D}, q.options());
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid((S + NW * QW - 1) / (NW * QW), B * Hq);
  const long qbs = q.stride(0), qts = q.stride(1), kbs = k.stride(0), kts = k.stride(1);

#define LAUNCH_FWD(DD)                                                                            \
  do {                                                                                            \
    const size_t lds = 4 * sub_bytes<KVB, DD>() + NW * (QW * KVB * 2);                            \
    set_lds_limit((const void*)fa_fwd_kernel<DD, false>, lds);                                    \
    hipLaunchKernelGGL((fa_fwd_kernel<DD, false>), grid, dim3(NT), lds, stream.stream(),          \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),  \
                       (const unsigned short*)v.data_ptr(), (unsigned short*)o.data_ptr(),        \
                       lse.data_ptr<float>(), cu, cuk, qbs, qts, kbs, kts,                        \
                       B, S, Sk, Hq, Hkv, (float)scale, causal ? 1 : 0);                              \
  } while (0)

  if (D == 128) LAUNCH_FWD(128);
  else LAUNCH_FWD(64);
#undef LAUNCH_FWD
  HIP_CHECK_LAST();
  return {o, lse};
}

// Packed ragged batch: q/k/v [total, H, D]; cu_seqlens int32 [n_seq+1].
// lse is [Hq, total]. (reference varlen surface: flash-attn 2's
// _flash_attn_varlen_forward as consumed by colossalai .../attn.py:139)
std::vector<at::Tensor> flash_attn_varlen_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                              at::Tensor cu_seqlens, long max_seqlen,
                                              bool causal, double scale) {
  using namespace fa;
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3, "varlen: q/k/v must be [total,H,D]");
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt && cu_seqlens.is_cuda() && cu_seqlens.dim() == 1);
  const int T = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
  const int Hkv = (int)k.size(1);
  const int n_seq = (int)cu_seqlens.numel() - 1;
  TORCH_CHECK(D == 64 || D == 128, "flash_attn: head dim must be 64 or 128");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == D, "varlen: q must be dense in [H,D]");
  TORCH_CHECK(k.strides() == v.strides() && k.sizes() == v.sizes());
  auto o = at::empty({T, Hq, D}, q.options());
  auto lse = at::empty({Hq, T}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  const dim3 grid(((int)max_seqlen + NW * QW - 1) / (NW * QW), n_seq * Hq);
  const long qts = q.stride(0), kts = k.stride(0);

#define LAUNCH_VFWD(DD)                                                                           \
  do {                                                                                            \
    const size_t lds = 4 * sub_bytes<KVB, DD>() + NW * (QW * KVB * 2);                            \
    set_lds_limit((const void*)fa_fwd_kernel<DD, true>, lds);                                     \
    hipLaunchKernelGGL((fa_fwd_kernel<DD, true>), grid, dim3(NT), lds, stream.stream(),           \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),  \
                       (const unsigned short*)v.data_ptr(), (unsigned short*)o.data_ptr(),        \
                       lse.data_ptr<float>(), cu_seqlens.data_ptr<int>(), nullptr,                \
                       0L, qts, 0L, kts,                                                          \
                       n_seq, T, T, Hq, Hkv, (float)scale, causal ? 1 : 0);                          \
  } while (0)

  if (D == 128) LAUNCH_VFWD(128);
  else LAUNCH_VFWD(64);
#undef LAUNCH_VFWD
  HIP_CHECK_LAST();
  return {o, lse};
}

namespace {

hipStream_t bwd_side_stream(hipEvent_t* fork, hipEvent_t* join) {
  static hipStream_t side_stream = nullptr;
  static hipEvent_t ev_fork = nullptr, ev_join = nullptr;
  if (side_stream == nullptr) {
    (void)hipStreamCreateWithFlags(&side_stream, hipStreamNonBlocking);
    (void)hipEventCreateWithFlags(&ev_fork, hipEventDisableTiming);
    (void)hipEventCreateWithFlags(&ev_join, hipEventDisableTiming);
  }
  *fork = ev_fork;
  *join = ev_join;
  return side_stream;
}

int dkdv_variant() {
  // dkdv occupancy-variant experiment switch (docs/KERNELS.md)
  static int dkdv_var = -1;
  if (dkdv_var < 0) {
    const char* e = getenv("CAI_FA_DKDV_VAR");
    dkdv_var = e ? atoi(e) : 2;
  }
  return dkdv_var;
}

}  // namespace

// dq/dk/dv may be caller-provided strided views (e.g. into a packed dQKV
// buffer) to avoid post-hoc gathers; pass empty tensors to allocate.
std::vector<at::Tensor> flash_attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                       at::Tensor out, at::Tensor lse, bool causal, double scale,
                                       at::Tensor dq, at::Tensor dk, at::Tensor dv,
                                       c10::optional<at::Tensor> seqlens_opt,
                                       c10::optional<at::Tensor> seqlens_k_opt) {
  at::Tensor seqlens = seqlens_opt.has_value() ? *seqlens_opt : at::Tensor();
  at::Tensor seqlens_k = seqlens_k_opt.has_value() ? *seqlens_k_opt : at::Tensor();
  using namespace fa;
  check_fa_inputs(q, k, v);
  TORCH_CHECK(dout.is_contiguous() && out.is_contiguous() && lse.is_contiguous(),
              "flash_attn_bwd: dout/out/lse must be contiguous");
  const int B = (int)q.size(0), S = (int)q.size(1), Hq = (int)q.size(2), D = (int)q.size(3);
  const int Hkv = (int)k.size(2);
  const int* cu = nullptr;
  const int* cuk = nullptr;
  if (seqlens.defined() && seqlens.numel() > 0) {
    TORCH_CHECK(seqlens.scalar_type() == at::kInt && seqlens.numel() == B && seqlens.is_cuda(),
                "flash_attn: seqlens must be int32 [B] on device");
    cu = seqlens.data_ptr<int>();
  }
  if (seqlens_k.defined() && seqlens_k.numel() > 0) {
    TORCH_CHECK(cu != nullptr, "flash_attn: seqlens_k requires seqlens");
    TORCH_CHECK(seqlens_k.scalar_type() == at::kInt && seqlens_k.numel() == B && seqlens_k.is_cuda());
    cuk = seqlens_k.data_ptr<int>();
  }
  const int Sk = (int)k.size(1);  // kv may be a shorter ring piece
  if (dq.numel() == 0) dq = at::empty({B, S, Hq, D}, q.options());
  if (dk.numel() == 0) dk = at::empty({B, Sk, Hkv, D}, k.options());
  if (dv.numel() == 0) dv = at::empty({B, Sk, Hkv, D}, v.options());
  check_fa_view(dq, "dq");
  check_fa_view(dk, "dk");
  TORCH_CHECK(dk.strides() == dv.strides(), "flash_attn_bwd: dk/dv strides must match");
  auto delta = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();

  {  // delta preprocess (pad rows of O are zero -> delta 0)
    const long rows = (long)B * S * Hq;
    const int grid = capped_grid((rows + 15) / 16, 4096);
    hipLaunchKernelGGL(fa_delta_kernel, dim3(grid), dim3(256), 0, stream.stream(),
                       (const unsigned short*)dout.data_ptr(), (const unsigned short*)out.data_ptr(),
                       delta.data_ptr<float>(), rows, S, Hq, D);
    HIP_CHECK_LAST();
  }

  const dim3 grid_kv((Sk + NWB * QW - 1) / (NWB * QW), B * Hkv);
  const dim3 grid_q((S + NWB * QW - 1) / (NWB * QW), B * Hq);
  const long qbs = q.stride(0), qts = q.stride(1), kbs = k.stride(0), kts = k.stride(1);
  const long dqbs = dq.stride(0), dqts = dq.stride(1), dkbs = dk.stride(0), dkts = dk.stride(1);

  // dK/dV and dQ are independent: run them on two streams so the two
  // latency-bound kernels co-occupy the CUs.
  hipEvent_t ev_fork, ev_join;
  hipStream_t side_stream = bwd_side_stream(&ev_fork, &ev_join);
  (void)hipEventRecord(ev_fork, stream.stream());
  (void)hipStreamWaitEvent(side_stream, ev_fork, 0);
  const int dkdv_var = dkdv_variant();

#define LAUNCH_BWD(DD)                                                                              \
  do {                                                                                              \
    constexpr int QB = 64;                                                                          \
    const size_t lds_kv = 2 * sub_bytes<QB, DD>() + NWB * sub_bytes<32, 32>()                       \
                          + 2 * QB * sizeof(float);                                                 \
    const void* dkdv_fn = dkdv_var == 0 ? (const void*)fa_bwd_dkdv_kernel<DD, 0>                    \
                        : dkdv_var == 1 ? (const void*)fa_bwd_dkdv_kernel<DD, 1>                    \
                        : dkdv_var == 3 ? (const void*)fa_bwd_dkdv_kernel<DD, 3>                    \
                                        : (const void*)fa_bwd_dkdv_kernel<DD, 2>;                   \
    set_lds_limit(dkdv_fn, lds_kv);                                                                 \
    hipLaunchKernelGGL((dkdv_var == 0 ? fa_bwd_dkdv_kernel<DD, 0>                                   \
                        : dkdv_var == 1 ? fa_bwd_dkdv_kernel<DD, 1>                                 \
                        : dkdv_var == 3 ? fa_bwd_dkdv_kernel<DD, 3>                                 \
                                        : fa_bwd_dkdv_kernel<DD, 2>),                               \
                       grid_kv, dim3(NTB), lds_kv, stream.stream(),                                 \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),    \
                       (const unsigned short*)v.data_ptr(), (const unsigned short*)dout.data_ptr(), \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),                              \
                       (unsigned short*)dk.data_ptr(), (unsigned short*)dv.data_ptr(),              \
                       cu, cuk, qbs, qts, kbs, kts, dkbs, dkts,                                    \
                       B, S, Sk, Hq, Hkv, (float)scale, causal ? 1 : 0);                            \
    HIP_CHECK_LAST();                                                                               \
    const size_t lds_q = 2 * sub_bytes<KVB, DD>() + NWB * sub_bytes<32, 32>()                       \
                         + 2 * NWB * QW * sizeof(float);                                            \
    set_lds_limit((const void*)fa_bwd_dq_kernel<DD>, lds_q);                                        \
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DD>), grid_q, dim3(NTB), lds_q, side_stream,               \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),    \
                       (const unsigned short*)v.data_ptr(), (const unsigned short*)dout.data_ptr(), \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),                              \
                       (unsigned short*)dq.data_ptr(),                                              \
                       cu, cuk, qbs, qts, kbs, kts, dqbs, dqts,                                    \
                       B, S, Sk, Hq, Hkv, (float)scale, causal ? 1 : 0);                            \
    HIP_CHECK_LAST();                                                                               \
  } while (0)

  if (D == 128) LAUNCH_BWD(128);
  else LAUNCH_BWD(64);
#undef LAUNCH_BWD
  (void)hipEventRecord(ev_join, side_stream);
  (void)hipStreamWaitEvent(stream.stream(), ev_join, 0);
  return {dq, dk, dv};
}

std::vector<at::Tensor> flash_attn_varlen_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                              at::Tensor out, at::Tensor lse, at::Tensor cu_seqlens,
                                              long max_seqlen, bool causal, double scale) {
  using namespace fa;
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3 && v.dim() == 3, "varlen: q/k/v must be [total,H,D]");
  TORCH_CHECK(dout.is_contiguous() && out.is_contiguous() && lse.is_contiguous());
  const int T = (int)q.size(0), Hq = (int)q.size(1), D = (int)q.size(2);
  const int Hkv = (int)k.size(1);
  const int n_seq = (int)cu_seqlens.numel() - 1;
  auto dq = at::empty({T, Hq, D}, q.options());
  auto dk = at::empty({T, Hkv, D}, k.options());
  auto dv = at::empty({T, Hkv, D}, v.options());
  auto delta = at::empty({Hq, T}, q.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();

  {  // delta preprocess over packed rows ([1, Hq, T] layout == [Hq, T])
    const long rows = (long)T * Hq;
    const int grid = capped_grid((rows + 15) / 16, 4096);
    hipLaunchKernelGGL(fa_delta_kernel, dim3(grid), dim3(256), 0, stream.stream(),
                       (const unsigned short*)dout.data_ptr(), (const unsigned short*)out.data_ptr(),
                       delta.data_ptr<float>(), rows, T, Hq, D);
    HIP_CHECK_LAST();
  }

  const dim3 grid_kv(((int)max_seqlen + NWB * QW - 1) / (NWB * QW), n_seq * Hkv);
  const dim3 grid_q(((int)max_seqlen + NWB * QW - 1) / (NWB * QW), n_seq * Hq);
  const long qts = q.stride(0), kts = k.stride(0);
  const long dqts = dq.stride(0), dkts = dk.stride(0);
  const int* cu = cu_seqlens.data_ptr<int>();

  hipEvent_t ev_fork, ev_join;
  hipStream_t side_stream = bwd_side_stream(&ev_fork, &ev_join);
  (void)hipEventRecord(ev_fork, stream.stream());
  (void)hipStreamWaitEvent(side_stream, ev_fork, 0);

#define LAUNCH_VBWD(DD)                                                                             \
  do {                                                                                              \
    constexpr int QB = 64;                                                                          \
    const size_t lds_kv = 2 * sub_bytes<QB, DD>() + NWB * sub_bytes<32, 32>()                       \
                          + 2 * QB * sizeof(float);                                                 \
    set_lds_limit((const void*)fa_bwd_dkdv_kernel<DD, 2, true>, lds_kv);                            \
    hipLaunchKernelGGL((fa_bwd_dkdv_kernel<DD, 2, true>), grid_kv, dim3(NTB), lds_kv,               \
                       stream.stream(),                                                             \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),    \
                       (const unsigned short*)v.data_ptr(), (const unsigned short*)dout.data_ptr(), \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),                              \
                       (unsigned short*)dk.data_ptr(), (unsigned short*)dv.data_ptr(),              \
                       cu, nullptr, 0L, qts, 0L, kts, 0L, dkts,                                              \
                       n_seq, T, T, Hq, Hkv, (float)scale, causal ? 1 : 0);                            \
    HIP_CHECK_LAST();                                                                               \
    const size_t lds_q = 2 * sub_bytes<KVB, DD>() + NWB * sub_bytes<32, 32>()                       \
                         + 2 * NWB * QW * sizeof(float);                                            \
    set_lds_limit((const void*)fa_bwd_dq_kernel<DD, true>, lds_q);                                  \
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DD, true>), grid_q, dim3(NTB), lds_q, side_stream,         \
                       (const unsigned short*)q.data_ptr(), (const unsigned short*)k.data_ptr(),    \
                       (const unsigned short*)v.data_ptr(), (const unsigned short*)dout.data_ptr(), \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),                              \
                       (unsigned short*)dq.data_ptr(),                                              \
                       cu, nullptr, 0L, qts, 0L, kts, 0L, dqts,                                              \
                       n_seq, T, T, Hq, Hkv, (float)scale, causal ? 1 : 0);                            \
    HIP_CHECK_LAST();                                                                               \
  } while (0)

  if (D == 128) LAUNCH_VBWD(128);
  else LAUNCH_VBWD(64);
#undef LAUNCH_VBWD
  (void)hipEventRecord(ev_join, side_stream);
  (void)hipStreamWaitEvent(stream.stream(), ev_join, 0);
  return {dq, dk, dv};
}

}  // namespace cai
