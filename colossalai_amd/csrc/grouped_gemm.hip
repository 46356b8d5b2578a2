// Grouped (per-expert segment) GEMM for gfx950 — the MoE expert compute.
//
// Tokens are pre-sorted by destination expert into contiguous row groups
// (offsets host-known from the routing bincount); each expert has its own
// weight slice of a packed [E, M, K] tensor. One launch covers all groups:
// blocks map to (group, token-tile) via a prefix scan over ceil(count/128)
// done in-kernel (E is small).
//
//   fwd   : y[n, m]  = sum_k  x[n, k]  * w[g(n), m, k]     (x @ w^T)
//   dgrad : dx[n, k] = sum_m  dy[n, m] * w[g(n), m, k]     (dy @ w)
//   wgrad : dw[g, m, k] = sum_{n in g} dy[n, m] * x[n, k]  (dy^T @ x)
//
// Design (CDNA4 guide §5 canonical GEMM + §5.5 T2/T10/T14):
//  - 256-thread blocks, 128x128 output tiles, BK=64 contraction steps,
//    mfma_f32_32x32x16_bf16; each of 4 waves owns 32 output rows.
//  - both operand tiles staged in LDS in the shared subtiled layout
//    (common.h): contraction-contiguous operands use 16 B row-slice reads,
//    contraction-strided operands (dgrad's w, wgrad's both) use
//    ds_read_b64_tr_b16 column fragments — no transposed staging anywhere.
//  - async-split staging (T14): next K-step's global loads issue before the
//    current step's MFMAs.
//  - wgrad zero-pads the ragged contraction tail (clamping would add
//    spurious rows).
//
// Reference behavioral equivalent: the per-expert GEMM loop the reference
// runs via torch (colossalai/shardformer/modeling/mixtral.py experts) and
// the grouped-GEMM requirement in BASELINE.json's kernel list; the
// reference's own MoE kernels are dispatch/combine only
// (extensions/csrc/kernel/cuda/moe_kernel.cu:276-367).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace cai {

namespace gg {

constexpr int TM = 128;  // token-tile rows
constexpr int TN = 128;  // output cols per block
constexpr int BK = 64;   // contraction step
constexpr int NT = 256;  // threads per block

// Cooperative staging of a [ROWS][COLS] bf16 tile into the subtiled layout.
// Global source is row-major with row stride `ld` elements. Rows are
// clamped to [0, row_hi) (harmless for output-masked rows) unless
// ZEROPAD, which writes zeros instead (for ragged contraction tails).
template <int ROWS, int COLS, bool ZEROPAD>
struct GGStage {
  static constexpr int CPR = COLS / 8;         // 16 B chunks per row
  static constexpr int NCH = ROWS * CPR / NT;  // chunks per thread
  bf16x8_v r[NCH];

  DEV_INLINE void load(const unsigned short* base, long ld, int row0, int row_hi, int col0) {
#pragma unroll
    for (int i = 0; i < NCH; ++i) {
      const int c = (int)threadIdx.x + i * NT;
      const int row = row0 + c / CPR, ch = c % CPR;
      if (ZEROPAD && row >= row_hi) {
        bf16x8_v z = {};
        r[i] = z;
      } else {
        const int rr = ZEROPAD ? row : min(row, row_hi - 1);
        r[i] = ld_g16b(base + (long)rr * ld + col0 + ch * 8);
      }
    }
  }
  DEV_INLINE void store(char* lds) const {
#pragma unroll
    for (int i = 0; i < NCH; ++i) {
      const int c = (int)threadIdx.x + i * NT;
      st_lds16b(lds, sub_off<ROWS>(c / CPR, (c % CPR) * 8), r[i]);
    }
  }
};

// block -> (group, token-tile row0) from host-computed tile ids
struct TileMap {
  const int* __restrict__ tile_g;     // [n_tiles] group of tile
  const int* __restrict__ tile_row0;  // [n_tiles] first token row of tile
};

}  // namespace gg

// ======================================================= fwd / dgrad kernel
//
// TRANS_W = false: fwd  (contraction over w's last dim, row-slice B-frags)
// TRANS_W = true : dgrad (contraction over w's M dim, tr-read B-frags)
// grid: (n_token_tiles, out_cols / TN)
template <bool TRANS_W>
__global__ __launch_bounds__(gg::NT) void grouped_gemm_kernel(
    const unsigned short* __restrict__ X,   // [N, Kin]
    const unsigned short* __restrict__ W,   // [E, M, K]
    unsigned short* __restrict__ Y,         // [N, Kout]
    const int* __restrict__ tile_g,
    const int* __restrict__ tile_row0,
    const int* __restrict__ offs,           // [E+1] group row offsets
    int Kin, int Kout, long w_group_stride) {
  using namespace gg;
  extern __shared__ char smem[];
  // fwd:   A [TM][BK] (x), B [TN][BK] (w rows m, cols k)
  // dgrad: A [TM][BK] (dy), B [BK][TN] (w rows m-contraction, cols k-out)
  constexpr int A_BYTES = sub_bytes<TM, BK>();
  char* Alds = smem;
  char* Blds = smem + A_BYTES;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int ln = lane & 31;

  int swz_x, swz_y;
  xcd_swizzle_xy(swz_x, swz_y);  // neighbor tiles share w[g] panels in the XCD L2
  const int g = tile_g[swz_x];
  const int row0 = tile_row0[swz_x];
  const int grp_lo = offs[g], grp_hi = offs[g + 1];
  const int n0 = swz_y * TN;  // first output col

  const unsigned short* x_base = X;
  const unsigned short* w_base = W + (long)g * w_group_stride;

  f32x16 acc[TN / 32];
#pragma unroll
  for (int nb = 0; nb < TN / 32; ++nb)
#pragma unroll
    for (int j = 0; j < 16; ++j) acc[nb][j] = 0.0f;

  const int ksteps = Kin / BK;  // Kin = contraction extent

  GGStage<TM, BK, false> astage;        // token rows (output-masked)
  GGStage<TRANS_W ? BK : TN, TRANS_W ? TN : BK, false> bstage;

  // prologue
  astage.load(x_base, Kin, grp_lo + row0, grp_hi, 0);
  if (TRANS_W) {
    // B tile = w[k0 : k0+BK (contraction m rows)][n0 : n0+TN (out cols)]
    bstage.load(w_base, Kout, 0, 1 << 30, n0);
  } else {
    // B tile = w[n0 : n0+TN (out rows m)][k0 : k0+BK]
    bstage.load(w_base, Kin, n0, 1 << 30, 0);
  }
  astage.store(Alds);
  bstage.store(Blds);
  __syncthreads();

  for (int t = 0; t < ksteps; ++t) {
    const int k0 = t * BK;
    const bool has_next = (t + 1 < ksteps);
    if (has_next) {
      astage.load(x_base, Kin, grp_lo + row0, grp_hi, k0 + BK);
      if (TRANS_W) bstage.load(w_base, Kout, k0 + BK, 1 << 30, n0);
      else bstage.load(w_base, Kin, n0, 1 << 30, k0 + BK);
    }

    // software-pipelined (as in flash_attn.hip): the next (af, bf) LDS
    // reads issue under the current MFMA
    {
      constexpr int KS = BK / 16, NB = TN / 32;
      auto ldb = [&](int ks, int nb) {
        return TRANS_W ? ld_frag_tr<BK>(Blds, lane, ks * 16, nb * 2)
                       : ld_lds16b(Blds, sub_off<TN>(nb * 32 + ln, ks * 16 + half * 8));
      };
      bf16x8_v af_cur = ld_lds16b(Alds, sub_off<TM>(w * 32 + ln, half * 8));
      bf16x8_v bf_cur = ldb(0, 0);
      bf16x8_v af_nxt, bf_nxt;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          if (nb + 1 < NB) {
            bf_nxt = ldb(ks, nb + 1);
          } else if (ks + 1 < KS) {
            af_nxt = ld_lds16b(Alds, sub_off<TM>(w * 32 + ln, (ks + 1) * 16 + half * 8));
            bf_nxt = ldb(ks + 1, 0);
          }
          __builtin_amdgcn_s_setprio(1);
          acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af_cur, bf_cur, acc[nb], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
          bf_cur = bf_nxt;
          if (nb + 1 == NB) af_cur = af_nxt;
        }
      }
    }

    __syncthreads();
    if (has_next) {
      astage.store(Alds);
      bstage.store(Blds);
    }
    __syncthreads();
  }

  // epilogue: C row = token (crow), col = n0 + nb*32 + ln
  const int my_row0 = grp_lo + row0 + w * 32;
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int r = (j & 3) + 8 * (j >> 2) + 4 * half;
    const int row = my_row0 + r;
    if (row < grp_hi) {
#pragma unroll
      for (int nb = 0; nb < TN / 32; ++nb) {
        Y[(long)row * Kout + n0 + nb * 32 + ln] = f2bf(acc[nb][j]);
      }
    }
  }
}

// ============================================================= wgrad kernel
//
// dw[g, m, k] = sum_{n in group g} dy[n, m] * x[n, k]
// grid: (M/TM_tiles * K/TN_tiles flattened per group? -> (m_tile, k_tile, E))
__global__ __launch_bounds__(gg::NT) void grouped_gemm_wgrad_kernel(
    const unsigned short* __restrict__ dY,  // [N, M]
    const unsigned short* __restrict__ X,   // [N, K]
    float* __restrict__ dW,                 // [E, M, K] fp32 (summed downstream to bf16 if wanted)
    const int* __restrict__ offs,
    int M, int K) {
  using namespace gg;
  extern __shared__ char smem[];
  // both tiles [BK contraction rows][128 cols] subtiled
  constexpr int T_BYTES = sub_bytes<BK, TM>();
  char* DYlds = smem;            // [BK n][128 m]
  char* Xlds = smem + T_BYTES;   // [BK n][128 k]

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int half = lane >> 5;
  const int ln = lane & 31;

  int swz_x, swz_y, swz_z;
  xcd_swizzle_xyz(swz_x, swz_y, swz_z);  // same-expert tiles share dy/x panels
  const int g = swz_z;
  const int m0 = swz_x * TM;
  const int k0 = swz_y * TN;
  const int grp_lo = offs[g], grp_hi = offs[g + 1];
  const int count = grp_hi - grp_lo;
  if (count == 0) {
    // zero the tile (each (m,k) owned by exactly one block)
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int m = m0 + w * 32 + (j & 3) + 8 * (j >> 2) + 4 * half;
      for (int nb = 0; nb < TN / 32; ++nb)
        dW[((long)g * M + m) * K + k0 + nb * 32 + ln] = 0.0f;
    }
    return;
  }

  f32x16 acc[TN / 32];
#pragma unroll
  for (int nb = 0; nb < TN / 32; ++nb)
#pragma unroll
    for (int j = 0; j < 16; ++j) acc[nb][j] = 0.0f;

  const int nsteps = (count + BK - 1) / BK;
  GGStage<BK, TM, true> dystage, xstage;  // ZEROPAD the ragged tail

  dystage.load(dY, M, grp_lo, grp_hi, m0);
  xstage.load(X, K, grp_lo, grp_hi, k0);
  dystage.store(DYlds);
  xstage.store(Xlds);
  __syncthreads();

  for (int t = 0; t < nsteps; ++t) {
    const bool has_next = (t + 1 < nsteps);
    if (has_next) {
      dystage.load(dY, M, grp_lo + (t + 1) * BK, grp_hi, m0);
      xstage.load(X, K, grp_lo + (t + 1) * BK, grp_hi, k0);
    }

    // A-frag col index = m = w*32 + (lane&31); ld_frag_tr's internal
    // cg = cg_base + ((lane>>4)&1) covers 32 consecutive cols from
    // cg_base*16 — cg_base = w*2 selects this wave's 32-m window.
    {
      constexpr int KS = BK / 16, NB = TN / 32;
      bf16x8_v af_cur = ld_frag_tr<BK>(DYlds, lane, 0, w * 2);
      bf16x8_v bf_cur = ld_frag_tr<BK>(Xlds, lane, 0, 0);
      bf16x8_v af_nxt, bf_nxt;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
#pragma unroll
        for (int nb = 0; nb < NB; ++nb) {
          if (nb + 1 < NB) {
            bf_nxt = ld_frag_tr<BK>(Xlds, lane, ks * 16, (nb + 1) * 2);
          } else if (ks + 1 < KS) {
            af_nxt = ld_frag_tr<BK>(DYlds, lane, (ks + 1) * 16, w * 2);
            bf_nxt = ld_frag_tr<BK>(Xlds, lane, (ks + 1) * 16, 0);
          }
          __builtin_amdgcn_s_setprio(1);
          acc[nb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af_cur, bf_cur, acc[nb], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
          bf_cur = bf_nxt;
          if (nb + 1 == NB) af_cur = af_nxt;
        }
      }
    }

    __syncthreads();
    if (has_next) {
      dystage.store(DYlds);
      xstage.store(Xlds);
    }
    __syncthreads();
  }

  // epilogue: C row = m (crow within this wave's 32-m window), col = k
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const int m = m0 + w * 32 + (j & 3) + 8 * (j >> 2) + 4 * half;
#pragma unroll
    for (int nb = 0; nb < TN / 32; ++nb) {
      dW[((long)g * M + m) * K + k0 + nb * 32 + ln] = acc[nb][j];
    }
  }
}

// =============================================================== host side

namespace {

std::pair<at::Tensor, at::Tensor> build_tile_map(const std::vector<long>& offs, int& n_tiles,
                                                 at::TensorOptions opts) {
  std::vector<int> tg, tr;
  for (size_t g = 0; g + 1 < offs.size(); ++g) {
    const long cnt = offs[g + 1] - offs[g];
    for (long r = 0; r < cnt; r += gg::TM) {
      tg.push_back((int)g);
      tr.push_back((int)r);
    }
  }
  n_tiles = (int)tg.size();
  if (n_tiles == 0) {
    tg.push_back(0);
    tr.push_back(0);
  }
  auto t_g = at::tensor(tg, opts.dtype(at::kInt));
  auto t_r = at::tensor(tr, opts.dtype(at::kInt));
  return {t_g.to(at::kCUDA, /*non_blocking=*/true), t_r.to(at::kCUDA, true)};
}

at::Tensor offs_to_device(const std::vector<long>& offs) {
  std::vector<int> o(offs.begin(), offs.end());
  return at::tensor(o, at::TensorOptions().dtype(at::kInt)).to(at::kCUDA, true);
}

}  // namespace

// y = x @ w[g]^T per row group. offsets: host ints (E+1).
at::Tensor grouped_gemm_fwd(at::Tensor x, at::Tensor w, std::vector<long> offsets) {
  using namespace gg;
  TORCH_CHECK(x.dim() == 2 && w.dim() == 3 && x.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16);
  const int N = (int)x.size(0), K = (int)x.size(1);
  const int E = (int)w.size(0), M = (int)w.size(1);
  TORCH_CHECK((int)w.size(2) == K && (int)offsets.size() == E + 1 && offsets[E] == N);
  TORCH_CHECK(K % BK == 0 && M % TN == 0, "grouped_gemm: K % 64 == 0 and M % 128 == 0 required");
  auto y = at::empty({N, M}, x.options());
  if (N == 0) return y;
  int n_tiles;
  auto [tg, tr] = build_tile_map(offsets, n_tiles, x.options());
  if (n_tiles == 0) return y;
  auto offs_d = offs_to_device(offsets);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds = sub_bytes<TM, BK>() + sub_bytes<TN, BK>();
  const dim3 grid(n_tiles, M / TN);
  hipLaunchKernelGGL((grouped_gemm_kernel<false>), grid, dim3(NT), lds, stream.stream(),
                     (const unsigned short*)x.data_ptr(), (const unsigned short*)w.data_ptr(),
                     (unsigned short*)y.data_ptr(), tg.data_ptr<int>(), tr.data_ptr<int>(),
                     offs_d.data_ptr<int>(), K, M, (long)M * K);
  HIP_CHECK_LAST();
  return y;
}

// dx = dy @ w[g] per row group.
at::Tensor grouped_gemm_dgrad(at::Tensor dy, at::Tensor w, std::vector<long> offsets) {
  using namespace gg;
  TORCH_CHECK(dy.dim() == 2 && w.dim() == 3 && dy.is_contiguous() && w.is_contiguous());
  const int N = (int)dy.size(0), M = (int)dy.size(1);
  const int E = (int)w.size(0), K = (int)w.size(2);
  TORCH_CHECK((int)w.size(1) == M && (int)offsets.size() == E + 1 && offsets[E] == N);
  TORCH_CHECK(M % BK == 0 && K % TN == 0, "grouped_gemm_dgrad: M % 64 == 0 and K % 128 == 0 required");
  auto dx = at::empty({N, K}, dy.options());
  if (N == 0) return dx;
  int n_tiles;
  auto [tg, tr] = build_tile_map(offsets, n_tiles, dy.options());
  if (n_tiles == 0) return dx;
  auto offs_d = offs_to_device(offsets);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds = sub_bytes<TM, BK>() + sub_bytes<BK, TN>();
  const dim3 grid(n_tiles, K / TN);
  hipLaunchKernelGGL((grouped_gemm_kernel<true>), grid, dim3(NT), lds, stream.stream(),
                     (const unsigned short*)dy.data_ptr(), (const unsigned short*)w.data_ptr(),
                     (unsigned short*)dx.data_ptr(), tg.data_ptr<int>(), tr.data_ptr<int>(),
                     offs_d.data_ptr<int>(), M, K, (long)M * K);
  HIP_CHECK_LAST();
  return dx;
}

// dw[g] = dy_g^T @ x_g per group (fp32 out).
at::Tensor grouped_gemm_wgrad(at::Tensor dy, at::Tensor x, std::vector<long> offsets) {
  using namespace gg;
  TORCH_CHECK(dy.dim() == 2 && x.dim() == 2 && dy.is_contiguous() && x.is_contiguous());
  const int N = (int)dy.size(0), M = (int)dy.size(1), K = (int)x.size(1);
  const int E = (int)offsets.size() - 1;
  TORCH_CHECK((int)x.size(0) == N && offsets[E] == N);
  TORCH_CHECK(M % TM == 0 && K % TN == 0, "grouped_gemm_wgrad: M,K % 128 == 0 required");
  auto dw = at::empty({E, M, K}, dy.options().dtype(at::kFloat));
  auto offs_d = offs_to_device(offsets);
  auto stream = at::hip::getCurrentHIPStream();
  const size_t lds = 2 * sub_bytes<BK, TM>();
  const dim3 grid(M / TM, K / TN, E);
  hipLaunchKernelGGL(grouped_gemm_wgrad_kernel, grid, dim3(NT), lds, stream.stream(),
                     (const unsigned short*)dy.data_ptr(), (const unsigned short*)x.data_ptr(),
                     dw.data_ptr<float>(), offs_d.data_ptr<int>(), M, K);
  HIP_CHECK_LAST();
  return dw;
}

}  // namespace cai
