// Fused multi-tensor Adam/AdamW for gfx950.
//
// Design (MI355X-native, not a CUDA port):
//  - memory-bound op → everything rides on vectorized 16 B/lane loads;
//    bf16 tensors are loaded as short4/short8 reinterpret (hipcc does not
//    auto-vectorize scalar bf16). 256-thread blocks (4 waves).
//  - the same kernel serves (a) HybridAdam-style per-parameter lists and
//    (b) the ZeRO flat-shard step (single huge tensor, caller passes a
//    chunk size that yields ~2048 blocks so the whole chip is filled and
//    the kernel grid-strides nothing).
//  - optional fused bf16 write-back: when training keeps fp32 masters and
//    bf16 working params, the updated master is converted and stored to
//    the working copy in the same pass — saves one full read+write of the
//    parameters per step vs. a separate cast kernel (HBM-bound op, so
//    this is a direct step-time win).
//
// Functional equivalent of the reference's
// extensions/csrc/kernel/cuda/multi_tensor_adam_kernel.cu (re-derived; no
// code shared). Python consumer: colossalai_amd/nn/optimizer/fused_adam.py.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <vector>

#include "common.h"

namespace cai {

constexpr int MTA_TENSORS = 24;
constexpr int MTA_BLOCKS = 320;
constexpr int ADAM_BLOCK = 256;

struct TensorListMeta {
  void* addresses[5][MTA_TENSORS];
  long sizes[MTA_TENSORS];
  short block_to_tensor[MTA_BLOCKS];
  int block_to_chunk[MTA_BLOCKS];
  int global_idx[MTA_TENSORS];  // position in the FULL tensor list (LAMB norms)
};

template <typename T>
struct VecIO;

template <>
struct VecIO<float> {
  // 4 floats = 16 B
  DEV_INLINE static void load4(const float* p, float (&out)[4]) {
    float4v v = *reinterpret_cast<const float4v*>(p);
    out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
  }
  DEV_INLINE static void store4(float* p, const float (&in)[4]) {
    float4v v = {in[0], in[1], in[2], in[3]};
    *reinterpret_cast<float4v*>(p) = v;
  }
  DEV_INLINE static float load1(const float* p) { return *p; }
  DEV_INLINE static void store1(float* p, float v) { *p = v; }
};

template <>
struct VecIO<unsigned short> {  // bf16 as raw u16
  DEV_INLINE static void load4(const unsigned short* p, float (&out)[4]) {
    short4v v = *reinterpret_cast<const short4v*>(p);
    out[0] = bf2f((unsigned short)v.x);
    out[1] = bf2f((unsigned short)v.y);
    out[2] = bf2f((unsigned short)v.z);
    out[3] = bf2f((unsigned short)v.w);
  }
  DEV_INLINE static void store4(unsigned short* p, const float (&in)[4]) {
    short4v v = {(short)f2bf(in[0]), (short)f2bf(in[1]), (short)f2bf(in[2]), (short)f2bf(in[3])};
    *reinterpret_cast<short4v*>(p) = v;
  }
  DEV_INLINE static float load1(const unsigned short* p) { return bf2f(*p); }
  DEV_INLINE static void store1(unsigned short* p, float v) { *p = f2bf(v); }
};

template <typename GT, typename PT, bool HAS_OUT, bool ADAMW>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_adam_kernel(
    TensorListMeta meta,
    long chunk_size,
    float beta1,
    float beta2,
    float bc1,        // 1 / (1 - beta1^t)
    float bc2_sqrt,   // 1 / sqrt(1 - beta2^t)
    float eps,
    float lr,
    float weight_decay,
    float inv_div_scale) {
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);

  const GT* g = reinterpret_cast<const GT*>(meta.addresses[0][tensor_id]) + offset;
  PT* p = reinterpret_cast<PT*>(meta.addresses[1][tensor_id]) + offset;
  float* m = reinterpret_cast<float*>(meta.addresses[2][tensor_id]) + offset;
  float* v = reinterpret_cast<float*>(meta.addresses[3][tensor_id]) + offset;
  unsigned short* out =
      HAS_OUT ? reinterpret_cast<unsigned short*>(meta.addresses[4][tensor_id]) + offset : nullptr;

  const long n4 = n & ~3L;
  for (long i = threadIdx.x * 4L; i < n4; i += (long)blockDim.x * 4L) {
    float gv[4], pv[4], mv[4], vv[4];
    VecIO<GT>::load4(g + i, gv);
    VecIO<PT>::load4(p + i, pv);
    VecIO<float>::load4(m + i, mv);
    VecIO<float>::load4(v + i, vv);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = gv[k] * inv_div_scale;
      if (!ADAMW) gk += weight_decay * pv[k];  // L2 mode
      mv[k] = beta1 * mv[k] + (1.0f - beta1) * gk;
      vv[k] = beta2 * vv[k] + (1.0f - beta2) * gk * gk;
      const float mhat = mv[k] * bc1;
      const float denom = sqrtf(vv[k]) * bc2_sqrt + eps;
      float update = mhat / denom;
      if (ADAMW) update += weight_decay * pv[k];
      pv[k] -= lr * update;
    }
    VecIO<PT>::store4(p + i, pv);
    VecIO<float>::store4(m + i, mv);
    VecIO<float>::store4(v + i, vv);
    if (HAS_OUT) VecIO<unsigned short>::store4(out + i, pv);
  }
  // tail
  for (long i = n4 + threadIdx.x; i < n; i += blockDim.x) {
    float gk = VecIO<GT>::load1(g + i) * inv_div_scale;
    float pk = VecIO<PT>::load1(p + i);
    float mk = m[i], vk = v[i];
    if (!ADAMW) gk += weight_decay * pk;
    mk = beta1 * mk + (1.0f - beta1) * gk;
    vk = beta2 * vk + (1.0f - beta2) * gk * gk;
    float update = (mk * bc1) / (sqrtf(vk) * bc2_sqrt + eps);
    if (ADAMW) update += weight_decay * pk;
    pk -= lr * update;
    VecIO<PT>::store1(p + i, pk);
    m[i] = mk;
    v[i] = vk;
    if (HAS_OUT) VecIO<unsigned short>::store1(out + i, pk);
  }
}

// ------------------------------------------------------------------- scale
template <typename T>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_scale_kernel(
    TensorListMeta meta, long chunk_size, float scale) {
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);
  const T* in = reinterpret_cast<const T*>(meta.addresses[0][tensor_id]) + offset;
  T* outp = reinterpret_cast<T*>(meta.addresses[1][tensor_id]) + offset;
  const long n4 = n & ~3L;
  for (long i = threadIdx.x * 4L; i < n4; i += (long)blockDim.x * 4L) {
    float x[4];
    VecIO<T>::load4(in + i, x);
#pragma unroll
    for (int k = 0; k < 4; ++k) x[k] *= scale;
    VecIO<T>::store4(outp + i, x);
  }
  for (long i = n4 + threadIdx.x; i < n; i += blockDim.x) {
    VecIO<T>::store1(outp + i, VecIO<T>::load1(in + i) * scale);
  }
}

// ----------------------------------------------------------------- l2 norm
// partials[block] = sum of squares over the block's chunk; finalized on the
// Python side with a single torch reduction over <= a few thousand floats.
template <typename T>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_l2norm_kernel(
    TensorListMeta meta, long chunk_size, float* partials) {
  __shared__ float smem[ADAM_BLOCK / WAVE];
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);
  const T* in = reinterpret_cast<const T*>(meta.addresses[0][tensor_id]) + offset;
  float acc = 0.0f;
  const long n4 = n & ~3L;
  for (long i = threadIdx.x * 4L; i < n4; i += (long)blockDim.x * 4L) {
    float x[4];
    VecIO<T>::load4(in + i, x);
#pragma unroll
    for (int k = 0; k < 4; ++k) acc += x[k] * x[k];
  }
  for (long i = n4 + threadIdx.x; i < n; i += blockDim.x) {
    float x = VecIO<T>::load1(in + i);
    acc += x * x;
  }
  acc = block_reduce_sum(acc, smem);
  if (threadIdx.x == 0) partials[blockIdx.x % MTA_BLOCKS] += acc;
}

// =============================================================== host side

namespace {

// Walk the tensor lists, fill TensorListMeta batches, invoke `launch` per batch.
template <typename LaunchFn>
void multi_tensor_apply(const std::vector<std::vector<at::Tensor>>& lists, long chunk_size, LaunchFn&& launch) {
  const int depth = (int)lists.size();
  TORCH_CHECK(depth >= 1 && depth <= 5, "multi_tensor_apply: depth must be in [1,5]");
  const size_t ntensors = lists[0].size();
  for (auto& l : lists) TORCH_CHECK(l.size() == ntensors, "tensor list length mismatch");
  if (ntensors == 0) return;

  TensorListMeta meta;
  int t_in_meta = 0;
  int b_in_meta = 0;
  for (size_t t = 0; t < ntensors; ++t) {
    const long numel = lists[0][t].numel();
    for (int d = 0; d < depth; ++d) {
      TORCH_CHECK(lists[d][t].numel() == numel, "tensor numel mismatch across lists");
      TORCH_CHECK(lists[d][t].is_contiguous(), "multi_tensor_apply requires contiguous tensors");
      meta.addresses[d][t_in_meta] = lists[d][t].data_ptr();
    }
    meta.sizes[t_in_meta] = numel;
    meta.global_idx[t_in_meta] = (int)t;
    const long nchunks = (numel + chunk_size - 1) / chunk_size;
    for (long c = 0; c < nchunks; ++c) {
      meta.block_to_tensor[b_in_meta] = (short)t_in_meta;
      meta.block_to_chunk[b_in_meta] = (int)c;
      ++b_in_meta;
      const bool tensors_full = (t_in_meta == MTA_TENSORS - 1) && (c == nchunks - 1);
      const bool blocks_full = (b_in_meta == MTA_BLOCKS);
      const bool last = (t == ntensors - 1) && (c == nchunks - 1);
      if (blocks_full || (tensors_full && !last) || last) {
        launch(meta, b_in_meta);
        b_in_meta = 0;
        if (blocks_full && c < nchunks - 1) {
          // continue the same tensor in a fresh batch
          for (int d = 0; d < depth; ++d) meta.addresses[d][0] = lists[d][t].data_ptr();
          meta.sizes[0] = numel;
          meta.global_idx[0] = (int)t;
          t_in_meta = 0;
        } else if (!last) {
          t_in_meta = -1;  // next tensor becomes index 0
        }
      }
    }
    ++t_in_meta;
    if (t_in_meta >= MTA_TENSORS) t_in_meta = 0;
  }
}

bool is_bf16(const at::Tensor& t) { return t.scalar_type() == at::kBFloat16; }

}  // namespace

void multi_tensor_adam(
    std::vector<at::Tensor> grads,
    std::vector<at::Tensor> params,
    std::vector<at::Tensor> exp_avgs,
    std::vector<at::Tensor> exp_avg_sqs,
    std::vector<at::Tensor> param_outs,  // empty, or bf16 working copies
    double lr,
    double beta1,
    double beta2,
    double eps,
    long step,
    bool adamw_mode,
    bool bias_correction,
    double weight_decay,
    double div_scale,
    long chunk_size) {
  TORCH_CHECK(!grads.empty(), "multi_tensor_adam: empty tensor list");
  const bool has_out = !param_outs.empty();
  float bc1 = 1.0f, bc2_sqrt = 1.0f;
  if (bias_correction) {
    bc1 = 1.0f / (1.0f - powf((float)beta1, (float)step));
    bc2_sqrt = 1.0f / sqrtf(1.0f - powf((float)beta2, (float)step));
  }
  const float inv_div_scale = (float)(1.0 / div_scale);
  auto stream = at::hip::getCurrentHIPStream();

  const bool g_bf16 = is_bf16(grads[0]);
  const bool p_bf16 = is_bf16(params[0]);

  std::vector<std::vector<at::Tensor>> lists = {grads, params, exp_avgs, exp_avg_sqs};
  if (has_out) lists.push_back(param_outs);

  auto run = [&](auto gt, auto pt, auto has_out_c, auto adamw_c) {
    using GT = decltype(gt);
    using PT = decltype(pt);
    multi_tensor_apply(lists, chunk_size, [&](const TensorListMeta& meta, int nblocks) {
      hipLaunchKernelGGL((multi_tensor_adam_kernel<GT, PT, decltype(has_out_c)::value, decltype(adamw_c)::value>),
                         dim3(nblocks), dim3(ADAM_BLOCK), 0, stream.stream(), meta, chunk_size,
                         (float)beta1, (float)beta2, bc1, bc2_sqrt, (float)eps, (float)lr,
                         (float)weight_decay, inv_div_scale);
      HIP_CHECK_LAST();
    });
  };

  using TrueT = std::integral_constant<bool, true>;
  using FalseT = std::integral_constant<bool, false>;
#define DISPATCH_ADAM(GT, PT)                                       \
  do {                                                              \
    if (has_out && adamw_mode) run(GT{}, PT{}, TrueT{}, TrueT{});   \
    else if (has_out) run(GT{}, PT{}, TrueT{}, FalseT{});           \
    else if (adamw_mode) run(GT{}, PT{}, FalseT{}, TrueT{});        \
    else run(GT{}, PT{}, FalseT{}, FalseT{});                       \
  } while (0)

  using US = unsigned short;
  if (g_bf16 && p_bf16) DISPATCH_ADAM(US, US);
  else if (g_bf16) DISPATCH_ADAM(US, float);
  else if (p_bf16) DISPATCH_ADAM(float, US);
  else DISPATCH_ADAM(float, float);
#undef DISPATCH_ADAM
}

// --------------------------------------------------------------------- sgd
template <typename GT, typename PT, bool HAS_OUT, bool MOMENTUM, bool NESTEROV>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_sgd_kernel(
    TensorListMeta meta, long chunk_size, float lr, float momentum, float dampening,
    float weight_decay, float inv_div_scale) {
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);

  const GT* g = reinterpret_cast<const GT*>(meta.addresses[0][tensor_id]) + offset;
  PT* p = reinterpret_cast<PT*>(meta.addresses[1][tensor_id]) + offset;
  float* buf = MOMENTUM ? reinterpret_cast<float*>(meta.addresses[2][tensor_id]) + offset : nullptr;
  unsigned short* out =
      HAS_OUT ? reinterpret_cast<unsigned short*>(meta.addresses[MOMENTUM ? 3 : 2][tensor_id]) + offset
              : nullptr;

  for (long i = threadIdx.x; i < n; i += blockDim.x) {
    float gk = VecIO<GT>::load1(g + i) * inv_div_scale;
    float pk = VecIO<PT>::load1(p + i);
    gk += weight_decay * pk;
    float upd = gk;
    if (MOMENTUM) {
      float bk = momentum * buf[i] + (1.0f - dampening) * gk;
      buf[i] = bk;
      upd = NESTEROV ? gk + momentum * bk : bk;
    }
    pk -= lr * upd;
    VecIO<PT>::store1(p + i, pk);
    if (HAS_OUT) VecIO<unsigned short>::store1(out + i, pk);
  }
}

void multi_tensor_sgd(
    std::vector<at::Tensor> grads,
    std::vector<at::Tensor> params,
    std::vector<at::Tensor> momentum_bufs,  // empty when momentum == 0
    std::vector<at::Tensor> param_outs,     // empty, or bf16 working copies
    double lr, double momentum, double dampening, double weight_decay,
    bool nesterov, double div_scale, long chunk_size) {
  TORCH_CHECK(!grads.empty(), "multi_tensor_sgd: empty tensor list");
  const bool has_mom = !momentum_bufs.empty();
  const bool has_out = !param_outs.empty();
  const float inv_div_scale = (float)(1.0 / div_scale);
  auto stream = at::hip::getCurrentHIPStream();
  const bool g_bf16 = is_bf16(grads[0]);
  const bool p_bf16 = is_bf16(params[0]);

  std::vector<std::vector<at::Tensor>> lists = {grads, params};
  if (has_mom) lists.push_back(momentum_bufs);
  if (has_out) lists.push_back(param_outs);

  auto run = [&](auto gt, auto pt, auto out_c, auto mom_c, auto nest_c) {
    using GT = decltype(gt);
    using PT = decltype(pt);
    multi_tensor_apply(lists, chunk_size, [&](const TensorListMeta& meta, int nblocks) {
      hipLaunchKernelGGL((multi_tensor_sgd_kernel<GT, PT, decltype(out_c)::value,
                                                  decltype(mom_c)::value, decltype(nest_c)::value>),
                         dim3(nblocks), dim3(ADAM_BLOCK), 0, stream.stream(), meta, chunk_size,
                         (float)lr, (float)momentum, (float)dampening, (float)weight_decay,
                         inv_div_scale);
      HIP_CHECK_LAST();
    });
  };
  using TrueT = std::integral_constant<bool, true>;
  using FalseT = std::integral_constant<bool, false>;
  using US = unsigned short;
#define DISPATCH_SGD(GT, PT)                                                        \
  do {                                                                              \
    if (has_mom && nesterov) {                                                      \
      if (has_out) run(GT{}, PT{}, TrueT{}, TrueT{}, TrueT{});                      \
      else run(GT{}, PT{}, FalseT{}, TrueT{}, TrueT{});                             \
    } else if (has_mom) {                                                           \
      if (has_out) run(GT{}, PT{}, TrueT{}, TrueT{}, FalseT{});                     \
      else run(GT{}, PT{}, FalseT{}, TrueT{}, FalseT{});                            \
    } else {                                                                        \
      if (has_out) run(GT{}, PT{}, TrueT{}, FalseT{}, FalseT{});                    \
      else run(GT{}, PT{}, FalseT{}, FalseT{}, FalseT{});                           \
    }                                                                               \
  } while (0)
  if (g_bf16 && p_bf16) DISPATCH_SGD(US, US);
  else if (g_bf16) DISPATCH_SGD(US, float);
  else if (p_bf16) DISPATCH_SGD(float, US);
  else DISPATCH_SGD(float, float);
#undef DISPATCH_SGD
}

// --------------------------------------------------------------------- lamb
// Two MTA passes (reference: multi_tensor_lamb_kernel.cu two-stage scheme):
// stage 1 runs the Adam-style moment update, writes the raw update u into a
// scratch list and block-atomically accumulates per-tensor ||p||^2 / ||u||^2;
// stage 2 applies p -= lr * trust * u with trust = ||p|| / ||u|| read from
// the accumulators (computed per block, no host sync).
template <typename GT, typename PT>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_lamb_stage1_kernel(
    TensorListMeta meta, long chunk_size, float beta1, float beta2, float bc1, float bc2,
    float eps, float weight_decay, float inv_div_scale, float* __restrict__ norms) {
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);

  const GT* g = reinterpret_cast<const GT*>(meta.addresses[0][tensor_id]) + offset;
  const PT* p = reinterpret_cast<const PT*>(meta.addresses[1][tensor_id]) + offset;
  float* m = reinterpret_cast<float*>(meta.addresses[2][tensor_id]) + offset;
  float* v = reinterpret_cast<float*>(meta.addresses[3][tensor_id]) + offset;
  float* u = reinterpret_cast<float*>(meta.addresses[4][tensor_id]) + offset;

  float pn = 0.0f, un = 0.0f;
  for (long i = threadIdx.x; i < n; i += blockDim.x) {
    const float gk = VecIO<GT>::load1(g + i) * inv_div_scale;
    const float pk = VecIO<PT>::load1(p + i);
    float mk = beta1 * m[i] + (1.0f - beta1) * gk;
    float vk = beta2 * v[i] + (1.0f - beta2) * gk * gk;
    m[i] = mk;
    v[i] = vk;
    float uk = (mk * bc1) / (sqrtf(vk * bc2) + eps) + weight_decay * pk;
    u[i] = uk;
    pn += pk * pk;
    un += uk * uk;
  }
  // block reduce (wave shuffle + LDS across the 4 waves)
  __shared__ float sm[2][4];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    pn += __shfl_xor(pn, off);
    un += __shfl_xor(un, off);
  }
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) { sm[0][wave] = pn; sm[1][wave] = un; }
  __syncthreads();
  if (threadIdx.x == 0) {
    pn = sm[0][0] + sm[0][1] + sm[0][2] + sm[0][3];
    un = sm[1][0] + sm[1][1] + sm[1][2] + sm[1][3];
    const int gi = meta.global_idx[tensor_id];
    atomicAdd(norms + 2 * gi, pn);
    atomicAdd(norms + 2 * gi + 1, un);
  }
}

template <typename PT, bool HAS_OUT>
__global__ __launch_bounds__(ADAM_BLOCK) void multi_tensor_lamb_stage2_kernel(
    TensorListMeta meta, long chunk_size, float lr, const float* __restrict__ norms) {
  const int tensor_id = meta.block_to_tensor[blockIdx.x];
  const long chunk_id = meta.block_to_chunk[blockIdx.x];
  const long offset = chunk_id * chunk_size;
  const long n = min(chunk_size, meta.sizes[tensor_id] - offset);

  const float* u = reinterpret_cast<const float*>(meta.addresses[0][tensor_id]) + offset;
  PT* p = reinterpret_cast<PT*>(meta.addresses[1][tensor_id]) + offset;
  unsigned short* out =
      HAS_OUT ? reinterpret_cast<unsigned short*>(meta.addresses[2][tensor_id]) + offset : nullptr;

  const int gi = meta.global_idx[tensor_id];
  const float pn = sqrtf(norms[2 * gi]);
  const float un = sqrtf(norms[2 * gi + 1]);
  const float trust = (pn > 0.0f && un > 0.0f) ? pn / un : 1.0f;
  const float step_sz = lr * trust;
  for (long i = threadIdx.x; i < n; i += blockDim.x) {
    const float pk = VecIO<PT>::load1(p + i) - step_sz * u[i];
    VecIO<PT>::store1(p + i, pk);
    if (HAS_OUT) VecIO<unsigned short>::store1(out + i, pk);
  }
}

void multi_tensor_lamb(
    std::vector<at::Tensor> grads,
    std::vector<at::Tensor> params,
    std::vector<at::Tensor> exp_avgs,
    std::vector<at::Tensor> exp_avg_sqs,
    std::vector<at::Tensor> param_outs,  // empty, or bf16 working copies
    double lr, double beta1, double beta2, double eps, long step,
    bool bias_correction, double weight_decay, double div_scale, long chunk_size) {
  TORCH_CHECK(!grads.empty(), "multi_tensor_lamb: empty tensor list");
  const bool has_out = !param_outs.empty();
  float bc1 = 1.0f, bc2 = 1.0f;
  if (bias_correction) {
    bc1 = 1.0f / (1.0f - powf((float)beta1, (float)step));
    bc2 = 1.0f / (1.0f - powf((float)beta2, (float)step));
  }
  const float inv_div_scale = (float)(1.0 / div_scale);
  auto stream = at::hip::getCurrentHIPStream();

  const long nt = (long)grads.size();
  auto norms = at::zeros({2 * nt}, params[0].options().dtype(at::kFloat));
  float* nptr = norms.data_ptr<float>();
  std::vector<at::Tensor> scratch;
  scratch.reserve(nt);
  for (auto& gt : grads) scratch.push_back(at::empty_like(gt, gt.options().dtype(at::kFloat)));

  const bool g_bf16 = is_bf16(grads[0]);
  const bool p_bf16 = is_bf16(params[0]);
  std::vector<std::vector<at::Tensor>> l1 = {grads, params, exp_avgs, exp_avg_sqs, scratch};
  auto run1 = [&](auto gt, auto pt) {
    using GT = decltype(gt);
    using PT = decltype(pt);
    multi_tensor_apply(l1, chunk_size, [&](const TensorListMeta& meta, int nblocks) {
      hipLaunchKernelGGL((multi_tensor_lamb_stage1_kernel<GT, PT>), dim3(nblocks),
                         dim3(ADAM_BLOCK), 0, stream.stream(), meta, chunk_size, (float)beta1,
                         (float)beta2, bc1, bc2, (float)eps, (float)weight_decay, inv_div_scale,
                         nptr);
      HIP_CHECK_LAST();
    });
  };
  using US = unsigned short;
  if (g_bf16 && p_bf16) run1(US{}, US{});
  else if (g_bf16) run1(US{}, float{});
  else if (p_bf16) run1(float{}, US{});
  else run1(float{}, float{});

  std::vector<std::vector<at::Tensor>> l2 = {scratch, params};
  if (has_out) l2.push_back(param_outs);
  auto run2 = [&](auto pt, auto out_c) {
    using PT = decltype(pt);
    multi_tensor_apply(l2, chunk_size, [&](const TensorListMeta& meta, int nblocks) {
      hipLaunchKernelGGL((multi_tensor_lamb_stage2_kernel<PT, decltype(out_c)::value>),
                         dim3(nblocks), dim3(ADAM_BLOCK), 0, stream.stream(), meta, chunk_size,
                         (float)lr, nptr);
      HIP_CHECK_LAST();
    });
  };
  using TrueT = std::integral_constant<bool, true>;
  using FalseT = std::integral_constant<bool, false>;
  if (p_bf16) { if (has_out) run2(US{}, TrueT{}); else run2(US{}, FalseT{}); }
  else { if (has_out) run2(float{}, TrueT{}); else run2(float{}, FalseT{}); }
}

void multi_tensor_scale(std::vector<at::Tensor> inputs, std::vector<at::Tensor> outputs, double scale,
                        long chunk_size) {
  auto stream = at::hip::getCurrentHIPStream();
  std::vector<std::vector<at::Tensor>> lists = {inputs, outputs};
  auto launch = [&](const TensorListMeta& meta, int nblocks) {
    if (is_bf16(inputs[0])) {
      hipLaunchKernelGGL((multi_tensor_scale_kernel<unsigned short>), dim3(nblocks), dim3(ADAM_BLOCK), 0,
                         stream.stream(), meta, chunk_size, (float)scale);
    } else {
      hipLaunchKernelGGL((multi_tensor_scale_kernel<float>), dim3(nblocks), dim3(ADAM_BLOCK), 0,
                         stream.stream(), meta, chunk_size, (float)scale);
    }
    HIP_CHECK_LAST();
  };
  multi_tensor_apply(lists, chunk_size, launch);
}

at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> inputs, long chunk_size) {
  auto stream = at::hip::getCurrentHIPStream();
  auto partials = at::zeros({MTA_BLOCKS}, inputs[0].options().dtype(at::kFloat));
  float* pptr = partials.data_ptr<float>();
  std::vector<std::vector<at::Tensor>> lists = {inputs};
  auto launch = [&](const TensorListMeta& meta, int nblocks) {
    if (is_bf16(inputs[0])) {
      hipLaunchKernelGGL((multi_tensor_l2norm_kernel<unsigned short>), dim3(nblocks), dim3(ADAM_BLOCK), 0,
                         stream.stream(), meta, chunk_size, pptr);
    } else {
      hipLaunchKernelGGL((multi_tensor_l2norm_kernel<float>), dim3(nblocks), dim3(ADAM_BLOCK), 0,
                         stream.stream(), meta, chunk_size, pptr);
    }
    HIP_CHECK_LAST();
  };
  multi_tensor_apply(lists, chunk_size, launch);
  return partials.sum().sqrt();
}

}  // namespace cai
