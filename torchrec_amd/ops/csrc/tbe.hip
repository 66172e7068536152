// Table-Batched Embedding (TBE) kernels for MI355X (gfx950).
//
// MI355X-native equivalent of the reference's FBGEMM
// SplitTableBatchedEmbeddingBagsCodegen (executable spec:
// reference torchrec/distributed/triton_tbe/triton_table_batched_embeddings.py,
// forward :367, backward :787/:1292, fused rowwise-Adagrad semantics).
//
// Design (not a port):
//  * all tables of a group live in ONE flat fp32 weights buffer; per-table
//    element offsets are 16B-aligned so rows load as float4.
//  * forward: each wave serves SLOTS = 64/LPS bags concurrently; a bag's
//    pooled row is accumulated in fp32 registers (CHUNKS x float4 per lane),
//    gathered rows stream through L2/LDS-free (gather has no reuse;
//    guideline: rely on L2/L3 for hot rows).
//  * backward: indices are linearized into the group's global row space,
//    radix-sorted (hipCUB) with their positions, run-length segmented on
//    device (no host sync), then ONE wave-slot per duplicate-run accumulates
//    the gradient over all occurrences and applies the optimizer update
//    in-place (rowwise Adagrad / SGD) — deterministic: one writer per row.
//  * mean-pooling / per-sample-weight scaling flows through a per-position
//    float scale array, so the same backward serves sum/mean/weighted and
//    the sequence (non-pooled) path (grad row = position).

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <algorithm>
#include <cstdlib>
#include <hipcub/hipcub.hpp>
#include <rocprim/block/block_radix_sort.hpp>
#include <rocprim/device/device_segmented_radix_sort.hpp>

#include <type_traits>

#include "common.h"

namespace trec_amd {


// ---------------------------------------------------------------------------
// UVM support: weights may live in pinned host memory (EmbeddingLocation::
// MANAGED) — kernels read/write it over PCIe via the device-visible alias.
// ---------------------------------------------------------------------------

template <typename T>
static T* uvm_ptr(const at::Tensor& t) {
  if (t.numel() == 0) return nullptr;
  if (t.is_cuda()) return t.data_ptr<T>();
  TORCH_CHECK(t.is_pinned(), "TBE host-resident tensors must be pinned (UVM/MANAGED)");
  void* dp = nullptr;
  TREC_HIP_CHECK(hipHostGetDevicePointer(&dp, t.data_ptr(), 0));
  return static_cast<T*>(dp);
}

// ---------------------------------------------------------------------------
// stochastic rounding (bf16): add 16 random bits below the kept mantissa and
// truncate — unbiased, carries across the rounding boundary (reference: TBE
// stochastic_rounding for low-precision weights). fp16/fp32 stores stay
// round-to-nearest (fp16 is not a bit-truncation of fp32).
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t mix64(uint64_t k) {
  k ^= k >> 33;
  k *= 0xff51afd7ed558ccdULL;
  k ^= k >> 33;
  k *= 0xc4ceb9fe1a85ec53ULL;
  k ^= k >> 33;
  return k;
}

__device__ __forceinline__ unsigned short stoch_bf16_bits(float v, uint32_t r16) {
  uint32_t u = __float_as_uint(v);
  if ((u & 0x7F800000u) != 0x7F800000u) u += (r16 & 0xFFFFu);
  return (unsigned short)(u >> 16);
}

template <typename emb_t>
__device__ __forceinline__ void store4_maybe_stoch(emb_t* row, int col4, float4 v,
                                                   uint64_t rk, bool stoch) {
  Vec4<emb_t>::store(row, col4, v);
}

template <>
__device__ __forceinline__ void store4_maybe_stoch<__hip_bfloat16>(
    __hip_bfloat16* row, int col4, float4 v, uint64_t rk, bool stoch) {
  if (!stoch) {
    Vec4<__hip_bfloat16>::store(row, col4, v);
    return;
  }
  uint64_t r = mix64(rk);
  ushort4 q;
  q.x = stoch_bf16_bits(v.x, (uint32_t)(r));
  q.y = stoch_bf16_bits(v.y, (uint32_t)(r >> 16));
  q.z = stoch_bf16_bits(v.z, (uint32_t)(r >> 32));
  q.w = stoch_bf16_bits(v.w, (uint32_t)(r >> 48));
  reinterpret_cast<ushort4*>(row)[col4] = q;
}

static inline hipStream_t tbe_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// ---------------------------------------------------------------------------
// forward (pooled)
// ---------------------------------------------------------------------------

template <typename emb_t, typename o_t, int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_fwd_pooled_kernel(
    const emb_t* __restrict__ weights,
    const int64_t* __restrict__ table_elem_offsets,  // [T]
    const int32_t* __restrict__ dims,                // [T]
    const int32_t* __restrict__ feat_table,          // [F]
    const int64_t* __restrict__ d_out_offsets,       // [F+1] output col offsets
    const int64_t* __restrict__ indices,
    const int64_t* __restrict__ offsets,  // [F*B+1] feature-major bags
    const float* __restrict__ psw,        // nullable per-sample weights
    const float* __restrict__ cache_weights,   // nullable lxu cache rows
    const int32_t* __restrict__ cache_loc,     // per-position slot or -1
    int64_t cache_stride,
    int F, int B, int64_t total_D, bool mean_pool,
    o_t* __restrict__ out /* [B, total_D] */) {
  constexpr int SLOTS = kWaveSize / LPS;
  int sl = threadIdx.x % LPS;                    // lane within slot
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  (void)SLOTS;
  int64_t n_bags = static_cast<int64_t>(F) * B;
  for (int64_t bag = slot; bag < n_bags; bag += n_slots) {
    int f = bag / B;
    int b = bag - static_cast<int64_t>(f) * B;
    int t = feat_table[f];
    int D = dims[t];
    const emb_t* tab = weights + table_elem_offsets[t];
    float4 acc[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) acc[c] = make_float4(0.f, 0.f, 0.f, 0.f);
    int64_t i0 = offsets[bag], i1 = offsets[bag + 1];
    for (int64_t i = i0; i < i1; ++i) {
      int64_t idx = indices[i];
      const emb_t* row = tab + idx * D;
      const float* crow = nullptr;
      if (cache_loc && cache_loc[i] >= 0) {
        crow = cache_weights + static_cast<int64_t>(cache_loc[i]) * cache_stride;
      }
      float w = psw ? psw[i] : 1.f;
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          float4 v = crow ? Vec4<float>::load(crow, col4) : Vec4<emb_t>::load(row, col4);
          acc[c].x += w * v.x;
          acc[c].y += w * v.y;
          acc[c].z += w * v.z;
          acc[c].w += w * v.w;
        }
      }
    }
    float scale = 1.f;
    if (mean_pool && i1 > i0) scale = 1.f / static_cast<float>(i1 - i0);
    o_t* orow = out + static_cast<int64_t>(b) * total_D + d_out_offsets[f];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int col4 = c * LPS + sl;
      if (col4 * 4 < D) {
        Vec4<o_t>::store(orow, col4,
                         make_float4(acc[c].x * scale, acc[c].y * scale,
                                     acc[c].z * scale, acc[c].w * scale));
      }
    }
  }
}

template <typename emb_t, typename o_t, typename host_o_t>
static void launch_tbe_fwd_pooled(
    const emb_t* weights, const at::Tensor& table_elem_offsets, const at::Tensor& dims,
    const at::Tensor& feat_table, const at::Tensor& d_out_offsets,
    const at::Tensor& indices, const at::Tensor& offsets, const float* psw_ptr,
    const float* cache_w_ptr, const int32_t* cache_loc_ptr, int64_t max_D, int F,
    int64_t B, int64_t total_D, bool mean_pool, at::Tensor& out, int lps, int chunks,
    int grid, hipStream_t stream) {
  o_t* out_ptr = reinterpret_cast<o_t*>(out.data_ptr<host_o_t>());
#define TBE_FWD_LAUNCH(LPS, CHUNKS)                                                       \
  hipLaunchKernelGGL((tbe_fwd_pooled_kernel<emb_t, o_t, LPS, CHUNKS>), dim3(grid),        \
                     dim3(kBlockThreads), 0, stream, weights,                             \
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(),    \
                     feat_table.data_ptr<int32_t>(), d_out_offsets.data_ptr<int64_t>(),   \
                     indices.data_ptr<int64_t>(), offsets.data_ptr<int64_t>(),            \
                     psw_ptr, cache_w_ptr, cache_loc_ptr, max_D, F, (int)B, total_D,      \
                     mean_pool, out_ptr)
  if (lps == 16) {
    TORCH_CHECK(chunks == 1);
    TBE_FWD_LAUNCH(16, 1);
  } else if (lps == 32) {
    TORCH_CHECK(chunks == 1);
    TBE_FWD_LAUNCH(32, 1);
  } else {
    switch (chunks) {
      case 1: TBE_FWD_LAUNCH(64, 1); break;
      case 2: TBE_FWD_LAUNCH(64, 2); break;
      case 3: case 4: TBE_FWD_LAUNCH(64, 4); break;
      default: TBE_FWD_LAUNCH(64, 8); break;
    }
  }
#undef TBE_FWD_LAUNCH
}

at::Tensor tbe_forward_pooled(
    const at::Tensor& weights, const at::Tensor& table_elem_offsets, const at::Tensor& dims,
    const at::Tensor& feat_table, const at::Tensor& d_out_offsets, const at::Tensor& indices,
    const at::Tensor& offsets, const at::Tensor& per_sample_weights, int64_t B,
    int64_t total_D, int64_t max_D, bool mean_pool, const at::Tensor& cache_weights,
    const at::Tensor& cache_loc, int64_t out_dtype) {
  TORCH_CHECK(max_D % 4 == 0 && max_D <= 2048, "TBE: dims must be %4==0 and <=2048");
  int F = feat_table.numel();
  // out_dtype: 0 = fp32, 1 = bf16, 2 = fp16 (reference SplitTBE output_dtype;
  // the pool still accumulates fp32 and rounds once on store)
  auto out_st = out_dtype == 1 ? at::kBFloat16 : (out_dtype == 2 ? at::kHalf : at::kFloat);
  auto out = at::empty({B, total_D}, indices.options().dtype(out_st));
  if (B == 0 || F == 0) return out;
  const float* psw_ptr =
      per_sample_weights.numel() > 0 ? per_sample_weights.data_ptr<float>() : nullptr;
  const float* cache_w_ptr =
      cache_weights.numel() > 0 ? cache_weights.data_ptr<float>() : nullptr;
  const int32_t* cache_loc_ptr =
      cache_loc.numel() > 0 ? cache_loc.data_ptr<int32_t>() : nullptr;
  auto stream = tbe_stream();
  int64_t n_bags = static_cast<int64_t>(F) * B;
  // pick lanes-per-slot and register chunks from max_D
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  // grid-strided (capped) launch: unlike the backward, the forward is
  // BANDWIDTH-bound — same-box A/B showed the full one-slot-per-bag grid
  // 4-10% SLOWER (55.1 vs 52.9 us fp32; block scheduling overhead with no
  // latency chain to hide). TREC_FWD_GRID_CAP overrides.
  int64_t fwd_cap = kMaxBlocks;
  if (const char* gc = std::getenv("TREC_FWD_GRID_CAP")) fwd_cap = std::atoll(gc);
  int grid = (int)std::min<int64_t>(
      (n_bags * lps + kBlockThreads - 1) / kBlockThreads, std::max<int64_t>(fwd_cap, 1));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, weights.scalar_type(),
                                  "tbe_fwd_pooled", [&] {
    using dev_t = typename DevType<scalar_t>::type;
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 embedding tables unsupported");
    } else {
      const dev_t* wp = reinterpret_cast<const dev_t*>(uvm_ptr<scalar_t>(weights));
      if (out_dtype == 1) {
        launch_tbe_fwd_pooled<dev_t, __hip_bfloat16, at::BFloat16>(
            wp, table_elem_offsets, dims, feat_table, d_out_offsets, indices, offsets,
            psw_ptr, cache_w_ptr, cache_loc_ptr, max_D, F, B, total_D, mean_pool, out,
            lps, chunks, grid, stream);
      } else if (out_dtype == 2) {
        launch_tbe_fwd_pooled<dev_t, __half, at::Half>(
            wp, table_elem_offsets, dims, feat_table, d_out_offsets, indices, offsets,
            psw_ptr, cache_w_ptr, cache_loc_ptr, max_D, F, B, total_D, mean_pool, out,
            lps, chunks, grid, stream);
      } else {
        launch_tbe_fwd_pooled<dev_t, float, float>(
            wp, table_elem_offsets, dims, feat_table, d_out_offsets, indices, offsets,
            psw_ptr, cache_w_ptr, cache_loc_ptr, max_D, F, B, total_D, mean_pool, out,
            lps, chunks, grid, stream);
      }
    }
  });
  return out;
}

// ---------------------------------------------------------------------------
// forward (pooled, VBE): per-feature batch sizes; output is 1-D packed
// [sum_f B_f * D_f] feature-major (reference: VBE TBE via generate_vbe_metadata,
// triton TBE b_t_map path :367). The backward needs no VBE variant: the
// generic (row, col) grad addressing treats the packed output as one row.
// ---------------------------------------------------------------------------

template <typename emb_t, int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_fwd_pooled_vbe_kernel(
    const emb_t* __restrict__ weights, const int64_t* __restrict__ table_elem_offsets,
    const int32_t* __restrict__ dims, const int32_t* __restrict__ feat_table,
    const int64_t* __restrict__ bag_offsets,   // [F+1] cumsum of B_f
    const int64_t* __restrict__ out_offsets,   // [F+1] cumsum of B_f * D_f
    const int64_t* __restrict__ indices, const int64_t* __restrict__ offsets,
    const float* __restrict__ psw, int F, int64_t n_bags, bool mean_pool,
    float* __restrict__ out) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  for (int64_t bag = slot; bag < n_bags; bag += n_slots) {
    int f = upper_bound_segment(bag_offsets, F, bag);
    int64_t b = bag - bag_offsets[f];
    int t = feat_table[f];
    int D = dims[t];
    const emb_t* tab = weights + table_elem_offsets[t];
    float4 acc[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) acc[c] = make_float4(0.f, 0.f, 0.f, 0.f);
    int64_t i0 = offsets[bag], i1 = offsets[bag + 1];
    for (int64_t i = i0; i < i1; ++i) {
      const emb_t* row = tab + indices[i] * D;
      float w = psw ? psw[i] : 1.f;
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          float4 v = Vec4<emb_t>::load(row, col4);
          acc[c].x += w * v.x;
          acc[c].y += w * v.y;
          acc[c].z += w * v.z;
          acc[c].w += w * v.w;
        }
      }
    }
    float scale = 1.f;
    if (mean_pool && i1 > i0) scale = 1.f / static_cast<float>(i1 - i0);
    float4* orow = reinterpret_cast<float4*>(out + out_offsets[f] + b * D);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int col4 = c * LPS + sl;
      if (col4 * 4 < D)
        orow[col4] = make_float4(acc[c].x * scale, acc[c].y * scale, acc[c].z * scale,
                                 acc[c].w * scale);
    }
  }
}

at::Tensor tbe_forward_pooled_vbe(
    const at::Tensor& weights, const at::Tensor& table_elem_offsets, const at::Tensor& dims,
    const at::Tensor& feat_table, const at::Tensor& bag_offsets,
    const at::Tensor& out_offsets, const at::Tensor& indices, const at::Tensor& offsets,
    const at::Tensor& per_sample_weights, int64_t n_bags, int64_t out_numel, int64_t max_D,
    bool mean_pool) {
  TORCH_CHECK(max_D % 4 == 0 && max_D <= 2048);
  int F = feat_table.numel();
  auto out = at::empty({out_numel}, indices.options().dtype(at::kFloat));
  if (n_bags == 0 || F == 0) return out;
  const float* psw_ptr =
      per_sample_weights.numel() > 0 ? per_sample_weights.data_ptr<float>() : nullptr;
  auto stream = tbe_stream();
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  int grid = grid_for(n_bags * lps, kBlockThreads);
#define TBE_VBE_LAUNCH(LPS, CHUNKS)                                                      \
  hipLaunchKernelGGL((tbe_fwd_pooled_vbe_kernel<dev_t, LPS, CHUNKS>), dim3(grid),        \
                     dim3(kBlockThreads), 0, stream,                                     \
                     reinterpret_cast<const dev_t*>(uvm_ptr<scalar_t>(weights)),         \
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(),   \
                     feat_table.data_ptr<int32_t>(), bag_offsets.data_ptr<int64_t>(),    \
                     out_offsets.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),       \
                     offsets.data_ptr<int64_t>(), psw_ptr, F, n_bags, mean_pool,         \
                     out.data_ptr<float>())
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, weights.scalar_type(),
                                  "tbe_fwd_vbe", [&] {
    using dev_t = typename DevType<scalar_t>::type;
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 embedding tables unsupported");
    } else {
    if (lps == 16) TBE_VBE_LAUNCH(16, 1);
    else if (lps == 32) TBE_VBE_LAUNCH(32, 1);
    else switch (chunks) {
      case 1: TBE_VBE_LAUNCH(64, 1); break;
      case 2: TBE_VBE_LAUNCH(64, 2); break;
      case 3: case 4: TBE_VBE_LAUNCH(64, 4); break;
      default: TBE_VBE_LAUNCH(64, 8); break;
    }
    }
  });
#undef TBE_VBE_LAUNCH
  return out;
}

// ---------------------------------------------------------------------------
// backward metadata: ONE launch for the per-position (row, col, linear-id)
// arrays, replacing torch's arange + repeat_interleave + div/sub + three
// index kernels (~12 launch-floor kernels per step in the captured graph).
// Each position binary-searches its owning bag in the offsets array (the
// FB+1 prefix sums stay L2-resident across the wave).
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256) tbe_bag_metadata_kernel(
    const int64_t* __restrict__ offsets,          // [FB+1]
    const int64_t* __restrict__ indices,          // [N]
    const int64_t* __restrict__ feat_d_out,       // [F]
    const int64_t* __restrict__ feat_row_offset,  // [F]
    int64_t FB, int64_t N, int B,
    int32_t* __restrict__ pos_row, int64_t* __restrict__ pos_col,
    int64_t* __restrict__ linear) {
  int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
  int64_t nthreads = static_cast<int64_t>(gridDim.x) * blockDim.x;
  for (; i < N; i += nthreads) {
    // first bag whose exclusive end exceeds position i
    int64_t lo = 0, hi = FB;
    while (lo < hi) {
      int64_t mid = (lo + hi) >> 1;
      if (offsets[mid + 1] <= i) lo = mid + 1; else hi = mid;
    }
    int64_t f = lo / B;
    pos_row[i] = static_cast<int32_t>(lo - f * B);
    pos_col[i] = feat_d_out[f];
    linear[i] = indices[i] + feat_row_offset[f];
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> tbe_bag_metadata(
    const at::Tensor& offsets, const at::Tensor& indices,
    const at::Tensor& feat_d_out, const at::Tensor& feat_row_offset, int64_t B) {
  int64_t N = indices.numel();
  int64_t FB = offsets.numel() - 1;
  auto pos_row = at::empty({N}, indices.options().dtype(at::kInt));
  auto pos_col = at::empty({N}, indices.options().dtype(at::kLong));
  auto linear = at::empty({N}, indices.options().dtype(at::kLong));
  if (N == 0) return {pos_row, pos_col, linear};
  auto stream = tbe_stream();
  int grid = grid_for(N, 256);
  hipLaunchKernelGGL(tbe_bag_metadata_kernel, dim3(grid), dim3(256), 0, stream,
                     offsets.data_ptr<int64_t>(), indices.data_ptr<int64_t>(),
                     feat_d_out.data_ptr<int64_t>(),
                     feat_row_offset.data_ptr<int64_t>(), FB, N, (int)B,
                     pos_row.data_ptr<int32_t>(), pos_col.data_ptr<int64_t>(),
                     linear.data_ptr<int64_t>());
  return {pos_row, pos_col, linear};
}

// ---------------------------------------------------------------------------
// forward (sequence / non-pooled): out[n, :] = W[table(f(n))][idx[n]]
// ---------------------------------------------------------------------------

template <typename emb_t, typename o_t, int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_fwd_seq_kernel(
    const emb_t* __restrict__ weights, const int64_t* __restrict__ table_elem_offsets,
    const int32_t* __restrict__ dims, const int32_t* __restrict__ feat_table,
    const int64_t* __restrict__ feat_val_offsets,  // [F+1] value range per feature
    const int64_t* __restrict__ indices, int F, int64_t N, int64_t D_out,
    o_t* __restrict__ out /* [N, D_out] */) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  for (int64_t n = slot; n < N; n += n_slots) {
    int f = upper_bound_segment(feat_val_offsets, F, n);
    int t = feat_table[f];
    int D = dims[t];
    const emb_t* row = weights + table_elem_offsets[t] + indices[n] * D;
    o_t* orow = out + n * D_out;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int col4 = c * LPS + sl;
      if (col4 * 4 < D) Vec4<o_t>::store(orow, col4, Vec4<emb_t>::load(row, col4));
    }
  }
}

template <typename emb_t, typename o_t, typename host_o_t>
static void launch_tbe_fwd_seq(const emb_t* wp,
                               const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                               const at::Tensor& feat_table,
                               const at::Tensor& feat_val_offsets,
                               const at::Tensor& indices, int F, int64_t N, int64_t D_out,
                               at::Tensor& out, int lps, int chunks, int grid,
                               hipStream_t stream) {
  o_t* out_ptr = reinterpret_cast<o_t*>(out.data_ptr<host_o_t>());
#define TBE_SEQ_LAUNCH(LPS, CHUNKS)                                                        \
  hipLaunchKernelGGL((tbe_fwd_seq_kernel<emb_t, o_t, LPS, CHUNKS>), dim3(grid),            \
                     dim3(kBlockThreads), 0, stream, wp,                                   \
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(),     \
                     feat_table.data_ptr<int32_t>(), feat_val_offsets.data_ptr<int64_t>(), \
                     indices.data_ptr<int64_t>(), F, N, D_out, out_ptr)
  if (lps == 16) TBE_SEQ_LAUNCH(16, 1);
  else if (lps == 32) TBE_SEQ_LAUNCH(32, 1);
  else switch (chunks) {
    case 1: TBE_SEQ_LAUNCH(64, 1); break;
    case 2: TBE_SEQ_LAUNCH(64, 2); break;
    case 3: case 4: TBE_SEQ_LAUNCH(64, 4); break;
    default: TBE_SEQ_LAUNCH(64, 8); break;
  }
#undef TBE_SEQ_LAUNCH
}

at::Tensor tbe_forward_seq(const at::Tensor& weights, const at::Tensor& table_elem_offsets,
                           const at::Tensor& dims, const at::Tensor& feat_table,
                           const at::Tensor& feat_val_offsets, const at::Tensor& indices,
                           int64_t D_out, int64_t max_D, int64_t out_dtype) {
  TORCH_CHECK(max_D % 4 == 0 && max_D <= 2048);
  int64_t N = indices.numel();
  auto out_st = out_dtype == 1 ? at::kBFloat16 : (out_dtype == 2 ? at::kHalf : at::kFloat);
  auto out = at::empty({N, D_out}, indices.options().dtype(out_st));
  if (N == 0) return out;
  int F = feat_table.numel();
  auto stream = tbe_stream();
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  int grid = grid_for(N * lps, kBlockThreads);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, weights.scalar_type(),
                                  "tbe_fwd_seq", [&] {
    using dev_t = typename DevType<scalar_t>::type;
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 embedding tables unsupported");
    } else {
      const dev_t* wp = reinterpret_cast<const dev_t*>(uvm_ptr<scalar_t>(weights));
      if (out_dtype == 1) {
        launch_tbe_fwd_seq<dev_t, __hip_bfloat16, at::BFloat16>(
            wp, table_elem_offsets, dims, feat_table, feat_val_offsets, indices, F, N,
            D_out, out, lps, chunks, grid, stream);
      } else if (out_dtype == 2) {
        launch_tbe_fwd_seq<dev_t, __half, at::Half>(
            wp, table_elem_offsets, dims, feat_table, feat_val_offsets, indices, F, N,
            D_out, out, lps, chunks, grid, stream);
      } else {
        launch_tbe_fwd_seq<dev_t, float, float>(
            wp, table_elem_offsets, dims, feat_table, feat_val_offsets, indices, F, N,
            D_out, out, lps, chunks, grid, stream);
      }
    }
  });
  return out;
}

// ---------------------------------------------------------------------------
// backward prep: radix sort + run-length segmentation, fully on device.
// ---------------------------------------------------------------------------

std::tuple<at::Tensor, at::Tensor> sort_pairs(const at::Tensor& keys, int64_t end_bit) {
  TORCH_CHECK(keys.is_cuda() && keys.scalar_type() == at::kLong);
  int64_t n = keys.numel();
  auto vals_in = at::arange(n, keys.options().dtype(at::kInt));
  auto vals_out = at::empty_like(vals_in);
  auto stream = tbe_stream();
  if (n == 0) return {at::empty_like(keys), vals_out};
  if (end_bit <= 31) {
    // id space fits 32-bit: halve the radix-sort key bandwidth
    auto k32 = keys.to(at::kInt);
    auto k32_out = at::empty_like(k32);
    size_t tmp_bytes = 0;
    hipcub::DeviceRadixSort::SortPairs(
        nullptr, tmp_bytes, k32.data_ptr<int32_t>(), k32_out.data_ptr<int32_t>(),
        vals_in.data_ptr<int32_t>(), vals_out.data_ptr<int32_t>(), n, 0, (int)end_bit,
        stream);
    auto tmp = at::empty({static_cast<int64_t>(tmp_bytes)}, keys.options().dtype(at::kByte));
    hipcub::DeviceRadixSort::SortPairs(
        tmp.data_ptr(), tmp_bytes, k32.data_ptr<int32_t>(), k32_out.data_ptr<int32_t>(),
        vals_in.data_ptr<int32_t>(), vals_out.data_ptr<int32_t>(), n, 0, (int)end_bit,
        stream);
    return {k32_out.to(at::kLong), vals_out};
  }
  auto keys_out = at::empty_like(keys);
  size_t tmp_bytes = 0;
  hipcub::DeviceRadixSort::SortPairs(
      nullptr, tmp_bytes, keys.data_ptr<int64_t>(), keys_out.data_ptr<int64_t>(),
      vals_in.data_ptr<int32_t>(), vals_out.data_ptr<int32_t>(), n, 0, (int)end_bit, stream);
  auto tmp = at::empty({static_cast<int64_t>(tmp_bytes)}, keys.options().dtype(at::kByte));
  hipcub::DeviceRadixSort::SortPairs(
      tmp.data_ptr(), tmp_bytes, keys.data_ptr<int64_t>(), keys_out.data_ptr<int64_t>(),
      vals_in.data_ptr<int32_t>(), vals_out.data_ptr<int32_t>(), n, 0, (int)end_bit, stream);
  return {keys_out, vals_out};
}

// ---------------------------------------------------------------------------
// segmented block radix sort: ONE launch replaces hipCUB's ~20-kernel device
// sort when every feature segment fits a workgroup (fixed-length bags, e.g.
// one-hot Criteo). Segment f of the feature-major layout is
// [offsets[f*B], offsets[(f+1)*B]); strictly-increasing feature_table_map
// keeps the concatenation globally grouped (caller-checked).
// ---------------------------------------------------------------------------

constexpr int kSortThreads = 1024;  // wide blocks: few segments must fill CUs

template <int ITEMS>
__global__ void __launch_bounds__(kSortThreads) seg_block_sort_kernel(
    const int64_t* __restrict__ linear, const int64_t* __restrict__ offsets, int B,
    int end_bit, int64_t* __restrict__ sorted_out, int32_t* __restrict__ perm_out,
    int32_t* __restrict__ overflow) {
  using sorter = rocprim::block_radix_sort<uint32_t, kSortThreads, ITEMS, uint32_t>;
  __shared__ typename sorter::storage_type storage;
  int f = blockIdx.x;
  int64_t lo = offsets[static_cast<int64_t>(f) * B];
  int64_t hi = offsets[static_cast<int64_t>(f + 1) * B];
  int count = static_cast<int>(hi - lo);
  if (count > kSortThreads * ITEMS) {
    if (threadIdx.x == 0) atomicOr(overflow, 1);
    count = kSortThreads * ITEMS;
  }
  uint32_t keys[ITEMS];
  uint32_t vals[ITEMS];
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    int k = static_cast<int>(threadIdx.x) * ITEMS + i;  // blocked arrangement
    if (k < count) {
      keys[i] = static_cast<uint32_t>(linear[lo + k]);
      vals[i] = static_cast<uint32_t>(lo + k);
    } else {
      keys[i] = 0xFFFFFFFFu;
      vals[i] = 0xFFFFFFFFu;
    }
  }
  sorter().sort(keys, vals, storage, 0u, static_cast<unsigned>(end_bit));
  // blocked arrangement out: thread t holds ranks [t*ITEMS, t*ITEMS+ITEMS)
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    int k = static_cast<int>(threadIdx.x) * ITEMS + i;
    if (k < count) {
      sorted_out[lo + k] = static_cast<int64_t>(keys[i]);
      perm_out[lo + k] = static_cast<int32_t>(vals[i]);
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs(
    const at::Tensor& linear, const at::Tensor& offsets, int64_t B, int64_t F,
    int64_t end_bit, int64_t capacity) {
  TORCH_CHECK(linear.is_cuda() && linear.scalar_type() == at::kLong);
  TORCH_CHECK(end_bit <= 32, "segmented sort requires 32-bit ids");
  auto sorted = at::empty_like(linear);
  auto perm = at::empty({linear.numel()}, linear.options().dtype(at::kInt));
  auto overflow = at::zeros({1}, linear.options().dtype(at::kInt));
  if (linear.numel() == 0 || F == 0) return {sorted, perm, overflow};
  auto stream = tbe_stream();
#define SEG_SORT_LAUNCH(ITEMS)                                                         \
  hipLaunchKernelGGL((seg_block_sort_kernel<ITEMS>), dim3((int)F), dim3(kSortThreads),  \
                     0, stream, linear.data_ptr<int64_t>(), offsets.data_ptr<int64_t>(), \
                     (int)B, (int)end_bit, sorted.data_ptr<int64_t>(),                  \
                     perm.data_ptr<int32_t>(), overflow.data_ptr<int32_t>())
  if (capacity <= kSortThreads * 2) SEG_SORT_LAUNCH(2);
  else if (capacity <= kSortThreads * 4) SEG_SORT_LAUNCH(4);
  else if (capacity <= kSortThreads * 8) SEG_SORT_LAUNCH(8);
  else {
    TORCH_CHECK(capacity <= kSortThreads * 16, "segment too large for block sort");
    SEG_SORT_LAUNCH(16);
  }
#undef SEG_SORT_LAUNCH
  return {sorted, perm, overflow};
}

// ---------------------------------------------------------------------------
// two-level segmented sort: stable tile sorts (512 keys/block — hundreds of
// workgroups fill the chip, vs ONE workgroup per segment in the single-level
// block sort) + log2(tiles) rounds of stable co-rank pairwise merges.
// No per-segment size cap (B = 65536 runs 128 tiles x 7 rounds).
// ---------------------------------------------------------------------------

constexpr int kTileThreads = 256;
constexpr int kTileItems = 2;
constexpr int kTileSize = kTileThreads * kTileItems;  // merge-chunk width

template <int ITEMS>
__global__ void __launch_bounds__(kTileThreads) seg_tile_sort_kernel(
    const int64_t* __restrict__ linear, const int64_t* __restrict__ offsets, int B,
    int tiles_per_seg, int end_bit, uint32_t* __restrict__ keys_out,
    uint32_t* __restrict__ perm_out) {
  constexpr int TS = kTileThreads * ITEMS;
  using sorter = rocprim::block_radix_sort<uint32_t, kTileThreads, ITEMS, uint32_t>;
  __shared__ typename sorter::storage_type storage;
  int seg = blockIdx.x / tiles_per_seg;
  int tile = blockIdx.x - seg * tiles_per_seg;
  int64_t lo = offsets[static_cast<int64_t>(seg) * B];
  int64_t hi = offsets[static_cast<int64_t>(seg + 1) * B];
  int64_t t0 = lo + static_cast<int64_t>(tile) * TS;
  if (t0 >= hi) return;
  int count = static_cast<int>(min(hi - t0, (int64_t)TS));
  uint32_t keys[ITEMS];
  uint32_t vals[ITEMS];
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    int k = static_cast<int>(threadIdx.x) * ITEMS + i;  // blocked
    if (k < count) {
      keys[i] = static_cast<uint32_t>(linear[t0 + k]);
      vals[i] = static_cast<uint32_t>(t0 + k);
    } else {
      keys[i] = 0xFFFFFFFFu;
      vals[i] = 0xFFFFFFFFu;
    }
  }
  sorter().sort(keys, vals, storage, 0u, static_cast<unsigned>(end_bit));
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    int k = static_cast<int>(threadIdx.x) * ITEMS + i;
    if (k < count) {
      keys_out[t0 + k] = keys[i];
      perm_out[t0 + k] = vals[i];
    }
  }
}

// co-rank of out position k in sorted runs A[0..la), B[0..lb): smallest ai
// such that merging is consistent; stable (ties take A first).
__device__ __forceinline__ int co_rank(int k, const uint32_t* A, int la,
                                       const uint32_t* Bp, int lb) {
  // monotone predicate: ai is too small while A[ai] <= B[bi-1] (stability:
  // equal keys take ALL of A's before any of B's). In-loop ai < hi <= la and
  // bi = k - ai >= 1, so both accesses are in range.
  int lo = max(0, k - lb), hi = min(k, la);
  while (lo < hi) {
    int ai = (lo + hi) >> 1;
    int bi = k - ai;
    if (A[ai] <= Bp[bi - 1]) {
      lo = ai + 1;
    } else {
      hi = ai;
    }
  }
  return lo;
}

template <bool OUT64>
__global__ void __launch_bounds__(kTileThreads) seg_merge_kernel(
    const uint32_t* __restrict__ keys_in, const uint32_t* __restrict__ vals_in,
    const int64_t* __restrict__ offsets, int B, int chunks_per_seg, int run_len,
    uint32_t* __restrict__ keys_out32, int64_t* __restrict__ keys_out64,
    uint32_t* __restrict__ vals_out32, int32_t* __restrict__ vals_out32_final) {
  // chunk boundaries are found with TWO global co-rank searches (thread 0),
  // then the <=512-element A/B windows stage through LDS so the per-element
  // co-ranks probe LDS, not HBM
  __shared__ uint32_t sA[kTileSize + 1], sBk[kTileSize + 1];
  __shared__ uint32_t sAv[kTileSize + 1], sBv[kTileSize + 1];
  __shared__ int bounds[4];
  int seg = blockIdx.x / chunks_per_seg;
  int chunk = blockIdx.x - seg * chunks_per_seg;
  int64_t lo = offsets[static_cast<int64_t>(seg) * B];
  int64_t hi = offsets[static_cast<int64_t>(seg + 1) * B];
  int seg_len = static_cast<int>(hi - lo);
  int out0 = chunk * kTileSize;
  if (out0 >= seg_len) return;
  int pair0 = (out0 / (2 * run_len)) * (2 * run_len);
  int la = min(run_len, seg_len - pair0);
  int lb = max(0, min(run_len, seg_len - (pair0 + run_len)));
  const uint32_t* A = keys_in + lo + pair0;
  const uint32_t* Bp = A + run_len;
  const uint32_t* Av = vals_in + lo + pair0;
  const uint32_t* Bv = Av + run_len;
  int rel0 = out0 - pair0;
  int rel1 = min(rel0 + kTileSize, la + lb);
  if (threadIdx.x == 0) {
    bounds[0] = co_rank(rel0, A, la, Bp, lb);
    bounds[1] = co_rank(rel1, A, la, Bp, lb);
  }
  __syncthreads();
  int a_lo = bounds[0], a_hi = bounds[1];
  int b_lo = rel0 - a_lo, b_hi = rel1 - a_hi;
  int wa = a_hi - a_lo, wb = b_hi - b_lo;
  for (int i = threadIdx.x; i < wa; i += kTileThreads) {
    sA[i] = A[a_lo + i];
    sAv[i] = Av[a_lo + i];
  }
  for (int i = threadIdx.x; i < wb; i += kTileThreads) {
    sBk[i] = Bp[b_lo + i];
    sBv[i] = Bv[b_lo + i];
  }
  __syncthreads();
  for (int i = threadIdx.x; i < kTileSize; i += kTileThreads) {
    if (rel0 + i >= rel1) break;
    int ai = co_rank(i, sA, wa, sBk, wb);
    int bi = i - ai;
    bool take_a = (bi >= wb) || (ai < wa && sA[ai] <= sBk[bi]);
    uint32_t kv = take_a ? sA[ai] : sBk[bi];
    uint32_t vv = take_a ? sAv[ai] : sBv[bi];
    int64_t out_idx = lo + out0 + i;
    if (OUT64) {
      keys_out64[out_idx] = static_cast<int64_t>(kv);
      vals_out32_final[out_idx] = static_cast<int32_t>(vv);
    } else {
      keys_out32[out_idx] = kv;
      vals_out32[out_idx] = vv;
    }
  }
}

__global__ void seg_cast_out_kernel(const uint32_t* __restrict__ keys,
                                    const uint32_t* __restrict__ vals, int64_t n,
                                    int64_t* __restrict__ keys64,
                                    int32_t* __restrict__ perm) {
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    keys64[i] = static_cast<int64_t>(keys[i]);
    perm[i] = static_cast<int32_t>(vals[i]);
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs_2level(
    const at::Tensor& linear, const at::Tensor& offsets, int64_t B, int64_t F,
    int64_t end_bit, int64_t capacity) {
  TORCH_CHECK(linear.is_cuda() && linear.scalar_type() == at::kLong);
  TORCH_CHECK(end_bit <= 32, "segmented sort requires 32-bit ids");
  int64_t n = linear.numel();
  auto sorted64 = at::empty_like(linear);
  auto perm = at::empty({n}, linear.options().dtype(at::kInt));
  auto overflow = at::zeros({1}, linear.options().dtype(at::kInt));
  if (n == 0 || F == 0) return {sorted64, perm, overflow};
  auto stream = tbe_stream();
  // base-tile size: bigger tiles = fewer merge rounds (each merge kernel has
  // a ~9 us latency floor), smaller tiles = more tile-sort parallelism.
  // 2048-key tiles balance at the bench shapes.
  int items = capacity >= 4096 ? 8 : 2;
  int base_tile = kTileThreads * items;
  int tiles_per_seg = (int)((capacity + base_tile - 1) / base_tile);
  auto opts32 = linear.options().dtype(at::kInt);
  auto keysA = at::empty({n}, opts32);
  auto valsA = at::empty({n}, opts32);
  if (items == 8) {
    hipLaunchKernelGGL((seg_tile_sort_kernel<8>), dim3((int)F * tiles_per_seg),
                       dim3(kTileThreads), 0, stream, linear.data_ptr<int64_t>(),
                       offsets.data_ptr<int64_t>(), (int)B, tiles_per_seg, (int)end_bit,
                       reinterpret_cast<uint32_t*>(keysA.data_ptr<int32_t>()),
                       reinterpret_cast<uint32_t*>(valsA.data_ptr<int32_t>()));
  } else {
    hipLaunchKernelGGL((seg_tile_sort_kernel<2>), dim3((int)F * tiles_per_seg),
                       dim3(kTileThreads), 0, stream, linear.data_ptr<int64_t>(),
                       offsets.data_ptr<int64_t>(), (int)B, tiles_per_seg, (int)end_bit,
                       reinterpret_cast<uint32_t*>(keysA.data_ptr<int32_t>()),
                       reinterpret_cast<uint32_t*>(valsA.data_ptr<int32_t>()));
  }
  if (tiles_per_seg == 1) {
    hipLaunchKernelGGL(seg_cast_out_kernel, dim3(grid_for(n, kBlockThreads)),
                       dim3(kBlockThreads), 0, stream,
                       reinterpret_cast<const uint32_t*>(keysA.data_ptr<int32_t>()),
                       reinterpret_cast<const uint32_t*>(valsA.data_ptr<int32_t>()), n,
                       sorted64.data_ptr<int64_t>(), perm.data_ptr<int32_t>());
    return {sorted64, perm, overflow};
  }
  auto keysB = at::empty({n}, opts32);
  auto valsB = at::empty({n}, opts32);
  // merge output chunks are kTileSize wide regardless of the base tile
  int chunks_per_seg = (int)((capacity + kTileSize - 1) / kTileSize);
  int run_len = base_tile;
  bool a_is_src = true;
  while (run_len < tiles_per_seg * base_tile) {
    bool last = (run_len * 2) >= tiles_per_seg * base_tile;
    auto& src_k = a_is_src ? keysA : keysB;
    auto& src_v = a_is_src ? valsA : valsB;
    auto& dst_k = a_is_src ? keysB : keysA;
    auto& dst_v = a_is_src ? valsB : valsA;
    if (last) {
      hipLaunchKernelGGL((seg_merge_kernel<true>), dim3((int)F * chunks_per_seg),
                         dim3(kTileThreads), 0, stream,
                         reinterpret_cast<const uint32_t*>(src_k.data_ptr<int32_t>()),
                         reinterpret_cast<const uint32_t*>(src_v.data_ptr<int32_t>()),
                         offsets.data_ptr<int64_t>(), (int)B, chunks_per_seg, run_len,
                         nullptr, sorted64.data_ptr<int64_t>(), nullptr,
                         perm.data_ptr<int32_t>());
    } else {
      hipLaunchKernelGGL((seg_merge_kernel<false>), dim3((int)F * chunks_per_seg),
                         dim3(kTileThreads), 0, stream,
                         reinterpret_cast<const uint32_t*>(src_k.data_ptr<int32_t>()),
                         reinterpret_cast<const uint32_t*>(src_v.data_ptr<int32_t>()),
                         offsets.data_ptr<int64_t>(), (int)B, chunks_per_seg, run_len,
                         reinterpret_cast<uint32_t*>(dst_k.data_ptr<int32_t>()),
                         nullptr,
                         reinterpret_cast<uint32_t*>(dst_v.data_ptr<int32_t>()),
                         nullptr);
    }
    run_len *= 2;
    a_is_src = !a_is_src;
  }
  return {sorted64, perm, overflow};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs_large(
    const at::Tensor& linear, const at::Tensor& feat_bounds /* [F+1] */, int64_t F,
    int64_t end_bit) {
  // device segmented radix sort (rocPRIM): segments of any size, ONE public
  // call — the large-batch (B > block-sort LDS tile) backward path
  TORCH_CHECK(linear.is_cuda() && linear.scalar_type() == at::kLong);
  TORCH_CHECK(end_bit <= 31, "segmented sort requires 32-bit ids");
  auto stream = tbe_stream();
  int64_t n = linear.numel();
  auto perm_in = at::arange(n, linear.options().dtype(at::kInt));
  auto perm = at::empty_like(perm_in);
  auto overflow = at::zeros({1}, linear.options().dtype(at::kInt));
  if (n == 0 || F == 0) return {at::empty_like(linear), perm, overflow};
  auto k32 = linear.to(at::kInt);
  auto k32_out = at::empty_like(k32);
  auto bounds32 = feat_bounds.to(at::kInt);
  size_t tmp_bytes = 0;
  auto err = rocprim::segmented_radix_sort_pairs(
      nullptr, tmp_bytes, k32.data_ptr<int32_t>(), k32_out.data_ptr<int32_t>(),
      perm_in.data_ptr<int32_t>(), perm.data_ptr<int32_t>(), (unsigned)n, (unsigned)F,
      bounds32.data_ptr<int32_t>(), bounds32.data_ptr<int32_t>() + 1, 0,
      (unsigned)end_bit, stream);
  TORCH_CHECK(err == hipSuccess, "segmented_radix_sort sizing failed");
  auto tmp = at::empty({static_cast<int64_t>(tmp_bytes)},
                       linear.options().dtype(at::kByte));
  err = rocprim::segmented_radix_sort_pairs(
      tmp.data_ptr(), tmp_bytes, k32.data_ptr<int32_t>(), k32_out.data_ptr<int32_t>(),
      perm_in.data_ptr<int32_t>(), perm.data_ptr<int32_t>(), (unsigned)n, (unsigned)F,
      bounds32.data_ptr<int32_t>(), bounds32.data_ptr<int32_t>() + 1, 0,
      (unsigned)end_bit, stream);
  TORCH_CHECK(err == hipSuccess, "segmented_radix_sort failed");
  return {k32_out.to(at::kLong), perm, overflow};
}

// ---------------------------------------------------------------------------
// single-kernel run detection and chunk planning (decoupled-lookback scans).
// The mark -> rocprim scan -> scatter pipeline was 10 launch-floor kernels
// (~50 us inside the captured step); each phase is now ONE kernel: compute
// the scan input inline, block-scan, decoupled lookback for the tile prefix
// (rocprim pattern: publish aggregate, spin-read predecessors, publish
// inclusive prefix), scatter the results. Tile ids come from a global
// counter so tiles start in issue order (no deadlock).
// ---------------------------------------------------------------------------

namespace {
constexpr int kScanItems = 8;
constexpr int kScanTile = kBlockThreads * kScanItems;

// single-word (status<<32 | value) protocol: the packed word is both the
// flag and the payload, so RELAXED atomics suffice (no ordering against
// other locations) and probe loads pipeline freely.
__device__ __forceinline__ void scan_publish(uint64_t* st, int tile, int32_t val,
                                             uint64_t status) {
  __hip_atomic_store(&st[tile], (status << 32) | (uint32_t)val, __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_AGENT);
}

// wave-parallel decoupled lookback (rocprim pattern): 64 predecessor tiles
// probed at once; a serial walk costs ~0.7 us per probed tile and dominated
// the kernel at >100 tiles. All 64 lanes of wave 0 must call this; every
// lane returns the exclusive prefix.
__device__ __forceinline__ int32_t scan_lookback_wave(const uint64_t* st, int tile,
                                                      int lane) {
  int32_t total = 0;
  int base = tile;  // probe tiles [base-64, base-1]
  while (base > 0) {
    int t = base - 64 + lane;
    uint64_t p = ((uint64_t)2) << 32;  // lanes below tile 0 act as prefix 0
    if (t >= 0) {
      do {
        p = __hip_atomic_load(&st[t], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      } while ((p >> 32) == 0);
    }
    uint64_t pref_mask = __ballot((p >> 32) == 2);
    // highest lane holding an inclusive prefix; cut = -1 (sum ALL aggregates,
    // keep walking) when no prefix is visible in this window
    int cut = pref_mask ? (63 - __clzll(pref_mask)) : -1;
    int32_t contrib = (lane >= cut) ? (int32_t)(uint32_t)p : 0;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      contrib += __shfl_down(contrib, off, kWaveSize);
    contrib = __shfl(contrib, 0, kWaveSize);
    total += contrib;
    if (cut >= 0) return total;  // prefix found: done
    base -= 64;
  }
  return total;
}
}  // namespace

__global__ void __launch_bounds__(kBlockThreads) prep_runs_lookback_kernel(
    const int64_t* __restrict__ sorted, int64_t n, int32_t* __restrict__ seg_offsets,
    int32_t* __restrict__ num_runs, uint64_t* __restrict__ tile_state,
    int* __restrict__ tile_counter) {
  using BlockScan = hipcub::BlockScan<int32_t, kBlockThreads>;
  __shared__ typename BlockScan::TempStorage temp;
  __shared__ int s_tile;
  __shared__ int32_t s_prefix;
  if (threadIdx.x == 0) s_tile = atomicAdd(tile_counter, 1);
  __syncthreads();
  const int tile = s_tile;
  const int64_t base = static_cast<int64_t>(tile) * kScanTile;
  int32_t flags[kScanItems];
  int32_t incl[kScanItems];
#pragma unroll
  for (int v = 0; v < kScanItems; ++v) {
    int64_t i = base + static_cast<int64_t>(threadIdx.x) * kScanItems + v;
    flags[v] = (i < n) ? ((i == 0) || (sorted[i] != sorted[i - 1])) : 0;
  }
  int32_t agg;
  BlockScan(temp).InclusiveSum(flags, incl, agg);
  __syncthreads();  // broadcast agg via BlockScan done; wave 0 runs lookback
  if (threadIdx.x < kWaveSize) {
    if (threadIdx.x == 0) scan_publish(tile_state, tile, agg, 1);
    int32_t pre = scan_lookback_wave(tile_state, tile, threadIdx.x);
    if (threadIdx.x == 0) {
      s_prefix = pre;
      scan_publish(tile_state, tile, pre + agg, 2);
    }
  }
  __syncthreads();
  const int32_t prefix = s_prefix;
#pragma unroll
  for (int v = 0; v < kScanItems; ++v) {
    int64_t i = base + static_cast<int64_t>(threadIdx.x) * kScanItems + v;
    if (i < n) {
      int32_t run_id = prefix + incl[v];  // inclusive count of run starts
      if (flags[v]) seg_offsets[run_id - 1] = static_cast<int32_t>(i);
      if (i == n - 1) {
        seg_offsets[run_id] = static_cast<int32_t>(n);
        *num_runs = run_id;
      }
    }
  }
}

std::tuple<at::Tensor, at::Tensor> tbe_backward_prep(const at::Tensor& sorted_linear) {
  int64_t n = sorted_linear.numel();
  auto opts = sorted_linear.options().dtype(at::kInt);
  auto seg_offsets = at::empty({n + 1}, opts);
  auto num_runs = at::empty({1}, opts);
  if (n == 0) return {seg_offsets, num_runs.zero_()};
  auto stream = tbe_stream();
  int64_t tiles = (n + kScanTile - 1) / kScanTile;
  // workspace: [tiles] packed lookback states + 1 tile counter, zeroed once
  auto ws = at::zeros({tiles + 1}, sorted_linear.options());  // int64 slots
  hipLaunchKernelGGL(prep_runs_lookback_kernel, dim3((int)tiles), dim3(kBlockThreads),
                     0, stream, sorted_linear.data_ptr<int64_t>(), n,
                     seg_offsets.data_ptr<int32_t>(), num_runs.data_ptr<int32_t>(),
                     reinterpret_cast<uint64_t*>(ws.data_ptr<int64_t>()),
                     reinterpret_cast<int*>(ws.data_ptr<int64_t>() + tiles));
  return {seg_offsets, num_runs};
}

// ---------------------------------------------------------------------------
// chunking of long duplicate-index runs.
//
// Hot rows (tiny tables: 3..128 rows under batch 8192 -> runs of thousands of
// occurrences) would serialize a single wave-slot. Runs longer than
// `chunk_size` are split into chunks processed by independent slots writing
// fp32 partials to scratch (phase 1), then one slot per run sums its chunks
// in order and applies the optimizer update (phase 2) — deterministic, no
// atomics. Mirrors the reference's short/long-run split (triton TBE
// :787/:1292) redesigned for wave-slot scheduling.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(kBlockThreads) prep_chunks_lookback_kernel(
    const int32_t* __restrict__ seg_offsets, const int32_t* __restrict__ num_runs_ptr,
    int64_t n, int chunk_size, int32_t* __restrict__ chunk_offsets,
    int32_t* __restrict__ total, uint64_t* __restrict__ tile_state,
    int* __restrict__ tile_counter) {
  using BlockScan = hipcub::BlockScan<int32_t, kBlockThreads>;
  __shared__ typename BlockScan::TempStorage temp;
  __shared__ int s_tile;
  __shared__ int32_t s_prefix;
  if (threadIdx.x == 0) s_tile = atomicAdd(tile_counter, 1);
  __syncthreads();
  const int tile = s_tile;
  const int64_t base = static_cast<int64_t>(tile) * kScanTile;
  const int32_t num_runs = *num_runs_ptr;
  int32_t nc[kScanItems];
  int32_t incl[kScanItems];
#pragma unroll
  for (int v = 0; v < kScanItems; ++v) {
    int64_t r = base + static_cast<int64_t>(threadIdx.x) * kScanItems + v;
    int32_t c = 0;
    if (r < num_runs) {
      int32_t len = seg_offsets[r + 1] - seg_offsets[r];
      if (len > chunk_size) c = (len + chunk_size - 1) / chunk_size;
    }
    nc[v] = c;
  }
  int32_t agg;
  BlockScan(temp).InclusiveSum(nc, incl, agg);
  __syncthreads();
  if (threadIdx.x < kWaveSize) {
    if (threadIdx.x == 0) scan_publish(tile_state, tile, agg, 1);
    int32_t pre = scan_lookback_wave(tile_state, tile, threadIdx.x);
    if (threadIdx.x == 0) {
      s_prefix = pre;
      scan_publish(tile_state, tile, pre + agg, 2);
    }
  }
  __syncthreads();
  const int32_t prefix = s_prefix;
#pragma unroll
  for (int v = 0; v < kScanItems; ++v) {
    int64_t r = base + static_cast<int64_t>(threadIdx.x) * kScanItems + v;
    if (r < n) {
      chunk_offsets[r + 1] = prefix + incl[v];
      if (r == 0) chunk_offsets[0] = 0;
      if (r == num_runs - 1) *total = prefix + incl[v];
    }
  }
}

std::tuple<at::Tensor, at::Tensor> tbe_backward_chunk_prep(const at::Tensor& seg_offsets,
                                                           const at::Tensor& num_runs,
                                                           int64_t chunk_size) {
  int64_t n = seg_offsets.numel() - 1;  // max possible runs
  auto opts = seg_offsets.options();
  auto chunk_offsets = at::empty({n + 1}, opts);
  auto total = at::empty({1}, opts);  // writer guaranteed: n>0 => num_runs>=1
  if (n == 0) return {chunk_offsets, total.zero_()};
  auto stream = tbe_stream();
  int64_t tiles = (n + kScanTile - 1) / kScanTile;
  auto ws = at::zeros({tiles + 1}, seg_offsets.options().dtype(at::kLong));
  hipLaunchKernelGGL(prep_chunks_lookback_kernel, dim3((int)tiles), dim3(kBlockThreads),
                     0, stream, seg_offsets.data_ptr<int32_t>(),
                     num_runs.data_ptr<int32_t>(), n, (int)chunk_size,
                     chunk_offsets.data_ptr<int32_t>(), total.data_ptr<int32_t>(),
                     reinterpret_cast<uint64_t*>(ws.data_ptr<int64_t>()),
                     reinterpret_cast<int*>(ws.data_ptr<int64_t>() + tiles));
  return {chunk_offsets, total};
}

__device__ __forceinline__ int upper_bound_segment_i32(const int32_t* offs, int n, int32_t x) {
  int lo = 0, hi = n;
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    if (offs[mid] <= x) lo = mid; else hi = mid;
  }
  return lo;
}

template <typename g_t, int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_bwd_long_partial_kernel(
    const g_t* __restrict__ grad,
    const int4* __restrict__ desc_a,   // per run: {k0, k1, c0, c1}
    const int* __restrict__ desc_d,    // per run: D
    const int64_t* __restrict__ grow_off,   // per sorted position: grad row offset
    const float* __restrict__ sorted_scale,  // nullable, sorted order
    const int32_t* __restrict__ num_runs_ptr,
    const int32_t* __restrict__ chunk_offsets, const int32_t* __restrict__ total_chunks_ptr,
    int chunk_size, int64_t max_D, float* __restrict__ scratch) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  int32_t total_chunks = *total_chunks_ptr;
  int32_t num_runs = *num_runs_ptr;
  for (int64_t c = slot; c < total_chunks; c += n_slots) {
    int r = upper_bound_segment_i32(chunk_offsets, num_runs, (int32_t)c);
    int4 da = desc_a[r];
    int32_t k0 = da.x + (c - da.z) * chunk_size;
    int32_t k1 = min(da.y, k0 + chunk_size);
    int D = desc_d[r];
    float4 acc[CHUNKS];
#pragma unroll
    for (int cc = 0; cc < CHUNKS; ++cc) acc[cc] = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int32_t k = k0; k < k1; ++k) {
      const g_t* grow = grad + grow_off[k];
      float s = sorted_scale ? sorted_scale[k] : 1.f;
#pragma unroll
      for (int cc = 0; cc < CHUNKS; ++cc) {
        int col4 = cc * LPS + sl;
        if (col4 * 4 < D) {
          float4 g = Vec4<g_t>::load(grow, col4);
          acc[cc].x += s * g.x;
          acc[cc].y += s * g.y;
          acc[cc].z += s * g.z;
          acc[cc].w += s * g.w;
        }
      }
    }
    float4* srow = reinterpret_cast<float4*>(scratch + c * max_D);
#pragma unroll
    for (int cc = 0; cc < CHUNKS; ++cc) {
      int col4 = cc * LPS + sl;
      if (col4 * 4 < D) srow[col4] = acc[cc];
    }
  }
}

// one streaming pass builds (a) a 36-byte descriptor per duplicate-index run
// — collapsing the fused kernel's 7-deep dependent scalar chain
// (seg_offsets -> sorted_linear -> table binary search -> chunk_offsets) to
// three independent vector loads — and (b) the grad-row offset (+ scale) per
// SORTED position, so the gather chain perm -> (row, col) -> grad row
// becomes one sequential read + one row load. Measured motivation: with
// ids made fully sequential the fused kernel only sped up 8% => it is
// latency-chain bound, not DRAM-randomness bound.
__global__ void __launch_bounds__(kBlockThreads) tbe_bwd_build_desc_kernel(
    const int32_t* __restrict__ seg_offsets, const int32_t* __restrict__ chunk_offsets,
    const int32_t* __restrict__ num_runs_ptr, const int64_t* __restrict__ sorted_linear,
    const int32_t* __restrict__ sort_perm, const int32_t* __restrict__ pos_row,
    const int64_t* __restrict__ pos_col, const float* __restrict__ pos_scale,
    int64_t grad_stride, const int64_t* __restrict__ table_row_offsets,
    const int64_t* __restrict__ table_elem_offsets, const int32_t* __restrict__ dims,
    int T, int64_t n, int4* __restrict__ desc_a, longlong2* __restrict__ desc_b,
    int* __restrict__ desc_d, int64_t* __restrict__ grow_off,
    float* __restrict__ sorted_scale) {
  int32_t num_runs = *num_runs_ptr;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int32_t p = sort_perm[i];
    grow_off[i] = static_cast<int64_t>(pos_row[p]) * grad_stride + pos_col[p];
    if (sorted_scale) sorted_scale[i] = pos_scale[p];
    if (i < num_runs) {
      int32_t k0 = seg_offsets[i], k1 = seg_offsets[i + 1];
      int64_t lin = sorted_linear[k0];
      int t = upper_bound_segment(table_row_offsets, T, lin);
      int D = dims[t];
      desc_a[i] = make_int4(k0, k1, chunk_offsets[i], chunk_offsets[i + 1]);
      longlong2 b;
      b.x = lin;
      b.y = table_elem_offsets[t] + (lin - table_row_offsets[t]) * static_cast<int64_t>(D);
      desc_b[i] = b;
      desc_d[i] = D;
    }
  }
}

// ---------------------------------------------------------------------------
// backward + fused optimizer. One wave-slot per duplicate-index run; long
// runs (chunk_offsets[r+1] > chunk_offsets[r]) sum pre-computed chunk
// partials instead of walking occurrences.
// mode: 0 = SGD, 1 = rowwise Adagrad, 2 = dense grad (write grad_weights).
// ---------------------------------------------------------------------------

template <typename emb_t, typename g_t, int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_bwd_fused_kernel(
    emb_t* __restrict__ weights, float* __restrict__ momentum,
    const g_t* __restrict__ grad,
    const int4* __restrict__ desc_a,        // per run: {k0, k1, c0, c1}
    const longlong2* __restrict__ desc_b,   // per run: {lin, elem_base}
    const int* __restrict__ desc_d,         // per run: D
    const int64_t* __restrict__ grow_off,   // per sorted position
    const float* __restrict__ sorted_scale, // nullable, sorted order
    const int32_t* __restrict__ sort_perm,  // cache-locator path only
    const int32_t* __restrict__ num_runs_ptr, const float* __restrict__ scratch,
    int64_t max_D, float lr, float eps, int mode,
    emb_t* __restrict__ grad_weights, float* __restrict__ cache_weights,
    const int32_t* __restrict__ cache_loc, int64_t cache_stride,
    float* __restrict__ m1, float* __restrict__ m2, float beta1, float beta2,
    const float* __restrict__ iter_ptr, const int64_t* __restrict__ rng_state,
    int stochastic) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  int32_t num_runs = *num_runs_ptr;
  const uint64_t rng_base = rng_state ? (uint64_t)*rng_state : 0;
  for (int64_t r = slot; r < num_runs; r += n_slots) {
    int4 da = desc_a[r];
    longlong2 dbv = desc_b[r];
    int32_t k0 = da.x, k1 = da.y;
    int64_t lin = dbv.x;
    int D = desc_d[r];
    float4 acc[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) acc[c] = make_float4(0.f, 0.f, 0.f, 0.f);
    if (da.w > da.z) {
      // long run: sum phase-1 chunk partials in order (deterministic)
      for (int32_t c = da.z; c < da.w; ++c) {
        const float4* srow = reinterpret_cast<const float4*>(scratch + c * max_D);
#pragma unroll
        for (int cc = 0; cc < CHUNKS; ++cc) {
          int col4 = cc * LPS + sl;
          if (col4 * 4 < D) {
            float4 g = srow[col4];
            acc[cc].x += g.x;
            acc[cc].y += g.y;
            acc[cc].z += g.z;
            acc[cc].w += g.w;
          }
        }
      }
    } else {
      for (int32_t k = k0; k < k1; ++k) {
        const g_t* grow = grad + grow_off[k];
        float s = sorted_scale ? sorted_scale[k] : 1.f;
#pragma unroll
        for (int c = 0; c < CHUNKS; ++c) {
          int col4 = c * LPS + sl;
          if (col4 * 4 < D) {
            float4 g = Vec4<g_t>::load(grow, col4);
            acc[c].x += s * g.x;
            acc[c].y += s * g.y;
            acc[c].z += s * g.z;
            acc[c].w += s * g.w;
          }
        }
      }
    }
    // weight row: emb_t in the table, fp32 if the row sits in the lxu cache
    emb_t* wrow = weights + dbv.y;
    float* crow = nullptr;
    int32_t cloc = cache_loc ? cache_loc[sort_perm[k0]] : -1;
    if (cloc >= 0) crow = cache_weights + static_cast<int64_t>(cloc) * cache_stride;
    if (mode == 1) {
      // rowwise Adagrad: m += mean(g^2); w -= lr * g / (sqrt(m) + eps)
      float gsq = 0.f;
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D)
          gsq += acc[c].x * acc[c].x + acc[c].y * acc[c].y + acc[c].z * acc[c].z +
                 acc[c].w * acc[c].w;
      }
      gsq = group_reduce_sum<LPS>(gsq);
      float m = momentum[lin] + gsq / static_cast<float>(D);
      if (sl == 0) momentum[lin] = m;
      float step = lr / (sqrtf(m) + eps);
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          float4 w = crow ? Vec4<float>::load(crow, col4) : Vec4<emb_t>::load(wrow, col4);
          w.x -= step * acc[c].x;
          w.y -= step * acc[c].y;
          w.z -= step * acc[c].z;
          w.w -= step * acc[c].w;
          if (crow) Vec4<float>::store(crow, col4, w);
          else store4_maybe_stoch(wrow, col4, w, rng_base ^ (uint64_t)(lin * 1024 + col4),
                                  stochastic != 0);
        }
      }
    } else if (mode == 0) {
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          float4 w = crow ? Vec4<float>::load(crow, col4) : Vec4<emb_t>::load(wrow, col4);
          w.x -= lr * acc[c].x;
          w.y -= lr * acc[c].y;
          w.z -= lr * acc[c].z;
          w.w -= lr * acc[c].w;
          if (crow) Vec4<float>::store(crow, col4, w);
          else store4_maybe_stoch(wrow, col4, w, rng_base ^ (uint64_t)(lin * 1024 + col4),
                                  stochastic != 0);
        }
      }
    } else if (mode == 3 || mode == 4) {
      // Adam (mode 3: elementwise m2) / partial-rowwise Adam (mode 4: one m2
      // scalar per row = beta2-EMA of mean(g^2)). Reference semantics: TBE
      // fused optimizers (batched_embedding_kernel.py:36-53 imports).
      float t_iter = iter_ptr ? *iter_ptr : 1.f;
      float bc1 = 1.f / (1.f - __powf(beta1, t_iter));
      float bc2 = 1.f / (1.f - __powf(beta2, t_iter));
      int64_t ebase = dbv.y;
      float inv_sqrt_row = 0.f;
      if (mode == 4) {
        float gsq = 0.f;
#pragma unroll
        for (int c = 0; c < CHUNKS; ++c) {
          int col4 = c * LPS + sl;
          if (col4 * 4 < D)
            gsq += acc[c].x * acc[c].x + acc[c].y * acc[c].y + acc[c].z * acc[c].z +
                   acc[c].w * acc[c].w;
        }
        gsq = group_reduce_sum<LPS>(gsq);
        float v = beta2 * m2[lin] + (1.f - beta2) * gsq / static_cast<float>(D);
        if (sl == 0) m2[lin] = v;
        inv_sqrt_row = 1.f / (sqrtf(v * bc2) + eps);
      }
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          float4* m1p = reinterpret_cast<float4*>(m1 + ebase);
          float4 mv = m1p[col4];
          mv.x = beta1 * mv.x + (1.f - beta1) * acc[c].x;
          mv.y = beta1 * mv.y + (1.f - beta1) * acc[c].y;
          mv.z = beta1 * mv.z + (1.f - beta1) * acc[c].z;
          mv.w = beta1 * mv.w + (1.f - beta1) * acc[c].w;
          m1p[col4] = mv;
          float4 upd;
          if (mode == 3) {
            float4* m2p = reinterpret_cast<float4*>(m2 + ebase);
            float4 vv = m2p[col4];
            vv.x = beta2 * vv.x + (1.f - beta2) * acc[c].x * acc[c].x;
            vv.y = beta2 * vv.y + (1.f - beta2) * acc[c].y * acc[c].y;
            vv.z = beta2 * vv.z + (1.f - beta2) * acc[c].z * acc[c].z;
            vv.w = beta2 * vv.w + (1.f - beta2) * acc[c].w * acc[c].w;
            m2p[col4] = vv;
            upd.x = lr * mv.x * bc1 / (sqrtf(vv.x * bc2) + eps);
            upd.y = lr * mv.y * bc1 / (sqrtf(vv.y * bc2) + eps);
            upd.z = lr * mv.z * bc1 / (sqrtf(vv.z * bc2) + eps);
            upd.w = lr * mv.w * bc1 / (sqrtf(vv.w * bc2) + eps);
          } else {
            upd.x = lr * mv.x * bc1 * inv_sqrt_row;
            upd.y = lr * mv.y * bc1 * inv_sqrt_row;
            upd.z = lr * mv.z * bc1 * inv_sqrt_row;
            upd.w = lr * mv.w * bc1 * inv_sqrt_row;
          }
          float4 w = crow ? Vec4<float>::load(crow, col4) : Vec4<emb_t>::load(wrow, col4);
          w.x -= upd.x;
          w.y -= upd.y;
          w.z -= upd.z;
          w.w -= upd.w;
          if (crow) Vec4<float>::store(crow, col4, w);
          else store4_maybe_stoch(wrow, col4, w, rng_base ^ (uint64_t)(lin * 1024 + col4),
                                  stochastic != 0);
        }
      }
    } else {
      emb_t* gw = grad_weights + dbv.y;
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) Vec4<emb_t>::store(gw, col4, acc[c]);
      }
    }
  }
}

template <typename emb_t, typename g_t, typename host_w_t>
static void launch_tbe_bwd(
    at::Tensor& weights, at::Tensor& momentum, const at::Tensor& grad,
    const at::Tensor& sorted_linear, const at::Tensor& sort_perm,
    const at::Tensor& seg_offsets, const at::Tensor& num_runs,
    const at::Tensor& chunk_offsets, const at::Tensor& total_chunks,
    const at::Tensor& scratch, const at::Tensor& pos_row, const at::Tensor& pos_col,
    const float* scale_ptr, const at::Tensor& table_row_offsets,
    const at::Tensor& table_elem_offsets, const at::Tensor& dims, int T,
    int chunk_size, int64_t max_D, float lr, float eps, int mode,
    at::Tensor& grad_weights, float* cache_w_ptr, const int32_t* cache_loc_ptr,
    float* m1_ptr, float* m2_ptr, float beta1, float beta2,
    const float* iter_ptr, const int64_t* rng_ptr, int stochastic,
    int lps, int chunks, int grid, int grid_long, hipStream_t stream) {
  const g_t* grad_ptr = reinterpret_cast<const g_t*>(grad.data_ptr());
  emb_t* gw_ptr = grad_weights.numel() > 0
      ? reinterpret_cast<emb_t*>(grad_weights.data_ptr<host_w_t>()) : nullptr;
  // one streaming pass collapses the per-run metadata chain + per-position
  // grad gather chain (see tbe_bwd_build_desc_kernel)
  int64_t n = sorted_linear.numel();
  auto opts_i = sorted_linear.options().dtype(at::kInt);
  auto desc_a_t = at::empty({n * 4}, opts_i);
  auto desc_b_t = at::empty({n * 2}, sorted_linear.options());
  auto desc_d_t = at::empty({n}, opts_i);
  auto grow_off_t = at::empty({n}, sorted_linear.options());
  at::Tensor sorted_scale_t;
  float* sscale_ptr = nullptr;
  if (scale_ptr) {
    sorted_scale_t = at::empty({n}, grad.options().dtype(at::kFloat));
    sscale_ptr = sorted_scale_t.data_ptr<float>();
  }
  int4* desc_a_p = reinterpret_cast<int4*>(desc_a_t.data_ptr<int32_t>());
  longlong2* desc_b_p = reinterpret_cast<longlong2*>(desc_b_t.data_ptr<int64_t>());
  hipLaunchKernelGGL(tbe_bwd_build_desc_kernel, dim3(grid_for(n, kBlockThreads)),
                     dim3(kBlockThreads), 0, stream,
                     seg_offsets.data_ptr<int32_t>(), chunk_offsets.data_ptr<int32_t>(),
                     num_runs.data_ptr<int32_t>(), sorted_linear.data_ptr<int64_t>(),
                     sort_perm.data_ptr<int32_t>(), pos_row.data_ptr<int32_t>(),
                     pos_col.data_ptr<int64_t>(), scale_ptr, grad.size(1),
                     table_row_offsets.data_ptr<int64_t>(),
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(), T,
                     n, desc_a_p, desc_b_p, desc_d_t.data_ptr<int32_t>(),
                     grow_off_t.data_ptr<int64_t>(), sscale_ptr);
#define TBE_BWD_LAUNCH(LPS, CHUNKS)                                                          \
  do {                                                                                       \
    hipLaunchKernelGGL((tbe_bwd_long_partial_kernel<g_t, LPS, CHUNKS>), dim3(grid_long),     \
                       dim3(kBlockThreads), 0, stream, grad_ptr, desc_a_p,                   \
                       desc_d_t.data_ptr<int32_t>(), grow_off_t.data_ptr<int64_t>(),         \
                       sscale_ptr, num_runs.data_ptr<int32_t>(),                             \
                       chunk_offsets.data_ptr<int32_t>(), total_chunks.data_ptr<int32_t>(),  \
                       chunk_size, max_D, scratch.data_ptr<float>());                        \
    hipLaunchKernelGGL((tbe_bwd_fused_kernel<emb_t, g_t, LPS, CHUNKS>), dim3(grid),          \
                       dim3(kBlockThreads),                                                  \
                       0, stream, reinterpret_cast<emb_t*>(uvm_ptr<host_w_t>(weights)),      \
                       uvm_ptr<float>(momentum),                                             \
                       grad_ptr, desc_a_p, desc_b_p, desc_d_t.data_ptr<int32_t>(),           \
                       grow_off_t.data_ptr<int64_t>(), sscale_ptr,                           \
                       sort_perm.data_ptr<int32_t>(),                                        \
                       num_runs.data_ptr<int32_t>(), scratch.data_ptr<float>(), max_D,       \
                       lr, eps, mode, gw_ptr, cache_w_ptr,                                   \
                       cache_loc_ptr, max_D, m1_ptr, m2_ptr, beta1, beta2, iter_ptr,         \
                       rng_ptr, stochastic);                                                 \
  } while (0)
  if (lps == 16) TBE_BWD_LAUNCH(16, 1);
  else if (lps == 32) TBE_BWD_LAUNCH(32, 1);
  else switch (chunks) {
    case 1: TBE_BWD_LAUNCH(64, 1); break;
    case 2: TBE_BWD_LAUNCH(64, 2); break;
    case 3: case 4: TBE_BWD_LAUNCH(64, 4); break;
    default: TBE_BWD_LAUNCH(64, 8); break;
  }
#undef TBE_BWD_LAUNCH
}

template <typename emb_t, typename host_w_t>
static void dispatch_tbe_bwd_grad(const at::ScalarType g_st, at::Tensor& weights,
    at::Tensor& momentum, const at::Tensor& grad, const at::Tensor& sorted_linear,
    const at::Tensor& sort_perm, const at::Tensor& seg_offsets, const at::Tensor& num_runs,
    const at::Tensor& chunk_offsets, const at::Tensor& total_chunks,
    const at::Tensor& scratch, const at::Tensor& pos_row, const at::Tensor& pos_col,
    const float* scale_ptr, const at::Tensor& table_row_offsets,
    const at::Tensor& table_elem_offsets, const at::Tensor& dims, int T, int chunk_size,
    int64_t max_D, float lr, float eps, int mode, at::Tensor& grad_weights,
    float* cache_w_ptr, const int32_t* cache_loc_ptr,
    float* m1_ptr, float* m2_ptr, float beta1, float beta2, const float* iter_ptr,
    const int64_t* rng_ptr, int stochastic, int lps, int chunks, int grid,
    int grid_long, hipStream_t stream) {
#define ARGS weights, momentum, grad, sorted_linear, sort_perm, seg_offsets, num_runs,  \
    chunk_offsets, total_chunks, scratch, pos_row, pos_col, scale_ptr,                  \
    table_row_offsets, table_elem_offsets, dims, T, chunk_size, max_D, lr, eps, mode,   \
    grad_weights, cache_w_ptr, cache_loc_ptr, m1_ptr, m2_ptr, beta1, beta2, iter_ptr,   \
    rng_ptr, stochastic, lps, chunks, grid, grid_long, stream
  if (g_st == at::kFloat) launch_tbe_bwd<emb_t, float, host_w_t>(ARGS);
  else if (g_st == at::kBFloat16) launch_tbe_bwd<emb_t, __hip_bfloat16, host_w_t>(ARGS);
  else if (g_st == at::kHalf) launch_tbe_bwd<emb_t, __half, host_w_t>(ARGS);
  else TORCH_CHECK(false, "unsupported TBE gradient dtype");
#undef ARGS
}

void tbe_backward_fused(
    at::Tensor weights, at::Tensor momentum, const at::Tensor& grad,
    const at::Tensor& sorted_linear, const at::Tensor& sort_perm,
    const at::Tensor& seg_offsets, const at::Tensor& num_runs, const at::Tensor& pos_row,
    const at::Tensor& pos_col, const at::Tensor& pos_scale,
    const at::Tensor& table_row_offsets, const at::Tensor& table_elem_offsets,
    const at::Tensor& dims, int64_t max_D, double lr, double eps, int64_t mode,
    at::Tensor grad_weights, at::Tensor cache_weights, const at::Tensor& cache_loc,
    at::Tensor m1, at::Tensor m2, double beta1, double beta2,
    const at::Tensor& iter_t, const at::Tensor& rng_state, bool stochastic) {
  int64_t n = sorted_linear.numel();
  if (n == 0) return;
  int T = table_elem_offsets.numel();
  auto stream = tbe_stream();
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  // full grid (one slot per run): the kernel is latency-chain bound, so a
  // grid-strided slot serializing ~3 runs costs ~3 dependent-chain lengths;
  // excess blocks just queue behind retiring ones (TREC_BWD_GRID_CAP to cap)
  int64_t grid_cap = 131072;
  if (const char* gc = std::getenv("TREC_BWD_GRID_CAP")) grid_cap = std::atoll(gc);
  int grid = (int)std::min<int64_t>(
      (n * lps + kBlockThreads - 1) / kBlockThreads, std::max<int64_t>(grid_cap, 1));
  const float* scale_ptr = pos_scale.numel() > 0 ? pos_scale.data_ptr<float>() : nullptr;
  float* cache_w_ptr = cache_weights.numel() > 0 ? cache_weights.data_ptr<float>() : nullptr;
  const int32_t* cache_loc_ptr =
      cache_loc.numel() > 0 ? cache_loc.data_ptr<int32_t>() : nullptr;

  // chunk long duplicate-runs: partials scratch sized by the host upper bound
  constexpr int kChunkSize = 32;
  auto [chunk_offsets, total_chunks] = tbe_backward_chunk_prep(seg_offsets, num_runs, kChunkSize);
  int64_t max_chunks = 2 * ((n + kChunkSize - 1) / kChunkSize) + 2;
  auto scratch = at::empty({max_chunks * max_D}, grad.options().dtype(at::kFloat));
  int grid_long = grid_for(max_chunks * lps, kBlockThreads);

  float* m1_ptr = m1.numel() > 0 ? uvm_ptr<float>(m1) : nullptr;
  float* m2_ptr = m2.numel() > 0 ? uvm_ptr<float>(m2) : nullptr;
  const float* iter_ptr = iter_t.numel() > 0 ? iter_t.data_ptr<float>() : nullptr;
  const int64_t* rng_ptr =
      rng_state.numel() > 0 ? rng_state.data_ptr<int64_t>() : nullptr;
#define BARGS weights, momentum, grad, sorted_linear, sort_perm, seg_offsets, num_runs, \
    chunk_offsets, total_chunks, scratch, pos_row, pos_col, scale_ptr,                  \
    table_row_offsets, table_elem_offsets, dims, T, kChunkSize, max_D, (float)lr,       \
    (float)eps, (int)mode, grad_weights, cache_w_ptr, cache_loc_ptr, m1_ptr, m2_ptr,    \
    (float)beta1, (float)beta2, iter_ptr, rng_ptr, stochastic ? 1 : 0, lps, chunks,     \
    grid, grid_long, stream
  auto g_st = grad.scalar_type();
  auto w_st = weights.scalar_type();
  if (w_st == at::kFloat) dispatch_tbe_bwd_grad<float, float>(g_st, BARGS);
  else if (w_st == at::kBFloat16)
    dispatch_tbe_bwd_grad<__hip_bfloat16, at::BFloat16>(g_st, BARGS);
  else if (w_st == at::kHalf) dispatch_tbe_bwd_grad<__half, at::Half>(g_st, BARGS);
  else TORCH_CHECK(false, "unsupported TBE weights dtype");
#undef BARGS
}

__global__ void gather_run_heads_kernel(const int64_t* __restrict__ sorted_lin,
                                        const int32_t* __restrict__ seg_offsets,
                                        const int32_t* __restrict__ num_runs_ptr, int64_t n,
                                        int64_t* __restrict__ out /* [n], -1 padded */) {
  int32_t num_runs = *num_runs_ptr;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    out[i] = (i < num_runs) ? sorted_lin[seg_offsets[i]] : -1;
  }
}

at::Tensor gather_run_heads(const at::Tensor& sorted_linear, const at::Tensor& seg_offsets,
                            const at::Tensor& num_runs) {
  int64_t n = sorted_linear.numel();
  auto out = at::empty({n}, sorted_linear.options());
  if (n == 0) return out;
  hipLaunchKernelGGL(gather_run_heads_kernel, dim3(grid_for(n, kBlockThreads)),
                     dim3(kBlockThreads), 0, tbe_stream(), sorted_linear.data_ptr<int64_t>(),
                     seg_offsets.data_ptr<int32_t>(), num_runs.data_ptr<int32_t>(), n,
                     out.data_ptr<int64_t>());
  return out;
}

// ---------------------------------------------------------------------------
// grad wrt per-sample weights: dL/dw_i = dot(grad_out_bag, W[idx_i])
// ---------------------------------------------------------------------------

template <typename emb_t, int LPS>
__global__ void __launch_bounds__(kBlockThreads) tbe_grad_psw_kernel(
    const emb_t* __restrict__ weights, const int64_t* __restrict__ table_elem_offsets,
    const int32_t* __restrict__ dims, const float* __restrict__ grad, int64_t grad_stride,
    const int64_t* __restrict__ indices, const int32_t* __restrict__ pos_row,
    const int64_t* __restrict__ pos_col, const int32_t* __restrict__ pos_table, int64_t N,
    float* __restrict__ grad_psw) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  for (int64_t p = slot; p < N; p += n_slots) {
    int t = pos_table[p];
    int D = dims[t];
    const emb_t* row = weights + table_elem_offsets[t] + indices[p] * static_cast<int64_t>(D);
    const float* grow = grad + static_cast<int64_t>(pos_row[p]) * grad_stride + pos_col[p];
    float acc = 0.f;
    for (int d = sl; d < D; d += LPS)
      acc += emb2float(row[d]) * grow[d];
    acc = group_reduce_sum<LPS>(acc);
    if (sl == 0) grad_psw[p] = acc;
  }
}

at::Tensor tbe_grad_per_sample_weights(
    const at::Tensor& weights, const at::Tensor& table_elem_offsets, const at::Tensor& dims,
    const at::Tensor& grad, const at::Tensor& indices, const at::Tensor& pos_row,
    const at::Tensor& pos_col, const at::Tensor& pos_table, int64_t max_D) {
  int64_t N = indices.numel();
  auto out = at::empty({N}, indices.options().dtype(at::kFloat));
  if (N == 0) return out;
  auto stream = tbe_stream();
  int grid = grid_for(N * 16, kBlockThreads);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, weights.scalar_type(),
                                  "tbe_grad_psw", [&] {
    using dev_t = typename DevType<scalar_t>::type;
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 embedding tables unsupported");
    } else {
    hipLaunchKernelGGL((tbe_grad_psw_kernel<dev_t, 16>), dim3(grid), dim3(kBlockThreads), 0,
                       stream, reinterpret_cast<const dev_t*>(uvm_ptr<scalar_t>(weights)),
                       table_elem_offsets.data_ptr<int64_t>(),
                       dims.data_ptr<int32_t>(), grad.data_ptr<float>(), grad.size(1),
                       indices.data_ptr<int64_t>(), pos_row.data_ptr<int32_t>(),
                       pos_col.data_ptr<int64_t>(), pos_table.data_ptr<int32_t>(), N,
                       out.data_ptr<float>());
    }
  });
  return out;
}

// ---------------------------------------------------------------------------
// bounds check: clamp invalid ids in-place, count violations.
// ---------------------------------------------------------------------------

__global__ void bounds_check_kernel(int64_t* __restrict__ indices,
                                    const int64_t* __restrict__ feat_val_offsets,
                                    const int64_t* __restrict__ rows,
                                    const int32_t* __restrict__ feat_table, int F, int64_t N,
                                    int32_t* __restrict__ warnings) {
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < N;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int f = upper_bound_segment(feat_val_offsets, F, i);
    int64_t r = rows[feat_table[f]];
    int64_t idx = indices[i];
    if (idx < 0 || idx >= r) {
      atomicAdd(warnings, 1);
      indices[i] = idx < 0 ? 0 : r - 1;
    }
  }
}

at::Tensor bounds_check_indices(at::Tensor indices, const at::Tensor& feat_val_offsets,
                                const at::Tensor& rows, const at::Tensor& feat_table) {
  int64_t N = indices.numel();
  auto warnings = at::zeros({1}, indices.options().dtype(at::kInt));
  if (N == 0) return warnings;
  hipLaunchKernelGGL(bounds_check_kernel, dim3(grid_for(N, kBlockThreads)),
                     dim3(kBlockThreads), 0, tbe_stream(), indices.data_ptr<int64_t>(),
                     feat_val_offsets.data_ptr<int64_t>(), rows.data_ptr<int64_t>(),
                     feat_table.data_ptr<int32_t>(), (int)feat_table.numel(), N,
                     warnings.data_ptr<int32_t>());
  return warnings;
}

}  // namespace trec_amd
