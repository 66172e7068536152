// Jagged / KJT ops for MI355X (gfx950) — native replacements for the
// reference's fbgemm jagged op surface (SURVEY.md §2.6 worklist):
// complete_cumsum, lengths_range, permute_2d_sparse_data,
// jagged_to_padded_dense / dense_to_jagged, segment_sum_csr,
// block_bucketize_sparse_features, permute_pooled_embs.
//
// All memory-bound: vectorized loads, grid-stride loops, capped grids
// (cdna_hip_programming.md Guideline 11/13).

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hipcub/hipcub.hpp>

#include "common.h"

namespace trec_amd {

static inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// ---------------------------------------------------------------------------
// complete_cumsum: out[0]=0, out[i+1]=sum(in[0..i])  (int32/int64)
// ---------------------------------------------------------------------------

at::Tensor complete_cumsum(const at::Tensor& lengths) {
  TORCH_CHECK(lengths.is_cuda() && lengths.dim() == 1);
  auto in = lengths.contiguous();
  int64_t n = in.numel();
  auto out = at::empty({n + 1}, in.options());
  auto stream = cur_stream();

  AT_DISPATCH_INDEX_TYPES(in.scalar_type(), "complete_cumsum", [&] {
    // out[1:] = inclusive_scan(in); out[0] = 0
    TREC_HIP_CHECK(hipMemsetAsync(out.data_ptr(), 0, sizeof(index_t), stream));
    if (n == 0) return;
    size_t tmp_bytes = 0;
    hipcub::DeviceScan::InclusiveSum(
        nullptr, tmp_bytes, in.data_ptr<index_t>(), out.data_ptr<index_t>() + 1, n, stream);
    auto tmp = at::empty({static_cast<int64_t>(tmp_bytes)},
                         in.options().dtype(at::kByte));
    hipcub::DeviceScan::InclusiveSum(
        tmp.data_ptr(), tmp_bytes, in.data_ptr<index_t>(), out.data_ptr<index_t>() + 1, n,
        stream);
  });
  return out;
}

// ---------------------------------------------------------------------------
// lengths_range: per segment emit [0..L_i)
// ---------------------------------------------------------------------------

template <typename index_t>
__global__ void lengths_range_kernel(
    const index_t* __restrict__ offsets, int n_seg, index_t total, index_t* __restrict__ out) {
  // one wave per segment, grid-stride
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  int l = lane_id();
  for (int64_t s = wave; s < n_seg; s += n_waves) {
    index_t start = offsets[s];
    index_t len = offsets[s + 1] - start;
    for (index_t i = l; i < len; i += kWaveSize) out[start + i] = i;
  }
}

at::Tensor lengths_range(const at::Tensor& offsets) {
  TORCH_CHECK(offsets.is_cuda() && offsets.dim() == 1 && offsets.numel() >= 1);
  auto offs = offsets.contiguous();
  int n_seg = offs.numel() - 1;
  int64_t total = offs[-1].item<int64_t>();  // sync: host needs output size anyway
  auto out = at::empty({total}, offs.options());
  if (total == 0 || n_seg == 0) return out;
  AT_DISPATCH_INDEX_TYPES(offs.scalar_type(), "lengths_range", [&] {
    hipLaunchKernelGGL(lengths_range_kernel<index_t>, dim3(grid_for(n_seg * kWaveSize, kBlockThreads)),
                       dim3(kBlockThreads), 0, cur_stream(), offs.data_ptr<index_t>(), n_seg,
                       static_cast<index_t>(total), out.data_ptr<index_t>());
  });
  return out;
}

// ---------------------------------------------------------------------------
// permute_2d_sparse_data: permute feature-major KJT storage by feature.
// One wave per OUTPUT bag; lanes copy the bag's values (8 B int64 coalesced).
// ---------------------------------------------------------------------------

template <typename val_t, bool HAS_W>
__global__ void permute_bags_kernel(
    const int64_t* __restrict__ permute,  // [K_out]
    const int64_t* __restrict__ in_offsets,   // [K_in*B+1]
    const int64_t* __restrict__ out_offsets,  // [K_out*B+1]
    const val_t* __restrict__ values,
    const float* __restrict__ weights,
    int64_t n_out_bags, int B,
    val_t* __restrict__ out_values,
    float* __restrict__ out_weights) {
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  int l = lane_id();
  for (int64_t bag = wave; bag < n_out_bags; bag += n_waves) {
    int64_t fo = bag / B;
    int64_t b = bag - fo * B;
    int64_t fi = permute[fo];
    int64_t src = in_offsets[fi * B + b];
    int64_t dst = out_offsets[bag];
    int64_t len = out_offsets[bag + 1] - dst;
    for (int64_t i = l; i < len; i += kWaveSize) {
      out_values[dst + i] = values[src + i];
      if (HAS_W) out_weights[dst + i] = weights[src + i];
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> permute_2d_sparse_data(
    const at::Tensor& permute, const at::Tensor& lengths, const at::Tensor& values,
    const at::Tensor& weights, int64_t out_size_hint) {
  TORCH_CHECK(values.is_cuda() && lengths.dim() == 2);
  int64_t K_in = lengths.size(0), B = lengths.size(1);
  auto perm = permute.to(at::kLong).contiguous();
  int64_t K_out = perm.numel();
  auto perm_lengths = lengths.index_select(0, perm).contiguous();
  auto in_offsets = complete_cumsum(lengths.reshape({-1}).to(at::kLong).contiguous());
  auto out_offsets = complete_cumsum(perm_lengths.reshape({-1}).to(at::kLong).contiguous());
  bool has_w = weights.numel() > 0;
  int64_t n_bags = K_out * B;
  // output size: caller-provided hint avoids a D2H sync; -1 -> read tail
  int64_t out_n = out_size_hint >= 0 ? out_size_hint : out_offsets[-1].item<int64_t>();
  auto out_values = at::empty({out_n}, values.options());
  auto out_weights = has_w ? at::empty({out_n}, weights.options()) : at::empty({0}, values.options().dtype(at::kFloat));
  if (n_bags > 0 && out_n > 0) {
    auto stream = cur_stream();
    int grid = grid_for(n_bags * kWaveSize, kBlockThreads);
    AT_DISPATCH_INDEX_TYPES(values.scalar_type(), "permute_bags", [&] {
      if (has_w) {
        hipLaunchKernelGGL((permute_bags_kernel<index_t, true>), dim3(grid), dim3(kBlockThreads), 0,
                           stream, perm.data_ptr<int64_t>(), in_offsets.data_ptr<int64_t>(),
                           out_offsets.data_ptr<int64_t>(), values.data_ptr<index_t>(),
                           weights.data_ptr<float>(), n_bags, (int)B,
                           out_values.data_ptr<index_t>(), out_weights.data_ptr<float>());
      } else {
        hipLaunchKernelGGL((permute_bags_kernel<index_t, false>), dim3(grid), dim3(kBlockThreads), 0,
                           stream, perm.data_ptr<int64_t>(), in_offsets.data_ptr<int64_t>(),
                           out_offsets.data_ptr<int64_t>(), values.data_ptr<index_t>(), nullptr,
                           n_bags, (int)B, out_values.data_ptr<index_t>(), nullptr);
      }
    });
  }
  return {perm_lengths, out_values, out_weights};
}

// ---------------------------------------------------------------------------
// jagged_to_padded_dense / dense_to_jagged ([sum_L, D] <-> [B, N, D] fp32)
// ---------------------------------------------------------------------------

template <typename scalar_t>
__global__ void jagged_to_padded_kernel(
    const scalar_t* __restrict__ values, const int64_t* __restrict__ offsets, int B, int N,
    int D, scalar_t pad, scalar_t* __restrict__ out) {
  // one block per sample row; threads cover N*D
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    int64_t start = offsets[b];
    int64_t len = offsets[b + 1] - start;
    int64_t nd = static_cast<int64_t>(N) * D;
    scalar_t* orow = out + static_cast<int64_t>(b) * nd;
    const scalar_t* vrow = values + start * D;
    for (int64_t t = threadIdx.x; t < nd; t += blockDim.x) {
      int64_t l = t / D;
      orow[t] = (l < len) ? vrow[t] : pad;
    }
  }
}

at::Tensor jagged_to_padded_dense(const at::Tensor& values, const at::Tensor& offsets,
                                  int64_t max_length, double padding_value) {
  TORCH_CHECK(values.is_cuda() && values.dim() == 2);
  auto offs = offsets.to(at::kLong).contiguous();
  int B = offs.numel() - 1;
  int D = values.size(1);
  auto out = at::empty({B, max_length, D}, values.options());
  if (B == 0) return out;
  auto v = values.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, values.scalar_type(),
                                  "jagged_to_padded", [&] {
    hipLaunchKernelGGL(jagged_to_padded_kernel<scalar_t>, dim3(std::min<int>(B, kMaxBlocks)),
                       dim3(kBlockThreads), 0, cur_stream(), v.data_ptr<scalar_t>(),
                       offs.data_ptr<int64_t>(), B, (int)max_length, D,
                       static_cast<scalar_t>(padding_value), out.data_ptr<scalar_t>());
  });
  return out;
}

template <typename scalar_t>
__global__ void dense_to_jagged_kernel(
    const scalar_t* __restrict__ dense, const int64_t* __restrict__ offsets, int B, int N,
    int D, scalar_t* __restrict__ out) {
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    int64_t start = offsets[b];
    int64_t len = offsets[b + 1] - start;
    if (len > N) len = N;
    const scalar_t* drow = dense + static_cast<int64_t>(b) * N * D;
    scalar_t* orow = out + start * D;
    int64_t nd = len * D;
    for (int64_t t = threadIdx.x; t < nd; t += blockDim.x) orow[t] = drow[t];
  }
}

at::Tensor dense_to_jagged(const at::Tensor& dense, const at::Tensor& offsets) {
  TORCH_CHECK(dense.is_cuda() && dense.dim() == 3);
  auto offs = offsets.to(at::kLong).contiguous();
  int B = offs.numel() - 1;
  int N = dense.size(1), D = dense.size(2);
  int64_t total = offs[-1].item<int64_t>();
  auto out = at::empty({total, D}, dense.options());
  if (B == 0 || total == 0) return out;
  auto d = dense.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dense.scalar_type(),
                                  "dense_to_jagged", [&] {
    hipLaunchKernelGGL(dense_to_jagged_kernel<scalar_t>, dim3(std::min<int>(B, kMaxBlocks)),
                       dim3(kBlockThreads), 0, cur_stream(), d.data_ptr<scalar_t>(),
                       offs.data_ptr<int64_t>(), B, N, D, out.data_ptr<scalar_t>());
  });
  return out;
}

// ---------------------------------------------------------------------------
// segment_sum_csr
// ---------------------------------------------------------------------------

template <typename scalar_t>
__global__ void segment_sum_csr_kernel(const int64_t* __restrict__ csr, int n_seg,
                                       const scalar_t* __restrict__ values,
                                       scalar_t* __restrict__ out) {
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  int l = lane_id();
  for (int64_t s = wave; s < n_seg; s += n_waves) {
    int64_t start = csr[s], end = csr[s + 1];
    float acc = 0.f;
    for (int64_t i = start + l; i < end; i += kWaveSize) acc += static_cast<float>(values[i]);
    acc = wave_reduce_sum(acc);
    if (l == 0) out[s] = static_cast<scalar_t>(acc);
  }
}

at::Tensor segment_sum_csr(const at::Tensor& csr, const at::Tensor& values) {
  auto c = csr.to(at::kLong).contiguous();
  int n_seg = c.numel() - 1;
  auto out = at::empty({n_seg}, values.options());
  if (n_seg == 0) return out;
  auto v = values.contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, values.scalar_type(),
                                  "segment_sum_csr", [&] {
    hipLaunchKernelGGL(segment_sum_csr_kernel<scalar_t>,
                       dim3(grid_for((int64_t)n_seg * kWaveSize, kBlockThreads)),
                       dim3(kBlockThreads), 0, cur_stream(), c.data_ptr<int64_t>(), n_seg,
                       v.data_ptr<scalar_t>(), out.data_ptr<scalar_t>());
  });
  return out;
}

// ---------------------------------------------------------------------------
// block_bucketize_sparse_features (RW sharding).
// Pass 1: one thread per input bag counts per-bucket lengths (deterministic,
// no atomics: each (bucket, f, b) slot has exactly one writer).
// Pass 2: cumsum -> new offsets; one thread per bag scatters values.
// ---------------------------------------------------------------------------

template <typename index_t>
__global__ void bucketize_count_kernel(
    const int64_t* __restrict__ offsets,  // [FB+1]
    const index_t* __restrict__ indices,
    const int64_t* __restrict__ block_sizes,  // [F]
    const int64_t* __restrict__ bag_bounds,   // [F+1] VBE bag->feature, or null
    int F, int B, int num_buckets, int64_t FB,
    int64_t* __restrict__ new_lengths /* [num_buckets*FB] zeroed */) {
  for (int64_t bag = blockIdx.x * blockDim.x + threadIdx.x; bag < FB;
       bag += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int f = bag_bounds ? upper_bound_segment(bag_bounds, F, bag) : (int)(bag / B);
    int64_t bs = block_sizes[f];
    for (int64_t p = offsets[bag]; p < offsets[bag + 1]; ++p) {
      int64_t bkt = indices[p] / bs;
      if (bkt >= num_buckets) bkt = num_buckets - 1;
      new_lengths[bkt * FB + bag] += 1;
    }
  }
}

template <typename index_t, bool HAS_W, bool POS, bool SEQ>
__global__ void bucketize_scatter_kernel(
    const int64_t* __restrict__ offsets, const index_t* __restrict__ indices,
    const float* __restrict__ weights, const int64_t* __restrict__ block_sizes,
    const int64_t* __restrict__ bag_bounds, int F, int B,
    int num_buckets, int64_t FB, const int64_t* __restrict__ new_offsets,
    index_t* __restrict__ new_indices, float* __restrict__ new_weights,
    index_t* __restrict__ new_pos, int64_t* __restrict__ unbucketize) {
  for (int64_t bag = blockIdx.x * blockDim.x + threadIdx.x; bag < FB;
       bag += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int f = bag_bounds ? upper_bound_segment(bag_bounds, F, bag) : (int)(bag / B);
    int64_t bs = block_sizes[f];
    int64_t start = offsets[bag];
    // per-bucket rank is recomputed by rescanning the bag prefix: O(L^2) per
    // bag but bags are short (tens of ids) and this keeps the scatter
    // deterministic with no atomics.
    for (int64_t p = start; p < offsets[bag + 1]; ++p) {
      int64_t idx = indices[p];
      int64_t bkt = idx / bs;
      if (bkt >= num_buckets) bkt = num_buckets - 1;
      // rank of p among same-bucket values of this bag so far
      int64_t rank = 0;
      for (int64_t q = start; q < p; ++q) {
        int64_t b2 = indices[q] / bs;
        if (b2 >= num_buckets) b2 = num_buckets - 1;
        if (b2 == bkt) ++rank;
      }
      int64_t slot = new_offsets[bkt * FB + bag] + rank;
      new_indices[slot] = static_cast<index_t>(idx - bkt * bs);
      if (HAS_W) new_weights[slot] = weights[p];
      if (POS) new_pos[slot] = static_cast<index_t>(p - start);
      if (SEQ) unbucketize[p] = slot;
    }
  }
}

std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor>
block_bucketize_sparse_features(const at::Tensor& lengths, const at::Tensor& indices,
                                const at::Tensor& block_sizes, int64_t num_buckets,
                                bool bucketize_pos, bool sequence, const at::Tensor& weights,
                                const at::Tensor& bag_feature_bounds) {
  // bag_feature_bounds: empty for uniform batch (bag -> feature = bag / B);
  // [F+1] cumulative per-feature bag counts for VBE inputs
  TORCH_CHECK(indices.is_cuda());
  int F = block_sizes.numel();
  int64_t FB = lengths.numel();
  int B = bag_feature_bounds.numel() > 0 ? 0 : (int)(FB / F);
  auto offsets = complete_cumsum(lengths.to(at::kLong).contiguous());
  auto bs = block_sizes.to(at::kLong).to(indices.device()).contiguous();
  auto new_lengths = at::zeros({num_buckets * FB}, lengths.options().dtype(at::kLong));
  bool has_w = weights.numel() > 0;
  auto stream = cur_stream();
  int grid = grid_for(FB, kBlockThreads);
  auto idx = indices.contiguous();
  std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor> result;

  auto bounds = bag_feature_bounds.numel() > 0
      ? bag_feature_bounds.to(at::kLong).to(indices.device()).contiguous()
      : at::Tensor();
  const int64_t* bounds_ptr = bounds.defined() ? bounds.data_ptr<int64_t>() : nullptr;
  AT_DISPATCH_INDEX_TYPES(idx.scalar_type(), "bucketize", [&] {
    hipLaunchKernelGGL(bucketize_count_kernel<index_t>, dim3(grid), dim3(kBlockThreads), 0,
                       stream, offsets.data_ptr<int64_t>(), idx.data_ptr<index_t>(),
                       bs.data_ptr<int64_t>(), bounds_ptr, F, B, (int)num_buckets, FB,
                       new_lengths.data_ptr<int64_t>());
    auto new_offsets = complete_cumsum(new_lengths);
    auto new_indices = at::empty_like(idx);
    auto new_weights = has_w ? at::empty_like(weights) : at::empty({0}, idx.options().dtype(at::kFloat));
    auto new_pos = bucketize_pos ? at::empty_like(idx) : at::empty({0}, idx.options());
    auto unbucketize =
        sequence ? at::empty({idx.numel()}, idx.options().dtype(at::kLong)) : at::empty({0}, idx.options().dtype(at::kLong));

    auto launch = [&](auto has_w_c, auto pos_c, auto seq_c) {
      hipLaunchKernelGGL((bucketize_scatter_kernel<index_t, decltype(has_w_c)::value,
                                                   decltype(pos_c)::value, decltype(seq_c)::value>),
                         dim3(grid), dim3(kBlockThreads), 0, stream, offsets.data_ptr<int64_t>(),
                         idx.data_ptr<index_t>(), has_w ? weights.data_ptr<float>() : nullptr,
                         bs.data_ptr<int64_t>(), bounds_ptr, F, B, (int)num_buckets, FB,
                         new_offsets.data_ptr<int64_t>(), new_indices.data_ptr<index_t>(),
                         has_w ? new_weights.data_ptr<float>() : nullptr,
                         bucketize_pos ? new_pos.data_ptr<index_t>() : nullptr,
                         sequence ? unbucketize.data_ptr<int64_t>() : nullptr);
    };
    using T = std::true_type; using Fa = std::false_type;
    if (has_w) { if (bucketize_pos) { if (sequence) launch(T{},T{},T{}); else launch(T{},T{},Fa{}); }
                 else { if (sequence) launch(T{},Fa{},T{}); else launch(T{},Fa{},Fa{}); } }
    else { if (bucketize_pos) { if (sequence) launch(Fa{},T{},T{}); else launch(Fa{},T{},Fa{}); }
           else { if (sequence) launch(Fa{},Fa{},T{}); else launch(Fa{},Fa{},Fa{}); } }
    result = std::make_tuple(new_lengths, new_indices, new_weights, new_pos, unbucketize);
  });
  return result;
}

// ---------------------------------------------------------------------------
// permute_pooled_embs: column-group permute of [B, sum_D]
// ---------------------------------------------------------------------------

template <typename scalar_t>
__global__ void permute_pooled_kernel(
    const scalar_t* __restrict__ in, const int64_t* __restrict__ in_offsets,
    const int64_t* __restrict__ out_offsets, const int64_t* __restrict__ order, int G, int B,
    int64_t D_total, scalar_t* __restrict__ out) {
  // threads cover B * D_total output elements; map column -> group via search
  int64_t total = static_cast<int64_t>(B) * D_total;
  for (int64_t t = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; t < total;
       t += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int64_t row = t / D_total;
    int64_t col = t - row * D_total;
    int g = upper_bound_segment(out_offsets, G, col);
    int64_t src_col = in_offsets[order[g]] + (col - out_offsets[g]);
    out[t] = in[row * D_total + src_col];
  }
}

at::Tensor permute_pooled_embs(const at::Tensor& values, const at::Tensor& in_offsets,
                               const at::Tensor& out_offsets, const at::Tensor& order) {
  TORCH_CHECK(values.is_cuda() && values.dim() == 2);
  int B = values.size(0);
  int64_t D_total = values.size(1);
  int G = order.numel();
  auto out = at::empty_like(values);
  if (values.numel() == 0) return out;
  auto v = values.contiguous();
  auto io = in_offsets.to(at::kLong).to(values.device()).contiguous();
  auto oo = out_offsets.to(at::kLong).to(values.device()).contiguous();
  auto ord = order.to(at::kLong).to(values.device()).contiguous();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, values.scalar_type(),
                                  "permute_pooled", [&] {
    hipLaunchKernelGGL(permute_pooled_kernel<scalar_t>,
                       dim3(grid_for((int64_t)B * D_total, kBlockThreads)), dim3(kBlockThreads),
                       0, cur_stream(), v.data_ptr<scalar_t>(), io.data_ptr<int64_t>(),
                       oo.data_ptr<int64_t>(), ord.data_ptr<int64_t>(), G, B, D_total,
                       out.data_ptr<scalar_t>());
  });
  return out;
}

}  // namespace trec_amd
