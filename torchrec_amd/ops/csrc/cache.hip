// Software-managed LRU embedding cache for UVM-backed tables (gfx950).
//
// MI355X-native equivalent of FBGEMM's lxu_cache (reference kernel:
// EmbeddingLocation.MANAGED_CACHING with prefetch(), used by
// torchrec/distributed/batched_embedding_kernel.py FUSED_UVM_CACHING).
//
// Layout: set-associative cache in HBM — `C` sets x `kWays`=32 ways.
//   cache_weights [C * 32, max_D] fp32   cache_tags [C * 32] int64 (-1 empty)
//   cache_lru     [C * 32] int64 (last-touch timestamp)
// set(id) = id % C. Population is race-free by construction: candidate ids
// are sorted by set (reusing the TBE backward's radix-sort + run-length
// machinery) and ONE wave owns each set-run; 32 lanes probe the 32 ways in
// parallel, the full wave moves the D-wide rows. Evicted dirty rows are
// written back to the pinned host table before being replaced (training
// updates hit the cache row; host is refreshed on eviction / flush()).

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "common.h"

namespace trec_amd {

constexpr int kWays = 32;

template <typename T>
T* uvm_ptr_cache(const at::Tensor& t) {
  if (t.numel() == 0) return nullptr;
  if (t.is_cuda()) return t.data_ptr<T>();
  TORCH_CHECK(t.is_pinned(), "host tensor must be pinned");
  void* dp = nullptr;
  TREC_HIP_CHECK(hipHostGetDevicePointer(&dp, t.data_ptr(), 0));
  return static_cast<T*>(dp);
}

static inline hipStream_t c_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// host row address of a linear id
__device__ __forceinline__ const float* host_row(
    const float* host_weights, const int64_t* table_row_offsets,
    const int64_t* table_elem_offsets, const int32_t* dims, int T, int64_t lin, int* D_out) {
  int t = upper_bound_segment(table_row_offsets, T, lin);
  int64_t local = lin - table_row_offsets[t];
  *D_out = dims[t];
  return host_weights + table_elem_offsets[t] + local * static_cast<int64_t>(dims[t]);
}

__global__ void __launch_bounds__(kBlockThreads) lxu_cache_populate_kernel(
    float* __restrict__ host_weights, const int64_t* __restrict__ table_row_offsets,
    const int64_t* __restrict__ table_elem_offsets, const int32_t* __restrict__ dims, int T,
    const int64_t* __restrict__ sorted_uniq_ids,  // unique ids sorted by (set)
    const int32_t* __restrict__ seg_offsets,      // per-set runs
    const int32_t* __restrict__ num_runs_ptr,
    float* __restrict__ cache_weights, int64_t* __restrict__ cache_tags,
    int64_t* __restrict__ cache_lru, int64_t C, int64_t max_D, int64_t timestamp) {
  int l = lane_id();
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  int32_t num_runs = *num_runs_ptr;
  for (int64_t r = wave; r < num_runs; r += n_waves) {
    int32_t k0 = seg_offsets[r], k1 = seg_offsets[r + 1];
    if (sorted_uniq_ids[k0] < 0) continue;  // sentinel padding run
    int64_t set = sorted_uniq_ids[k0] % C;
    int64_t base = set * kWays;
    for (int32_t k = k0; k < k1; ++k) {
      int64_t id = sorted_uniq_ids[k];
      // probe: lanes 0..31 check one way each
      int64_t tag = (l < kWays) ? cache_tags[base + l] : -2;
      unsigned long long hitmask = __ballot(tag == id);
      if (hitmask != 0ull) {
        int way = __ffsll((long long)hitmask) - 1;
        if (l == 0) cache_lru[base + way] = timestamp;
        continue;
      }
      // miss: pick LRU way (min lru over lanes 0..31; empty ways are oldest)
      int64_t mylru = (l < kWays) ? ((tag == -1) ? INT64_MIN : cache_lru[base + l])
                                  : INT64_MAX;
      int64_t best = mylru;
      int bestway = l;
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) {
        int64_t o_lru = __shfl_xor(best, off, kWaveSize);
        int o_way = __shfl_xor(bestway, off, kWaveSize);
        if (o_lru < best || (o_lru == best && o_way < bestway)) {
          best = o_lru;
          bestway = o_way;
        }
      }
      int way = bestway;
      int64_t old_tag = cache_tags[base + way];
      float* crow = cache_weights + (base + way) * max_D;
      if (old_tag >= 0) {
        // write back the evicted row to the host table
        int Dv = 0;
        const float* hr = host_row(host_weights, table_row_offsets, table_elem_offsets,
                                   dims, T, old_tag, &Dv);
        float* hw = const_cast<float*>(hr);
        for (int d = l; d < Dv; d += kWaveSize) hw[d] = crow[d];
      }
      int Dn = 0;
      const float* src = host_row(host_weights, table_row_offsets, table_elem_offsets,
                                  dims, T, id, &Dn);
      for (int d = l; d < Dn; d += kWaveSize) crow[d] = src[d];
      if (l == 0) {
        cache_tags[base + way] = id;
        cache_lru[base + way] = timestamp;
      }
      __builtin_amdgcn_s_waitcnt(0);  // order tag publish after row copy (wave-local)
    }
  }
}

void lxu_cache_populate(at::Tensor host_weights, const at::Tensor& table_row_offsets,
                        const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                        const at::Tensor& sorted_uniq_ids, const at::Tensor& seg_offsets,
                        const at::Tensor& num_runs, at::Tensor cache_weights,
                        at::Tensor cache_tags, at::Tensor cache_lru, int64_t max_D,
                        int64_t timestamp) {
  int T = table_elem_offsets.numel();
  int64_t C = cache_tags.numel() / kWays;
  int64_t n = sorted_uniq_ids.numel();
  if (n == 0) return;
  hipLaunchKernelGGL(lxu_cache_populate_kernel,
                     dim3(grid_for(n * kWaveSize, kBlockThreads)), dim3(kBlockThreads), 0,
                     c_stream(), uvm_ptr_cache<float>(host_weights),
                     table_row_offsets.data_ptr<int64_t>(),
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(), T,
                     sorted_uniq_ids.data_ptr<int64_t>(), seg_offsets.data_ptr<int32_t>(),
                     num_runs.data_ptr<int32_t>(), cache_weights.data_ptr<float>(),
                     cache_tags.data_ptr<int64_t>(), cache_lru.data_ptr<int64_t>(), C, max_D,
                     timestamp);
}

__global__ void lxu_cache_lookup_kernel(const int64_t* __restrict__ ids, int64_t N,
                                        const int64_t* __restrict__ cache_tags, int64_t C,
                                        int32_t* __restrict__ out) {
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < N;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int64_t id = ids[i];
    int64_t base = (id % C) * kWays;
    int32_t loc = -1;
    for (int w = 0; w < kWays; ++w) {
      if (cache_tags[base + w] == id) {
        loc = static_cast<int32_t>(base + w);
        break;
      }
    }
    out[i] = loc;
  }
}

at::Tensor lxu_cache_lookup(const at::Tensor& ids, const at::Tensor& cache_tags) {
  int64_t N = ids.numel();
  auto out = at::empty({N}, ids.options().dtype(at::kInt));
  if (N == 0) return out;
  int64_t C = cache_tags.numel() / kWays;
  hipLaunchKernelGGL(lxu_cache_lookup_kernel, dim3(grid_for(N, kBlockThreads)),
                     dim3(kBlockThreads), 0, c_stream(), ids.data_ptr<int64_t>(), N,
                     cache_tags.data_ptr<int64_t>(), C, out.data_ptr<int32_t>());
  return out;
}

__global__ void __launch_bounds__(kBlockThreads) lxu_cache_flush_kernel(
    float* __restrict__ host_weights, const int64_t* __restrict__ table_row_offsets,
    const int64_t* __restrict__ table_elem_offsets, const int32_t* __restrict__ dims, int T,
    const float* __restrict__ cache_weights, const int64_t* __restrict__ cache_tags,
    int64_t slots, int64_t max_D) {
  int l = lane_id();
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  for (int64_t slot = wave; slot < slots; slot += n_waves) {
    int64_t tag = cache_tags[slot];
    if (tag < 0) continue;
    int Dv = 0;
    const float* hr =
        host_row(host_weights, table_row_offsets, table_elem_offsets, dims, T, tag, &Dv);
    float* hw = const_cast<float*>(hr);
    const float* crow = cache_weights + slot * max_D;
    for (int d = l; d < Dv; d += kWaveSize) hw[d] = crow[d];
  }
}

void lxu_cache_flush(at::Tensor host_weights, const at::Tensor& table_row_offsets,
                     const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                     const at::Tensor& cache_weights, const at::Tensor& cache_tags,
                     int64_t max_D) {
  int T = table_elem_offsets.numel();
  int64_t slots = cache_tags.numel();
  if (slots == 0) return;
  hipLaunchKernelGGL(lxu_cache_flush_kernel,
                     dim3(grid_for(slots * kWaveSize, kBlockThreads)), dim3(kBlockThreads),
                     0, c_stream(), uvm_ptr_cache<float>(host_weights),
                     table_row_offsets.data_ptr<int64_t>(),
                     table_elem_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(), T,
                     cache_weights.data_ptr<float>(), cache_tags.data_ptr<int64_t>(), slots,
                     max_D);
}

}  // namespace trec_amd
