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


// ---------------------------------------------------------------------------
// MPZCH: GPU zero-collision hash remapping (reference:
// torchrec/modules/hash_mc_modules.py HashZchManagedCollisionModule :196,
// fbgemm zero_collision_hash :451).
//
// identity[z] holds the raw id owning slot z (-1 = free). Each input id
// linear-probes from hash(id) % Z; in training mode a free slot is claimed
// with a 64-bit atomicCAS (first-claimer wins — the id->slot map stays
// consistent because every probe sequence re-reads the claimed value).
// metadata[z] is a last-seen counter for eviction.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t zch_mix(uint64_t x) {
  // splitmix64 finalizer
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

__global__ void hash_zch_kernel(const int64_t* __restrict__ ids, int64_t n,
                                int64_t* __restrict__ identity,
                                int32_t* __restrict__ metadata, int64_t Z,
                                int max_probe, int32_t stamp, bool train,
                                int64_t* __restrict__ out) {
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; i < n;
       i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    int64_t id = ids[i];
    uint64_t h = zch_mix(static_cast<uint64_t>(id));
    int64_t slot = -1;
    for (int p = 0; p < max_probe; ++p) {
      int64_t z = static_cast<int64_t>((h + p) % static_cast<uint64_t>(Z));
      int64_t cur = identity[z];
      if (cur == id) { slot = z; break; }
      if (cur == -1 && train) {
        int64_t prev = atomicCAS(
            reinterpret_cast<unsigned long long*>(identity + z),
            static_cast<unsigned long long>(-1ll),
            static_cast<unsigned long long>(id));
        if (prev == static_cast<unsigned long long>(-1ll) ||
            static_cast<int64_t>(prev) == id) {
          slot = z;
          break;
        }
        cur = static_cast<int64_t>(prev);
        if (cur == id) { slot = z; break; }
      }
    }
    if (slot < 0) {
      // probe budget exhausted: shared fallback bucket (hash slot)
      slot = static_cast<int64_t>(h % static_cast<uint64_t>(Z));
    }
    if (train) metadata[slot] = stamp;
    out[i] = slot;
  }
}

__global__ void hash_zch_evict_kernel(int64_t* __restrict__ identity,
                                      int32_t* __restrict__ metadata, int64_t Z,
                                      int32_t older_than,
                                      int64_t* __restrict__ evicted /* [Z] -1 pad */) {
  for (int64_t z = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x; z < Z;
       z += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    if (identity[z] != -1 && metadata[z] < older_than) {
      evicted[z] = z;
      identity[z] = -1;
      metadata[z] = 0;
    } else {
      evicted[z] = -1;
    }
  }
}

at::Tensor hash_zch_remap(const at::Tensor& ids, at::Tensor identity,
                          at::Tensor metadata, int64_t max_probe, int64_t stamp,
                          bool train) {
  TORCH_CHECK(ids.is_cuda() && identity.is_cuda());
  int64_t n = ids.numel();
  int64_t Z = identity.numel();
  auto out = at::empty({n}, ids.options());
  if (n == 0) return out;
  hipLaunchKernelGGL(hash_zch_kernel, dim3(grid_for(n, kBlockThreads)),
                     dim3(kBlockThreads), 0, c_stream(), ids.data_ptr<int64_t>(), n,
                     identity.data_ptr<int64_t>(), metadata.data_ptr<int32_t>(), Z,
                     (int)max_probe, (int32_t)stamp, train, out.data_ptr<int64_t>());
  return out;
}

at::Tensor hash_zch_evict(at::Tensor identity, at::Tensor metadata, int64_t older_than) {
  int64_t Z = identity.numel();
  auto evicted = at::empty({Z}, identity.options());
  if (Z == 0) return evicted;
  hipLaunchKernelGGL(hash_zch_evict_kernel, dim3(grid_for(Z, kBlockThreads)),
                     dim3(kBlockThreads), 0, c_stream(), identity.data_ptr<int64_t>(),
                     metadata.data_ptr<int32_t>(), Z, (int32_t)older_than,
                     evicted.data_ptr<int64_t>());
  return evicted;
}

}  // namespace trec_amd
