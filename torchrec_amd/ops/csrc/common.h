// Common device helpers for torchrec_amd CDNA4 (gfx950) kernels.
//
// Written for MI355X: 64-wide wavefronts, 32-bank LDS, 8 XCDs, HBM3E.
// See /opt/skills/guides/cdna_hip_programming.md for the hardware model.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace trec_amd {

constexpr int kWaveSize = 64;  // CDNA wavefront — NOT 32
constexpr int kBlockThreads = 256;
constexpr int kNumXCD = 8;
constexpr int kNumCU = 256;
// memory-bound launch cap (Guideline 11): ~8 blocks/CU
constexpr int kMaxBlocks = kNumCU * 8;

#define TREC_HIP_CHECK(cmd)                                                   \
  do {                                                                        \
    hipError_t e_ = (cmd);                                                    \
    TORCH_CHECK(e_ == hipSuccess, "HIP error: ", hipGetErrorString(e_));      \
  } while (0)

__device__ __forceinline__ int lane_id() { return threadIdx.x & (kWaveSize - 1); }
__device__ __forceinline__ int wave_id() { return threadIdx.x >> 6; }

// Reduce across a power-of-two lane group of size `width` (<= 64).
template <int WIDTH>
__device__ __forceinline__ float group_reduce_sum(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, kWaveSize);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
  return group_reduce_sum<kWaveSize>(v);
}

__device__ __forceinline__ int64_t cdiv(int64_t a, int64_t b) { return (a + b - 1) / b; }

inline int grid_for(int64_t work, int per_block) {
  int64_t blocks = (work + per_block - 1) / per_block;
  if (blocks > kMaxBlocks) blocks = kMaxBlocks;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

// Binary search: greatest t such that offs[t] <= x < offs[t+1]; offs has n+1
// monotonically non-decreasing entries.
__device__ __forceinline__ int upper_bound_segment(const int64_t* offs, int n, int64_t x) {
  int lo = 0, hi = n;  // invariant: offs[lo] <= x < offs[hi]
  while (hi - lo > 1) {
    int mid = (lo + hi) >> 1;
    if (offs[mid] <= x) lo = mid; else hi = mid;
  }
  return lo;
}

}  // namespace trec_amd
