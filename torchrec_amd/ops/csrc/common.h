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

// -----------------------------------------------------------------------
// dtype-generic 4-element row accessor for embedding tables.
// fp32 rows move as float4 (16 B/lane); bf16/fp16 rows as 8 B/lane with
// fp32 accumulate and round-to-nearest store (reference weights_precision).
// -----------------------------------------------------------------------

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

namespace trec_amd {

// scalar emb_t -> fp32 (PyTorch extension builds define
// __HIP_NO_HALF_CONVERSIONS__, so implicit casts are unavailable)
__device__ __forceinline__ float emb2float(float v) { return v; }
__device__ __forceinline__ float emb2float(__half v) { return __half2float(v); }
__device__ __forceinline__ float emb2float(__hip_bfloat16 v) { return __bfloat162float(v); }


// host scalar -> device scalar for AT_DISPATCH'd kernels (requires ATen,
// which every .hip TU in this tree includes via torch/extension.h)
template <typename scalar_t>
struct DevType { using type = scalar_t; };
template <> struct DevType<c10::Half> { using type = __half; };
template <> struct DevType<c10::BFloat16> { using type = __hip_bfloat16; };

// fp32 -> scalar emb_t (tag-dispatched: the second arg selects the overload)
__device__ __forceinline__ float float2emb(float v, float) { return v; }
__device__ __forceinline__ __half float2emb(float v, __half) { return __float2half(v); }
__device__ __forceinline__ __hip_bfloat16 float2emb(float v, __hip_bfloat16) {
  return __float2bfloat16(v);
}

template <typename emb_t>
struct Vec4;

template <>
struct Vec4<float> {
  __device__ static float4 load(const float* row, int col4) {
    return reinterpret_cast<const float4*>(row)[col4];
  }
  __device__ static void store(float* row, int col4, float4 v) {
    reinterpret_cast<float4*>(row)[col4] = v;
  }
};

template <>
struct Vec4<__hip_bfloat16> {
  __device__ static float4 load(const __hip_bfloat16* row, int col4) {
    const ushort4 q = reinterpret_cast<const ushort4*>(row)[col4];
    __hip_bfloat16 a, b, c, d;
    *reinterpret_cast<unsigned short*>(&a) = q.x;
    *reinterpret_cast<unsigned short*>(&b) = q.y;
    *reinterpret_cast<unsigned short*>(&c) = q.z;
    *reinterpret_cast<unsigned short*>(&d) = q.w;
    return make_float4(__bfloat162float(a), __bfloat162float(b), __bfloat162float(c),
                       __bfloat162float(d));
  }
  __device__ static void store(__hip_bfloat16* row, int col4, float4 v) {
    ushort4 q;
    __hip_bfloat16 a = __float2bfloat16(v.x), b = __float2bfloat16(v.y),
                   c = __float2bfloat16(v.z), d = __float2bfloat16(v.w);
    q.x = *reinterpret_cast<unsigned short*>(&a);
    q.y = *reinterpret_cast<unsigned short*>(&b);
    q.z = *reinterpret_cast<unsigned short*>(&c);
    q.w = *reinterpret_cast<unsigned short*>(&d);
    reinterpret_cast<ushort4*>(row)[col4] = q;
  }
};

template <>
struct Vec4<__half> {
  __device__ static float4 load(const __half* row, int col4) {
    const ushort4 q = reinterpret_cast<const ushort4*>(row)[col4];
    __half a, b, c, d;
    *reinterpret_cast<unsigned short*>(&a) = q.x;
    *reinterpret_cast<unsigned short*>(&b) = q.y;
    *reinterpret_cast<unsigned short*>(&c) = q.z;
    *reinterpret_cast<unsigned short*>(&d) = q.w;
    return make_float4(__half2float(a), __half2float(b), __half2float(c), __half2float(d));
  }
  __device__ static void store(__half* row, int col4, float4 v) {
    ushort4 q;
    __half a = __float2half(v.x), b = __float2half(v.y), c = __float2half(v.z),
           d = __float2half(v.w);
    q.x = *reinterpret_cast<unsigned short*>(&a);
    q.y = *reinterpret_cast<unsigned short*>(&b);
    q.z = *reinterpret_cast<unsigned short*>(&c);
    q.w = *reinterpret_cast<unsigned short*>(&d);
    reinterpret_cast<ushort4*>(row)[col4] = q;
  }
};

}  // namespace trec_amd
