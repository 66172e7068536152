// Fused MLP backward helpers for MI355X (gfx950).
//
// relu_bwd_col_sum: one pass over dY producing BOTH the relu-masked gradient
// g = dY * (y > 0) and the bias gradient db = colsum(g) (fp32 accumulate,
// deterministic two-phase fixed-partition reduction). Replaces torch's
// separate compare + mul + column reduce (three dY-sized passes) with one.
// 16-bit dtypes move as 8-element (16 B) vectors per lane (Guideline 13:
// scalar bf16 loads are ~2.5x slower).

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <cstdlib>
#include <type_traits>

#include "common.h"

namespace trec_amd {

static inline hipStream_t mlp_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// VPT elements per thread along the column axis; rows_per_group rows per
// partial. Work item = (column 8-chunk, row group), grid-strided.
template <typename scalar_t, int VPT>
__global__ void __launch_bounds__(kBlockThreads) relu_bwd_colsum_partial_kernel(
    const scalar_t* __restrict__ dy, const scalar_t* __restrict__ y, int64_t M,
    int64_t N, int rows_per_group, int n_groups, scalar_t* __restrict__ g,
    float* __restrict__ partial /* [G, N] */, int* __restrict__ counters,
    int n_tiles) {
  // zero the finish kernel's per-tile semaphores (it launches after us on
  // the same stream, so ordering is guaranteed)
  if (counters && blockIdx.x == 0)
    for (int t = threadIdx.x; t < n_tiles; t += blockDim.x) counters[t] = 0;
  const int64_t col_chunks = (N + VPT - 1) / VPT;
  const int64_t items = col_chunks * n_groups;
  for (int64_t it = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       it < items; it += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int grp = static_cast<int>(it / col_chunks);
    const int64_t c0 = (it - static_cast<int64_t>(grp) * col_chunks) * VPT;
    const int64_t r0 = static_cast<int64_t>(grp) * rows_per_group;
    const int64_t r1 = min(M, r0 + static_cast<int64_t>(rows_per_group));
    float acc[VPT];
#pragma unroll
    for (int v = 0; v < VPT; ++v) acc[v] = 0.f;
    const bool full = (c0 + VPT <= N);
    for (int64_t r = r0; r < r1; ++r) {
      const int64_t base = r * N + c0;
      if (full && sizeof(scalar_t) == 2 && VPT == 8) {
        // 8 x 16-bit = one 16 B load per operand, one 16 B store
        uint4 qy = *reinterpret_cast<const uint4*>(y + base);
        uint4 qd = *reinterpret_cast<const uint4*>(dy + base);
        const scalar_t* py = reinterpret_cast<const scalar_t*>(&qy);
        const scalar_t* pd = reinterpret_cast<const scalar_t*>(&qd);
        uint4 qo;
        scalar_t* po = reinterpret_cast<scalar_t*>(&qo);
#pragma unroll
        for (int v = 0; v < VPT; ++v) {
          float gv = (emb2float(py[v]) > 0.f) ? emb2float(pd[v]) : 0.f;
          po[v] = float2emb(gv, scalar_t{});
          acc[v] += gv;
        }
        *reinterpret_cast<uint4*>(g + base) = qo;
      } else if (full && sizeof(scalar_t) == 4 && VPT == 4) {
        uint4 qy = *reinterpret_cast<const uint4*>(y + base);
        uint4 qd = *reinterpret_cast<const uint4*>(dy + base);
        const scalar_t* py = reinterpret_cast<const scalar_t*>(&qy);
        const scalar_t* pd = reinterpret_cast<const scalar_t*>(&qd);
        uint4 qo;
        scalar_t* po = reinterpret_cast<scalar_t*>(&qo);
#pragma unroll
        for (int v = 0; v < VPT; ++v) {
          float gv = (emb2float(py[v]) > 0.f) ? emb2float(pd[v]) : 0.f;
          po[v] = float2emb(gv, scalar_t{});
          acc[v] += gv;
        }
        *reinterpret_cast<uint4*>(g + base) = qo;
      } else {
        for (int v = 0; v < VPT && c0 + v < N; ++v) {
          float gv = (emb2float(y[base + v]) > 0.f) ? emb2float(dy[base + v]) : 0.f;
          g[base + v] = float2emb(gv, scalar_t{});
          acc[v] += gv;
        }
      }
    }
    float* prow = partial + static_cast<int64_t>(grp) * N + c0;
#pragma unroll
    for (int v = 0; v < VPT; ++v)
      if (c0 + v < N) prow[v] = acc[v];
  }
}

// one-kernel segment+final fold, take 2 (first try rejected at 21.8 us for
// per-element ORDERED atomics): phase 1 writes per-segment sums with plain
// float4 stores; publication is one device fence + a RELAXED counter; the
// last block per column tile re-fences and folds the 32 segment rows with
// PLAIN float4 loads (independent, pipelined). Fold order is fixed =>
// deterministic.
template <typename o_t>
__global__ void __launch_bounds__(kBlockThreads) colsum_finish2_kernel(
    const float* __restrict__ partial, int G, int seg_rows, int64_t N,
    float* __restrict__ seg /* [S, N] */, int* __restrict__ counters,
    int S, o_t* __restrict__ db) {
  // each thread owns 4 consecutive columns (float4 lanes)
  int64_t c4 = (static_cast<int64_t>(blockIdx.x) * kBlockThreads + threadIdx.x) * 4;
  int sgm = blockIdx.y;
  if (c4 + 4 <= N) {
    int g0 = sgm * seg_rows;
    int g1 = min(G, g0 + seg_rows);
    float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int grp = g0; grp < g1; ++grp) {
      float4 v = *reinterpret_cast<const float4*>(
          partial + static_cast<int64_t>(grp) * N + c4);
      acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
    }
    *reinterpret_cast<float4*>(seg + static_cast<int64_t>(sgm) * N + c4) = acc;
  } else {
    for (int64_t c = c4; c < N; ++c) {
      int g0 = sgm * seg_rows, g1 = min(G, g0 + seg_rows);
      float a = 0.f;
      for (int grp = g0; grp < g1; ++grp)
        a += partial[static_cast<int64_t>(grp) * N + c];
      seg[static_cast<int64_t>(sgm) * N + c] = a;
    }
  }
  __threadfence();
  __syncthreads();
  __shared__ int last;
  if (threadIdx.x == 0)
    last = (__hip_atomic_fetch_add(&counters[blockIdx.x], 1, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT) == S - 1);
  __syncthreads();
  if (!last) return;
  __threadfence();
  if (c4 + 4 <= N) {
    float4 tot = make_float4(0.f, 0.f, 0.f, 0.f);
    for (int ss = 0; ss < S; ++ss) {
      float4 v = *reinterpret_cast<const float4*>(
          seg + static_cast<int64_t>(ss) * N + c4);
      tot.x += v.x; tot.y += v.y; tot.z += v.z; tot.w += v.w;
    }
    db[c4 + 0] = float2emb(tot.x, o_t{});
    db[c4 + 1] = float2emb(tot.y, o_t{});
    db[c4 + 2] = float2emb(tot.z, o_t{});
    db[c4 + 3] = float2emb(tot.w, o_t{});
  } else {
    for (int64_t c = c4; c < N; ++c) {
      float tot = 0.f;
      for (int ss = 0; ss < S; ++ss)
        tot += seg[static_cast<int64_t>(ss) * N + c];
      db[c] = float2emb(tot, o_t{});
    }
  }
}

// level-1 segment reduce: 2-D grid ((column tile) x S) collapses G rows to
// S per-tile sums. Measured better than a single-kernel semaphore finish
// (21.8 us) as a pair with the cast-fused final (9 + 8.5 us): the last-block
// fold's 32-row tail ran on too few blocks.
__global__ void __launch_bounds__(kBlockThreads) colsum_seg_f32_kernel(
    const float* __restrict__ partial, int G, int seg_rows, int64_t N,
    float* __restrict__ seg_out /* [S, N] */) {
  int64_t c = static_cast<int64_t>(blockIdx.x) * kBlockThreads + threadIdx.x;
  if (c >= N) return;
  int s = blockIdx.y;
  int g0 = s * seg_rows;
  int g1 = min(G, g0 + seg_rows);
  float acc = 0.f;
  for (int grp = g0; grp < g1; ++grp)
    acc += partial[static_cast<int64_t>(grp) * N + c];
  seg_out[static_cast<int64_t>(s) * N + c] = acc;
}

template <typename o_t>
__global__ void __launch_bounds__(kBlockThreads) colsum_final_cast_kernel(
    const float* __restrict__ partial, int G, int64_t N, o_t* __restrict__ out) {
  int64_t c = static_cast<int64_t>(blockIdx.x) * kBlockThreads + threadIdx.x;
  if (c >= N) return;
  float acc = 0.f;
  for (int grp = 0; grp < G; ++grp) acc += partial[static_cast<int64_t>(grp) * N + c];
  out[c] = float2emb(acc, o_t{});
}

std::tuple<at::Tensor, at::Tensor> relu_bwd_col_sum(const at::Tensor& grad_out,
                                                    const at::Tensor& y) {
  TORCH_CHECK(grad_out.dim() == 2 && grad_out.is_cuda());
  TORCH_CHECK(grad_out.sizes() == y.sizes() && grad_out.scalar_type() == y.scalar_type());
  int64_t M = grad_out.size(0), N = grad_out.size(1);
  auto dy = grad_out.contiguous();
  auto yc = y.contiguous();
  auto g = at::empty_like(dy);
  auto db = at::empty({N}, dy.options());  // bias grad in the grad dtype
  if (M == 0 || N == 0) {
    db.zero_();
    return {g, db};
  }
  auto stream = mlp_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dy.scalar_type(),
                                  "relu_bwd_col_sum", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      constexpr int VPT = sizeof(dev_t) == 2 ? 8 : 4;
      int64_t col_chunks = (N + VPT - 1) / VPT;
      // enough (col-chunk, row-group) items to fill ~2048 blocks
      int G = std::max<int>(1, std::min<int64_t>(
          M, (static_cast<int64_t>(kMaxBlocks) * kBlockThreads) / std::max<int64_t>(col_chunks, 1)));
      G = std::min(G, 1024);
      int rows_per_group = (int)((M + G - 1) / G);
      G = (int)((M + rows_per_group - 1) / rows_per_group);
      auto partial = at::empty({(int64_t)G * N}, dy.options().dtype(at::kFloat));
      int ftiles = (int)((N + kBlockThreads - 1) / kBlockThreads);
      int grid = grid_for(col_chunks * G, kBlockThreads);
      // same-box A/B: finish2 1.344 vs pair 1.326 ms/step — the one-kernel
      // fold still loses to the two-kernel pair (the last-block tail runs
      // on tiles4 blocks only); kept opt-in for future shapes
      static const bool use_finish2 = [] {
        const char* e = std::getenv("TREC_COLSUM_FINISH2");
        return e && e[0] == '1';
      }();
      int tiles4 = (int)((N / 4 + kBlockThreads - 1) / kBlockThreads);
      if (tiles4 < 1) tiles4 = 1;
      at::Tensor counters;
      int* cnt_ptr = nullptr;
      if (G > 64 && use_finish2) {
        // zeroed by block 0 of the partial kernel (same stream => ordered)
        counters = at::empty({tiles4}, dy.options().dtype(at::kInt));
        cnt_ptr = counters.data_ptr<int>();
      }
      hipLaunchKernelGGL((relu_bwd_colsum_partial_kernel<dev_t, VPT>), dim3(grid),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(dy.data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(yc.data_ptr<scalar_t>()), M, N,
                         rows_per_group, G,
                         reinterpret_cast<dev_t*>(g.data_ptr<scalar_t>()),
                         partial.data_ptr<float>(), cnt_ptr, tiles4);
      if (G > 64 && use_finish2) {
        constexpr int S = 32;
        int seg_rows = (G + S - 1) / S;
        auto seg = at::empty({(int64_t)S * N}, dy.options().dtype(at::kFloat));
        hipLaunchKernelGGL((colsum_finish2_kernel<dev_t>), dim3(tiles4, S),
                           dim3(kBlockThreads), 0, stream, partial.data_ptr<float>(),
                           G, seg_rows, N, seg.data_ptr<float>(),
                           cnt_ptr, S,
                           reinterpret_cast<dev_t*>(db.data_ptr<scalar_t>()));
      } else if (G > 64) {
        constexpr int S = 32;
        int seg_rows = (G + S - 1) / S;
        auto seg = at::empty({(int64_t)S * N}, dy.options().dtype(at::kFloat));
        hipLaunchKernelGGL(colsum_seg_f32_kernel, dim3(ftiles, S),
                           dim3(kBlockThreads), 0, stream, partial.data_ptr<float>(),
                           G, seg_rows, N, seg.data_ptr<float>());
        hipLaunchKernelGGL((colsum_final_cast_kernel<dev_t>), dim3(ftiles),
                           dim3(kBlockThreads), 0, stream, seg.data_ptr<float>(),
                           S, N, reinterpret_cast<dev_t*>(db.data_ptr<scalar_t>()));
      } else {
        hipLaunchKernelGGL((colsum_final_cast_kernel<dev_t>), dim3(ftiles),
                           dim3(kBlockThreads), 0, stream, partial.data_ptr<float>(),
                           G, N, reinterpret_cast<dev_t*>(db.data_ptr<scalar_t>()));
      }
    }
  });
  return {g, db};
}

// relu mask only: g = dy * (y > 0) — one vectorized pass; the bias grad
// comes out of the wgrad GEMM's BGRADB epilogue (lt_gemm.hip), so the
// column-sum machinery is skipped entirely on that path.
template <typename scalar_t, int VPT>
__global__ void __launch_bounds__(kBlockThreads) relu_bwd_mask_kernel(
    const scalar_t* __restrict__ dy, const scalar_t* __restrict__ y, int64_t n,
    scalar_t* __restrict__ g) {
  const int64_t chunks = (n + VPT - 1) / VPT;
  for (int64_t it = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       it < chunks; it += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    const int64_t base = it * VPT;
    if (base + VPT <= n) {
      uint4 qy = *reinterpret_cast<const uint4*>(y + base);
      uint4 qd = *reinterpret_cast<const uint4*>(dy + base);
      const scalar_t* py = reinterpret_cast<const scalar_t*>(&qy);
      const scalar_t* pd = reinterpret_cast<const scalar_t*>(&qd);
      uint4 qo;
      scalar_t* po = reinterpret_cast<scalar_t*>(&qo);
#pragma unroll
      for (int v = 0; v < VPT; ++v)
        po[v] = (emb2float(py[v]) > 0.f) ? pd[v] : float2emb(0.f, scalar_t{});
      *reinterpret_cast<uint4*>(g + base) = qo;
    } else {
      for (int64_t i = base; i < n; ++i)
        g[i] = (emb2float(y[i]) > 0.f) ? dy[i] : float2emb(0.f, scalar_t{});
    }
  }
}

at::Tensor relu_bwd_mask(const at::Tensor& grad_out, const at::Tensor& y) {
  TORCH_CHECK(grad_out.is_cuda() && grad_out.sizes() == y.sizes());
  auto dy = grad_out.contiguous();
  auto yc = y.contiguous();
  auto g = at::empty_like(dy);
  int64_t n = dy.numel();
  if (n == 0) return g;
  auto stream = mlp_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dy.scalar_type(),
                                  "relu_bwd_mask", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      constexpr int VPT = sizeof(dev_t) == 2 ? 8 : 4;
      int grid = grid_for((n + VPT - 1) / VPT, kBlockThreads);
      hipLaunchKernelGGL((relu_bwd_mask_kernel<dev_t, VPT>), dim3(grid),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(dy.data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(yc.data_ptr<scalar_t>()), n,
                         reinterpret_cast<dev_t*>(g.data_ptr<scalar_t>()));
    }
  });
  return g;
}

// ---------------------------------------------------------------------------
// fused BCE-with-logits (mean reduction). torch's BCEWithLogitsLoss expands
// to ~12 launch-floor kernels per step inside the captured graph
// (log_sigmoid fwd chain + sigmoid/sub/scale bwd); these two kernels do one
// pass each. fwd: deterministic two-level mean of
//   max(x,0) - x*y + log1p(exp(-|x|)); bwd: dx = go * (sigmoid(x) - y) / B.
// ---------------------------------------------------------------------------

__device__ __forceinline__ float bce_ldf(float v) { return v; }
__device__ __forceinline__ float bce_ldf(__half v) { return __half2float(v); }
__device__ __forceinline__ float bce_ldf(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename x_t>
__global__ void __launch_bounds__(kBlockThreads) bce_logits_partial_kernel(
    const x_t* __restrict__ logits, const float* __restrict__ labels, int64_t B,
    float* __restrict__ partial /* [G] */) {
  __shared__ float red[kBlockThreads / kWaveSize];
  float acc = 0.f;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < B; i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    float x = bce_ldf(logits[i]);
    float y = labels[i];
    acc += fmaxf(x, 0.f) - x * y + log1pf(__expf(-fabsf(x)));
  }
  // wave then block reduce
  for (int off = kWaveSize / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, kWaveSize);
  int wave = threadIdx.x / kWaveSize;
  if (threadIdx.x % kWaveSize == 0) red[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int w = 0; w < kBlockThreads / kWaveSize; ++w) s += red[w];
    partial[blockIdx.x] = s;
  }
}

__global__ void bce_logits_final_kernel(const float* __restrict__ partial, int G,
                                        float inv_B, float* __restrict__ out) {
  if (threadIdx.x == 0) {
    float s = 0.f;
    for (int g = 0; g < G; ++g) s += partial[g];
    *out = s * inv_B;
  }
}

template <typename x_t>
__global__ void __launch_bounds__(kBlockThreads) bce_logits_bwd_kernel(
    const x_t* __restrict__ logits, const float* __restrict__ labels,
    const float* __restrict__ grad_out, float inv_B, int64_t B,
    x_t* __restrict__ dx) {
  float g = *grad_out * inv_B;
  for (int64_t i = static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < B; i += static_cast<int64_t>(gridDim.x) * blockDim.x) {
    float x = bce_ldf(logits[i]);
    float sig = 1.f / (1.f + __expf(-x));
    dx[i] = float2emb(g * (sig - labels[i]), x_t{});
  }
}

at::Tensor bce_with_logits_fwd(const at::Tensor& logits, const at::Tensor& labels) {
  TORCH_CHECK(logits.is_cuda() && labels.scalar_type() == at::kFloat);
  int64_t B = logits.numel();
  auto out = at::empty({}, logits.options().dtype(at::kFloat));
  auto stream = mlp_stream();
  int G = std::min<int64_t>(512, (B + kBlockThreads - 1) / kBlockThreads);
  G = std::max(G, 1);
  auto partial = at::empty({G}, logits.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, logits.scalar_type(),
                                  "bce_fwd", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      hipLaunchKernelGGL((bce_logits_partial_kernel<dev_t>), dim3(G),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(logits.data_ptr<scalar_t>()),
                         labels.data_ptr<float>(), B, partial.data_ptr<float>());
    }
  });
  hipLaunchKernelGGL(bce_logits_final_kernel, dim3(1), dim3(kWaveSize), 0, stream,
                     partial.data_ptr<float>(), G, B > 0 ? 1.f / B : 0.f,
                     out.data_ptr<float>());
  return out;
}

at::Tensor bce_with_logits_bwd(const at::Tensor& logits, const at::Tensor& labels,
                               const at::Tensor& grad_out) {
  int64_t B = logits.numel();
  auto dx = at::empty_like(logits);
  if (B == 0) return dx;
  auto stream = mlp_stream();
  int grid = grid_for(B, kBlockThreads);
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, logits.scalar_type(),
                                  "bce_bwd", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      hipLaunchKernelGGL((bce_logits_bwd_kernel<dev_t>), dim3(grid),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(logits.data_ptr<scalar_t>()),
                         labels.data_ptr<float>(),
                         grad_out.data_ptr<float>(), 1.f / B, B,
                         reinterpret_cast<dev_t*>(dx.data_ptr<scalar_t>()));
    }
  });
  return dx;
}

}  // namespace trec_amd
