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
    float* __restrict__ partial /* [G, N] */) {
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

// two-level deterministic reduce of [G, N] -> [N]: level 1 collapses G into
// S segments with a 2-D grid (full-chip parallel), level 2 folds S
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

__global__ void __launch_bounds__(kBlockThreads) colsum_final_f32_kernel(
    const float* __restrict__ partial, int G, int64_t N, float* __restrict__ out) {
  int64_t c = static_cast<int64_t>(blockIdx.x) * kBlockThreads + threadIdx.x;
  if (c >= N) return;
  float acc = 0.f;
  for (int grp = 0; grp < G; ++grp) acc += partial[static_cast<int64_t>(grp) * N + c];
  out[c] = acc;
}

std::tuple<at::Tensor, at::Tensor> relu_bwd_col_sum(const at::Tensor& grad_out,
                                                    const at::Tensor& y) {
  TORCH_CHECK(grad_out.dim() == 2 && grad_out.is_cuda());
  TORCH_CHECK(grad_out.sizes() == y.sizes() && grad_out.scalar_type() == y.scalar_type());
  int64_t M = grad_out.size(0), N = grad_out.size(1);
  auto dy = grad_out.contiguous();
  auto yc = y.contiguous();
  auto g = at::empty_like(dy);
  auto db = at::empty({N}, dy.options().dtype(at::kFloat));
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
      int grid = grid_for(col_chunks * G, kBlockThreads);
      hipLaunchKernelGGL((relu_bwd_colsum_partial_kernel<dev_t, VPT>), dim3(grid),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(dy.data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(yc.data_ptr<scalar_t>()), M, N,
                         rows_per_group, G,
                         reinterpret_cast<dev_t*>(g.data_ptr<scalar_t>()),
                         partial.data_ptr<float>());
      int ftiles = (int)((N + kBlockThreads - 1) / kBlockThreads);
      if (G > 64) {
        constexpr int S = 32;
        int seg_rows = (G + S - 1) / S;
        auto seg = at::empty({(int64_t)S * N}, dy.options().dtype(at::kFloat));
        hipLaunchKernelGGL(colsum_seg_f32_kernel, dim3(ftiles, S),
                           dim3(kBlockThreads), 0, stream, partial.data_ptr<float>(),
                           G, seg_rows, N, seg.data_ptr<float>());
        hipLaunchKernelGGL(colsum_final_f32_kernel, dim3(ftiles), dim3(kBlockThreads),
                           0, stream, seg.data_ptr<float>(), S, N,
                           db.data_ptr<float>());
      } else {
        hipLaunchKernelGGL(colsum_final_f32_kernel, dim3(ftiles), dim3(kBlockThreads),
                           0, stream, partial.data_ptr<float>(), G, N,
                           db.data_ptr<float>());
      }
    }
  });
  return {g, db};
}

}  // namespace trec_amd
