// Fused MLP backward helpers for MI355X (gfx950).
//
// relu_bwd_col_sum: one pass over dY producing BOTH the relu-masked gradient
// g = dY * (y > 0) and the bias gradient db = colsum(g) (fp32 accumulate,
// deterministic two-phase fixed-partition reduction). Replaces torch's
// separate threshold-backward elementwise + column reduce — one read of dY
// instead of two, and two fewer kernel launches per Linear+ReLU layer.

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <type_traits>

#include "common.h"

namespace trec_amd {

static inline hipStream_t mlp_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

template <typename scalar_t>
__global__ void __launch_bounds__(kBlockThreads) relu_bwd_colsum_partial_kernel(
    const scalar_t* __restrict__ dy, const scalar_t* __restrict__ y, int64_t M,
    int64_t N, int rows_per_group, scalar_t* __restrict__ g,
    float* __restrict__ partial /* [G, N] */) {
  int tile = blockIdx.x;
  int grp = blockIdx.y;
  int64_t c = static_cast<int64_t>(tile) * kBlockThreads + threadIdx.x;
  if (c >= N) return;
  int64_t r0 = static_cast<int64_t>(grp) * rows_per_group;
  int64_t r1 = min(M, r0 + rows_per_group);
  float acc = 0.f;
  for (int64_t r = r0; r < r1; ++r) {
    int64_t i = r * N + c;
    float v = (emb2float(y[i]) > 0.f) ? emb2float(dy[i]) : 0.f;
    g[i] = float2emb(v, scalar_t{});
    acc += v;
  }
  partial[static_cast<int64_t>(grp) * N + c] = acc;
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
  int ntiles = (int)((N + kBlockThreads - 1) / kBlockThreads);
  int G = std::max(1, std::min<int>(kMaxBlocks / std::max(ntiles, 1),
                                    (int)((M + 31) / 32)));
  int rows_per_group = (int)((M + G - 1) / G);
  auto partial = at::empty({(int64_t)G * N}, dy.options().dtype(at::kFloat));
  auto stream = mlp_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dy.scalar_type(),
                                  "relu_bwd_col_sum", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      hipLaunchKernelGGL((relu_bwd_colsum_partial_kernel<dev_t>), dim3(ntiles, G),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(dy.data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(yc.data_ptr<scalar_t>()), M, N,
                         rows_per_group, reinterpret_cast<dev_t*>(g.data_ptr<scalar_t>()),
                         partial.data_ptr<float>());
      hipLaunchKernelGGL(colsum_final_f32_kernel, dim3(ntiles), dim3(kBlockThreads), 0,
                         stream, partial.data_ptr<float>(), G, N, db.data_ptr<float>());
    }
  });
  return {g, db};
}

}  // namespace trec_amd
