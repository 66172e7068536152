// Fused DLRM pairwise-dot interaction for MI355X (gfx950).
//
// Replaces the eager chain cat -> bmm(T, T^T) -> triu gather -> cat (and its
// backward: index scatter + two bmms) with two kernels:
//   forward:  out[b] = [dense[b], {dot(T_i, T_j)}_{i<j}]  where
//             T = [dense[b]; sparse[b, 0..F-1]]  (F1 = F+1 rows, D cols)
//   backward: dT_i = sum_{j != i} dz_{ij} * T_j ; d_dense += dOut[:, :D]
//
// Per-sample tiles are staged in LDS with a +1-element row pad to break the
// D-stride bank conflict (cdna_hip_programming.md Guideline 4); one wave per
// sample (fwd) / one block per sample with one wave per 4 rows (bwd).
// Reference semantics: torchrec/models/dlrm.py:155 InteractionArch.

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "common.h"

namespace trec_amd {

static inline hipStream_t ia_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// pair p (i<j) column = D + p; host precomputes i_of_pair / j_of_pair.

__global__ void __launch_bounds__(64) interaction_fwd_kernel(
    const float* __restrict__ dense,   // [B, D]
    const float* __restrict__ sparse,  // [B, F, D]
    const int8_t* __restrict__ pi,     // [P] row i of pair p
    const int8_t* __restrict__ pj,     // [P] row j of pair p
    int B, int F1, int D, int P, float* __restrict__ out /* [B, D+P] */) {
  // ONE wave per block: __syncthreads() is a cheap wave-local fence, no
  // cross-wave iteration-count hazards in the grid-stride loop.
  extern __shared__ float lds[];  // [F1 * (D+1)]
  int l = lane_id();
  float* T = lds;
  int64_t out_w = D + P;
  for (int64_t b = blockIdx.x; b < B; b += gridDim.x) {
    // stage T: row 0 = dense, rows 1..F1-1 = sparse (+1 pad kills the
    // D-stride bank conflict on the dot reads)
    for (int t = l; t < D; t += kWaveSize) T[t] = dense[b * D + t];
    for (int t = l; t < (F1 - 1) * D; t += kWaveSize) {
      int r = t / D, c = t - r * D;
      T[(r + 1) * (D + 1) + c] = sparse[(b * (F1 - 1) + r) * D + c];
    }
    __syncthreads();
    float* orow = out + b * out_w;
    for (int t = l; t < D; t += kWaveSize) orow[t] = T[t];
    for (int p = l; p < P; p += kWaveSize) {
      const float* Ti = T + pi[p] * (D + 1);
      const float* Tj = T + pj[p] * (D + 1);
      float acc = 0.f;
      for (int d = 0; d < D; ++d) acc += Ti[d] * Tj[d];
      orow[D + p] = acc;
    }
    __syncthreads();
  }
}

__global__ void __launch_bounds__(kBlockThreads) interaction_bwd_kernel(
    const float* __restrict__ grad_out,  // [B, D+P]
    const float* __restrict__ dense,
    const float* __restrict__ sparse,
    const int32_t* __restrict__ pair_col,  // [F1*F1] col in grad_out or -1
    int B, int F1, int D, int P,
    float* __restrict__ d_dense,   // [B, D]
    float* __restrict__ d_sparse   // [B, F, D]
) {
  extern __shared__ float lds[];  // [F1 * (D+1)]
  float* T = lds;
  int l = lane_id();
  int wave = wave_id();
  int64_t out_w = D + P;
  for (int64_t b = blockIdx.x; b < B; b += gridDim.x) {
    for (int t = threadIdx.x; t < D; t += blockDim.x) T[t] = dense[b * D + t];
    for (int t = threadIdx.x; t < (F1 - 1) * D; t += blockDim.x) {
      int r = t / D, c = t - r * D;
      T[(r + 1) * (D + 1) + c] = sparse[(b * (F1 - 1) + r) * D + c];
    }
    __syncthreads();
    const float* grow = grad_out + b * out_w;
    for (int i = wave; i < F1; i += kBlockThreads / kWaveSize) {
      // lanes cover D
      for (int d0 = l; d0 < D; d0 += kWaveSize) {
        float acc = (i == 0) ? grow[d0] : 0.f;
        for (int j = 0; j < F1; ++j) {
          int c = pair_col[i * F1 + j];
          if (c >= 0) acc += grow[D + c] * T[j * (D + 1) + d0];
        }
        if (i == 0) {
          d_dense[b * D + d0] = acc;
        } else {
          d_sparse[(b * (F1 - 1) + (i - 1)) * D + d0] = acc;
        }
      }
    }
    __syncthreads();
  }
}

at::Tensor interaction_forward(const at::Tensor& dense, const at::Tensor& sparse,
                               const at::Tensor& pi, const at::Tensor& pj) {
  TORCH_CHECK(dense.is_cuda() && dense.dim() == 2 && sparse.dim() == 3);
  TORCH_CHECK(dense.scalar_type() == at::kFloat && sparse.scalar_type() == at::kFloat);
  int B = dense.size(0);
  int D = dense.size(1);
  int F1 = sparse.size(1) + 1;
  int P = pi.numel();
  auto out = at::empty({B, D + P}, dense.options());
  if (B == 0) return out;
  int lds_bytes = F1 * (D + 1) * sizeof(float);
  int grid = std::min<int>(B, kNumCU * 16);
  hipLaunchKernelGGL(interaction_fwd_kernel, dim3(grid), dim3(64), lds_bytes, ia_stream(),
                     dense.contiguous().data_ptr<float>(),
                     sparse.contiguous().data_ptr<float>(), pi.data_ptr<int8_t>(),
                     pj.data_ptr<int8_t>(), B, F1, D, P, out.data_ptr<float>());
  return out;
}

std::tuple<at::Tensor, at::Tensor> interaction_backward(
    const at::Tensor& grad_out, const at::Tensor& dense, const at::Tensor& sparse,
    const at::Tensor& pair_col_plus_d) {
  int B = dense.size(0);
  int D = dense.size(1);
  int F1 = sparse.size(1) + 1;
  int P = grad_out.size(1) - D;
  auto d_dense = at::empty_like(dense);
  auto d_sparse = at::empty_like(sparse);
  if (B == 0) return {d_dense, d_sparse};
  int lds_bytes = F1 * (D + 1) * sizeof(float);
  int grid = std::min<int>(B, kMaxBlocks);
  hipLaunchKernelGGL(interaction_bwd_kernel, dim3(grid), dim3(kBlockThreads), lds_bytes,
                     ia_stream(), grad_out.contiguous().data_ptr<float>(),
                     dense.contiguous().data_ptr<float>(),
                     sparse.contiguous().data_ptr<float>(),
                     pair_col_plus_d.data_ptr<int32_t>(), B, F1, D, P,
                     d_dense.data_ptr<float>(), d_sparse.data_ptr<float>());
  return {d_dense, d_sparse};
}

}  // namespace trec_amd
