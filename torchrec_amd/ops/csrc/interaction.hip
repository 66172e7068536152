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

#include <algorithm>
#include <type_traits>

#include "common.h"

namespace trec_amd {

static inline hipStream_t ia_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// pair p (i<j) column = D + p; host precomputes i_of_pair / j_of_pair.

__global__ void __launch_bounds__(kBlockThreads) interaction_fwd_kernel(
    const float* __restrict__ dense,   // [B, D]
    const float* __restrict__ sparse,  // [B, F, D]
    const int8_t* __restrict__ pi,     // [P] row i of pair p
    const int8_t* __restrict__ pj,     // [P] row j of pair p
    int B, int F1, int D, int P, float* __restrict__ out /* [B, D+P] */) {
  // 256 threads (4 waves) per sample: staging and the P pair-dots spread
  // across the whole block (351 pairs -> ~1.4 per lane instead of 5.5 on a
  // single wave). The grid-stride trip count depends only on blockIdx, so
  // __syncthreads() inside the loop is uniform per block.
  // LDS tile is float4-strided with a +1 float4 pad: rows stay 16B-aligned
  // for ds_read_b128 while the pad staggers banks across rows.
  extern __shared__ float lds[];  // [F1][D/4+1] float4
  int tid = threadIdx.x;
  float4* T4 = reinterpret_cast<float4*>(lds);
  const int d4 = D / 4;
  const int stride4 = d4 + 1;
  int64_t out_w = D + P;
  for (int64_t b = blockIdx.x; b < B; b += gridDim.x) {
    // stage T: row 0 = dense, rows 1..F1-1 = sparse
    const float4* drow = reinterpret_cast<const float4*>(dense + b * D);
    for (int t = tid; t < d4; t += blockDim.x) T4[t] = drow[t];
    const float4* srow = reinterpret_cast<const float4*>(sparse + b * (int64_t)(F1 - 1) * D);
    for (int t = tid; t < (F1 - 1) * d4; t += blockDim.x) {
      int r = t / d4, c = t - r * d4;
      T4[(r + 1) * stride4 + c] = srow[t];
    }
    __syncthreads();
    float* orow = out + b * out_w;
    float4* orow4 = reinterpret_cast<float4*>(orow);
    for (int t = tid; t < d4; t += blockDim.x) orow4[t] = T4[t];
    for (int p = tid; p < P; p += blockDim.x) {
      const float4* Ti = T4 + pi[p] * stride4;
      const float4* Tj = T4 + pj[p] * stride4;
      float acc = 0.f;
      for (int k = 0; k < d4; ++k) {
        float4 a = Ti[k], c = Tj[k];
        acc += a.x * c.x + a.y * c.y + a.z * c.z + a.w * c.w;
      }
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
  extern __shared__ float lds[];  // [F1][D/4+1] float4, [P] dz, [F1*F1] cols
  float4* T4 = reinterpret_cast<float4*>(lds);
  float* dzbuf = lds + 4 * F1 * (D / 4 + 1);
  int32_t* pc = reinterpret_cast<int32_t*>(dzbuf + P);
  int l = lane_id();
  int wave = wave_id();
  const int d4 = D / 4;
  const int stride4 = d4 + 1;
  int64_t out_w = D + P;
  // pair_col is sample-invariant: stage it once per block
  for (int t = threadIdx.x; t < F1 * F1; t += blockDim.x) pc[t] = pair_col[t];
  __syncthreads();
  for (int64_t b = blockIdx.x; b < B; b += gridDim.x) {
    const float4* drow = reinterpret_cast<const float4*>(dense + b * D);
    for (int t = threadIdx.x; t < d4; t += blockDim.x) T4[t] = drow[t];
    const float4* srow = reinterpret_cast<const float4*>(sparse + b * (int64_t)(F1 - 1) * D);
    for (int t = threadIdx.x; t < (F1 - 1) * d4; t += blockDim.x) {
      int r = t / d4, c = t - r * d4;
      T4[(r + 1) * stride4 + c] = srow[t];
    }
    const float* grow = grad_out + b * out_w;
    // stage the pair-gradient row once (each i-row re-reads F1-1 of them)
    for (int t = threadIdx.x; t < P; t += blockDim.x) dzbuf[t] = grow[D + t];
    __syncthreads();
    // when a row needs <= 32 float4 columns, each half-wave takes its own
    // row so no lanes idle (D=128 -> rows i and i+ROWS_PER_WAVE/2)
    const int halves = (d4 <= kWaveSize / 2) ? 2 : 1;
    const int rows_per_iter = (kBlockThreads / kWaveSize) * halves;
    const int half = (halves == 2) ? (l >> 5) : 0;
    const int lk = (halves == 2) ? (l & 31) : l;  // half-wave split, lint: wave-ok
    for (int i = wave * halves + half; i < F1; i += rows_per_iter) {
      // lanes cover D/4 float4 columns
      for (int k = lk; k < d4; k += kWaveSize / halves) {
        float4 acc;
        if (i == 0) {
          acc = reinterpret_cast<const float4*>(grow)[k];
        } else {
          acc = make_float4(0.f, 0.f, 0.f, 0.f);
        }
        for (int j = 0; j < F1; ++j) {
          int c = pc[i * F1 + j];
          if (c >= 0) {
            float dz = dzbuf[c];
            float4 t = T4[j * stride4 + k];
            acc.x += dz * t.x;
            acc.y += dz * t.y;
            acc.z += dz * t.z;
            acc.w += dz * t.w;
          }
        }
        if (i == 0) {
          reinterpret_cast<float4*>(d_dense + b * D)[k] = acc;
        } else {
          reinterpret_cast<float4*>(d_sparse + (b * (int64_t)(F1 - 1) + (i - 1)) * D)[k] = acc;
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
  TORCH_CHECK(D % 4 == 0, "interaction kernel needs D %% 4 == 0");
  int lds_bytes = F1 * (D / 4 + 1) * sizeof(float4);
  int grid = std::min<int>(B, kNumCU * 32);
  hipLaunchKernelGGL(interaction_fwd_kernel, dim3(grid), dim3(kBlockThreads), lds_bytes, ia_stream(),
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
  TORCH_CHECK(D % 4 == 0, "interaction kernel needs D %% 4 == 0");
  int lds_bytes = F1 * (D / 4 + 1) * sizeof(float4) + P * sizeof(float)
                  + F1 * F1 * sizeof(int32_t);
  int grid = std::min<int>(B, kMaxBlocks);
  hipLaunchKernelGGL(interaction_bwd_kernel, dim3(grid), dim3(kBlockThreads), lds_bytes,
                     ia_stream(), grad_out.contiguous().data_ptr<float>(),
                     dense.contiguous().data_ptr<float>(),
                     sparse.contiguous().data_ptr<float>(),
                     pair_col_plus_d.data_ptr<int32_t>(), B, F1, D, P,
                     d_dense.data_ptr<float>(), d_sparse.data_ptr<float>());
  return {d_dense, d_sparse};
}


// ---------------------------------------------------------------------------
// deterministic column sum (bias gradient): two-phase fixed-partition
// reduction over the batch dim. torch's bf16 column reduce_kernel takes
// ~17 us per layer at [8192, 1024]; this is a plain HBM-bound sweep.
// ---------------------------------------------------------------------------

template <typename scalar_t, int VPT>
__global__ void __launch_bounds__(kBlockThreads) col_sum_partial_kernel(
    const scalar_t* __restrict__ in, int64_t M, int64_t N, int rows_per_group,
    float* __restrict__ partial /* [G, N] */) {
  // VPT consecutive columns per thread -> one 16 B load per row for 16-bit
  // dtypes (wave reads 1 KB contiguous), fp32 accumulate
  int tile = blockIdx.x;
  int g = blockIdx.y;
  int64_t c0 = (static_cast<int64_t>(tile) * kBlockThreads + threadIdx.x) * VPT;
  if (c0 >= N) return;
  int64_t r0 = static_cast<int64_t>(g) * rows_per_group;
  int64_t r1 = min(M, r0 + rows_per_group);
  float acc[VPT];
#pragma unroll
  for (int v = 0; v < VPT; ++v) acc[v] = 0.f;
  if (c0 + VPT <= N) {
    for (int64_t r = r0; r < r1; ++r) {
      const scalar_t* row = in + r * N + c0;
      if constexpr (VPT == 4 && sizeof(scalar_t) == 2) {
        uint2 q = *reinterpret_cast<const uint2*>(row);  // 4 halfs, 8 B
        const scalar_t* h = reinterpret_cast<const scalar_t*>(&q);
#pragma unroll
        for (int v = 0; v < 4; ++v) acc[v] += emb2float(h[v]);
      } else if constexpr (VPT == 4 && sizeof(scalar_t) == 4) {
        uint4 q = *reinterpret_cast<const uint4*>(row);  // 4 floats, 16 B
        const scalar_t* h = reinterpret_cast<const scalar_t*>(&q);
#pragma unroll
        for (int v = 0; v < 4; ++v) acc[v] += emb2float(h[v]);
      } else {
#pragma unroll
        for (int v = 0; v < VPT; ++v) acc[v] += emb2float(row[v]);
      }
    }
  } else {
    for (int64_t r = r0; r < r1; ++r)
      for (int v = 0; v < VPT && c0 + v < N; ++v)
        acc[v] += emb2float(in[r * N + c0 + v]);
  }
  float* prow = partial + static_cast<int64_t>(g) * N + c0;
#pragma unroll
  for (int v = 0; v < VPT; ++v)
    if (c0 + v < N) prow[v] = acc[v];
}

template <typename scalar_t>
__global__ void __launch_bounds__(kBlockThreads) col_sum_final_kernel(
    const float* __restrict__ partial, int G, int64_t N,
    scalar_t* __restrict__ out) {
  int64_t c = static_cast<int64_t>(blockIdx.x) * kBlockThreads + threadIdx.x;
  if (c >= N) return;
  float acc = 0.f;
  for (int g = 0; g < G; ++g) acc += partial[static_cast<int64_t>(g) * N + c];
  if constexpr (std::is_same_v<scalar_t, float>) {
    out[c] = acc;
  } else if constexpr (std::is_same_v<scalar_t, __half>) {
    out[c] = __float2half(acc);
  } else {
    out[c] = __float2bfloat16(acc);
  }
}

at::Tensor col_sum(const at::Tensor& input) {
  TORCH_CHECK(input.dim() == 2 && input.is_cuda());
  int64_t M = input.size(0), N = input.size(1);
  auto out = at::empty({N}, input.options());
  if (N == 0) return out;
  auto in = input.contiguous();
  constexpr int kVPT = 1;  // scalar columns: measured best at [8192, 1024] bf16
  int ntiles = (int)((N + kBlockThreads - 1) / kBlockThreads);
  int G = std::max(1, std::min<int>(kMaxBlocks / std::max(ntiles, 1), (int)((M + 31) / 32)));
  int rows_per_group = (int)((M + G - 1) / G);
  int ftiles = ntiles;  // final pass covers kBlockThreads columns per block
  auto partial = at::empty({(int64_t)G * N}, input.options().dtype(at::kFloat));
  auto stream = ia_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, input.scalar_type(),
                                  "col_sum", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = std::conditional_t<
          std::is_same_v<scalar_t, at::Half>, __half,
          std::conditional_t<std::is_same_v<scalar_t, at::BFloat16>, __hip_bfloat16,
                             float>>;
      hipLaunchKernelGGL((col_sum_partial_kernel<dev_t, kVPT>), dim3(ntiles, G),
                         dim3(kBlockThreads), 0, stream,
                         reinterpret_cast<const dev_t*>(in.data_ptr<scalar_t>()), M, N,
                         rows_per_group, partial.data_ptr<float>());
      hipLaunchKernelGGL((col_sum_final_kernel<dev_t>), dim3(ftiles),
                         dim3(kBlockThreads), 0, stream, partial.data_ptr<float>(), G, N,
                         reinterpret_cast<dev_t*>(out.data_ptr<scalar_t>()));
    }
  });
  return out;
}

}  // namespace trec_amd

