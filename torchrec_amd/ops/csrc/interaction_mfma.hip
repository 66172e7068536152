// MFMA (matrix-core) DLRM pairwise-dot interaction for MI355X (gfx950).
//
// Same semantics as interaction.hip (reference: torchrec/models/dlrm.py:155
// InteractionArch) but the per-sample GEMMs run on the CDNA4 matrix cores:
//   forward:  Z = T * T^T   (T = [dense; sparse] as [F1, D], F1 <= 32)
//             out[b] = [dense[b], triu(Z, 1)]
//   backward: dT = G * T    (G = symmetrized pair-grad matrix, zero diag)
//             d_dense[b] = dT[0] + dOut[b, :D];  d_sparse[b, i-1] = dT[i]
//
// v_mfma_f32_16x16x32_bf16 tiles, fp32 accumulate. Inputs convert to bf16
// during LDS staging (CDNA4 has no fp32-input MFMA); each wave owns one
// sample. LDS tiles are XOR-swizzled (byte ^= (row&7)<<4) to break the
// row-major D-stride bank conflict on ds_read_b128
// (cdna_hip_programming.md Guideline 4 / technique T2).
//
// Fragment layouts (guide section 3, HW-verified m89):
//   A: lane l holds A[l&15][(l>>4)*8 + e], e = 0..7
//   B: lane l holds B[(l>>4)*8 + e][l&15]
//   C/D: lane l, reg r holds D[(l>>4)*4 + r][l&15]

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include <algorithm>

#include "common.h"

namespace trec_amd {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

static inline hipStream_t imf_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

constexpr int kIWaves = 4;  // waves (= samples in flight) per block

// XOR-swizzle constrained to the row (rb = row stride in bytes, multiple of
// 64): spreads the 16-B slots of up to 8 consecutive rows across banks.
__device__ __forceinline__ int swz(int row, int byte_in_row, int rb) {
  return byte_in_row ^ (((row & 7) << 4) & (rb - 16));
}

// load one A/B fragment from a row-major bf16 LDS tile with swizzled rows.
// row r = r_off + (l&15); bytes [k_byte + (l>>4)*16, +16).
__device__ __forceinline__ bf16x8 load_frag(const char* tile, int row_stride_bytes,
                                            int r_off, int k_byte, int l) {
  int row = r_off + (l & 15);
  int off = row * row_stride_bytes +
            swz(row, k_byte + ((l >> 4) << 4), row_stride_bytes);
  return *reinterpret_cast<const bf16x8*>(tile + off);
}

template <typename io_t>
__global__ void __launch_bounds__(kIWaves * kWaveSize) interaction_mfma_fwd_kernel(
    const io_t* __restrict__ dense,   // [B, D]
    const io_t* __restrict__ sparse,  // [B, F1-1, D]
    const int8_t* __restrict__ pi, const int8_t* __restrict__ pj,  // [P]
    int B, int F1, int D, int P, io_t* __restrict__ out /* [B, D+P] */) {
  extern __shared__ char lds[];
  const int l = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int t_bytes = 32 * D * 2;            // bf16 T tile per wave
  const int z_bytes = 32 * 32 * 4;           // fp32 Z per wave
  char* t_tile = lds + w * (t_bytes + z_bytes);
  float* zbuf = reinterpret_cast<float*>(t_tile + t_bytes);
  const int row_bytes = D * 2;
  const int64_t out_w = D + P;
  const int n_mt = (F1 + 15) / 16;  // 1 or 2 16-row tiles
  const int ksteps = D / 32;

  // zero the pad rows once (per-wave tile): rows F1..31 feed unused Z
  // entries but must not hold NaN bit patterns (0 * NaN = NaN)
  for (int r = F1; r < 32; ++r)
    for (int c = l; c < D / 8; c += 64)
      *reinterpret_cast<bf16x8*>(t_tile + r * row_bytes + swz(r, c * 16, row_bytes)) = bf16x8{};
  __syncthreads();

  const int64_t iters = (B + kIWaves - 1) / kIWaves;
  for (int64_t it = blockIdx.x; it < iters; it += gridDim.x) {
    const int64_t b = it * kIWaves + w;
    const bool active = b < B;
    if (active) {
      // stage T (bf16, swizzled): row 0 = dense, rows 1..F1-1 = sparse.
      // each lane converts 8-element chunks (16 B writes keep the XOR valid).
      const io_t* drow = dense + b * D;
      const io_t* srow = sparse + b * static_cast<int64_t>(F1 - 1) * D;
      const int chunks_per_row = D / 8;
      for (int cidx = l; cidx < F1 * chunks_per_row; cidx += 64) {
        int r = cidx / chunks_per_row, c = cidx - r * chunks_per_row;
        const io_t* src = (r == 0) ? drow + c * 8 : srow + (r - 1) * D + c * 8;
        bf16x8 v;
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = static_cast<__bf16>(emb2float(src[e]));
        *reinterpret_cast<bf16x8*>(t_tile + r * row_bytes + swz(r, c * 16, row_bytes)) = v;
      }
    }
    __syncthreads();
    if (active) {
      // Z = T * T^T. A and B fragments have the SAME gather pattern (B[k][j]
      // = T[j][k]) so the tile (mi, ni) uses frag(mi) x frag(ni).
      f32x4 acc00 = {0.f, 0.f, 0.f, 0.f}, acc01 = acc00, acc11 = acc00;
      for (int k = 0; k < ksteps; ++k) {
        bf16x8 f0 = load_frag(t_tile, row_bytes, 0, k * 64, l);
        acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f0, f0, acc00, 0, 0, 0);
        if (n_mt == 2) {
          bf16x8 f1 = load_frag(t_tile, row_bytes, 16, k * 64, l);
          acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f0, f1, acc01, 0, 0, 0);
          acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(f1, f1, acc11, 0, 0, 0);
        }
      }
      // scatter Z tiles to zbuf: lane l reg r -> Z[(l>>4)*4+r][l&15]
      int zr = (l >> 4) * 4, zc = l & 15;
#pragma unroll
      for (int r = 0; r < 4; ++r) zbuf[(zr + r) * 32 + zc] = acc00[r];
      if (n_mt == 2) {
#pragma unroll
        for (int r = 0; r < 4; ++r) zbuf[(zr + r) * 32 + zc + 16] = acc01[r];
#pragma unroll
        for (int r = 0; r < 4; ++r) zbuf[(zr + r + 16) * 32 + zc + 16] = acc11[r];
      }
    }
    __syncthreads();
    if (active) {
      io_t* orow = out + b * out_w;
      // dense passthrough from the staged tile (row 0)
      for (int c = l; c < D / 8; c += 64) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(t_tile + swz(0, c * 16, row_bytes));
#pragma unroll
        for (int e = 0; e < 8; ++e) orow[c * 8 + e] = float2emb(static_cast<float>(v[e]), io_t{});
      }
      for (int p = l; p < P; p += 64)
        orow[D + p] = float2emb(zbuf[pi[p] * 32 + pj[p]], io_t{});
    }
    __syncthreads();
  }
}

template <typename io_t>
__global__ void __launch_bounds__(kIWaves * kWaveSize) interaction_mfma_bwd_kernel(
    const io_t* __restrict__ grad_out,  // [B, D+P]
    const io_t* __restrict__ dense, const io_t* __restrict__ sparse,
    const int32_t* __restrict__ pair_col,  // [F1*F1], -1 or pair index
    int B, int F1, int D, int P,
    io_t* __restrict__ d_dense, io_t* __restrict__ d_sparse) {
  // ONE sample per block (4 waves cooperate): ~16.5 KB LDS keeps 8 blocks
  // resident per CU (32 waves of latency hiding vs 8 for a wave-per-sample
  // split — this kernel is HBM-latency/-bandwidth bound, not MFMA bound).
  extern __shared__ char lds[];
  const int l = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  constexpr int kStrideE = 40;  // 80 B rows: ds_read_b128 lands ~2-way
  int32_t* pc = reinterpret_cast<int32_t*>(lds);
  char* g_tile = lds + 32 * 32 * 4;
  char* tt_tile = g_tile + 32 * kStrideE * 2;
  const int64_t out_w = D + P;
  const int n_mt = (F1 + 15) / 16;
  const int ntiles = n_mt * (D / 16);

  // sample-invariant init: pc table (padded -1) and Tt pad columns (zeros —
  // the K-pad products are G_pad (=0) * Tt_pad, which must not be NaN)
  for (int e = threadIdx.x; e < 32 * 32; e += blockDim.x) {
    int i = e >> 5, j = e & 31;  // 32-wide pc table index, lint: wave-ok
    pc[e] = (i < F1 && j < F1) ? pair_col[i * F1 + j] : -1;
  }
  if (F1 < 32) {
    const int padw = 32 - F1;
    for (int e = threadIdx.x; e < D * padw; e += blockDim.x) {
      int r = e / padw, c = F1 + e % padw;
      *reinterpret_cast<__bf16*>(tt_tile + r * (kStrideE * 2) + c * 2) =
          static_cast<__bf16>(0.f);
    }
  }
  __syncthreads();

  for (int64_t b = blockIdx.x; b < B; b += gridDim.x) {
    const io_t* grow = grad_out + b * out_w;
    // G[i][j] = dOut[pair(i,j)] (symmetric, zero diagonal / pads): the 1.9 KB
    // grad row is L1-resident, so the scattered rereads are cache-served
    for (int e = threadIdx.x; e < 32 * 32; e += blockDim.x) {
      int i = e >> 5, j = e & 31;  // 32-wide G index, lint: wave-ok
      int32_t p = pc[e];
      *reinterpret_cast<__bf16*>(g_tile + i * (kStrideE * 2) + j * 2) =
          static_cast<__bf16>(p >= 0 ? emb2float(grow[D + p]) : 0.f);
    }
    // Tt[d][f] = T[f][d] (bf16): read T rows vectorized, scatter-transpose
    const io_t* drow = dense + b * D;
    const io_t* srow = sparse + b * static_cast<int64_t>(F1 - 1) * D;
    const int chunks_per_row = D / 4;
    for (int cidx = threadIdx.x; cidx < F1 * chunks_per_row; cidx += blockDim.x) {
      int f = cidx / chunks_per_row, c4 = cidx - f * chunks_per_row;
      const io_t* src = (f == 0) ? drow + c4 * 4 : srow + (f - 1) * D + c4 * 4;
#pragma unroll
      for (int e = 0; e < 4; ++e)
        *reinterpret_cast<__bf16*>(tt_tile + (c4 * 4 + e) * (kStrideE * 2) + f * 2) =
            static_cast<__bf16>(emb2float(src[e]));
    }
    __syncthreads();
    io_t* ddrow = d_dense + b * D;
    io_t* dsrow = d_sparse + b * static_cast<int64_t>(F1 - 1) * D;
    const int zr = (l >> 4) * 4, zc = l & 15;
    for (int t = w; t < ntiles; t += kIWaves) {
      const int mi = t / (D / 16), ni = t - mi * (D / 16);
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          g_tile + (mi * 16 + (l & 15)) * (kStrideE * 2) + ((l >> 4) << 4));
      bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
          tt_tile + (ni * 16 + (l & 15)) * (kStrideE * 2) + ((l >> 4) << 4));
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int i = mi * 16 + zr + r;     // T row (0 = dense)
        int dcol = ni * 16 + zc;      // dim column
        if (i >= F1) continue;
        float v = acc[r];
        if (i == 0) {
          ddrow[dcol] = float2emb(v + emb2float(grow[dcol]), io_t{});
        } else {
          dsrow[static_cast<int64_t>(i - 1) * D + dcol] = float2emb(v, io_t{});
        }
      }
    }
    __syncthreads();  // next sample's staging overwrites G/Tt
  }
}

at::Tensor interaction_mfma_forward(const at::Tensor& dense, const at::Tensor& sparse,
                                    const at::Tensor& pi, const at::Tensor& pj) {
  TORCH_CHECK(dense.is_cuda() && dense.dim() == 2 && sparse.dim() == 3);
  TORCH_CHECK(dense.scalar_type() == sparse.scalar_type(),
              "interaction_mfma: dense/sparse dtype mismatch");
  int B = dense.size(0);
  int D = dense.size(1);
  int F1 = sparse.size(1) + 1;
  int P = pi.numel();
  TORCH_CHECK(D % 32 == 0 && D <= 256, "interaction_mfma needs D % 32 == 0, D <= 256");
  TORCH_CHECK(F1 <= 32, "interaction_mfma needs <= 31 sparse features");
  auto out = at::empty({B, D + P}, dense.options());
  if (B == 0) return out;
  int lds_bytes = kIWaves * (32 * D * 2 + 32 * 32 * 4);
  int grid = std::min<int64_t>((B + kIWaves - 1) / kIWaves, kNumCU * 8);
  auto stream = imf_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dense.scalar_type(),
                                  "interaction_mfma_fwd", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      hipLaunchKernelGGL((interaction_mfma_fwd_kernel<dev_t>), dim3(grid),
                         dim3(kIWaves * kWaveSize), lds_bytes, stream,
                         reinterpret_cast<const dev_t*>(dense.contiguous().data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(sparse.contiguous().data_ptr<scalar_t>()),
                         pi.data_ptr<int8_t>(), pj.data_ptr<int8_t>(), B, F1, D, P,
                         reinterpret_cast<dev_t*>(out.data_ptr<scalar_t>()));
    }
  });
  return out;
}

std::tuple<at::Tensor, at::Tensor> interaction_mfma_backward(
    const at::Tensor& grad_out, const at::Tensor& dense, const at::Tensor& sparse,
    const at::Tensor& pair_col) {
  int B = dense.size(0);
  int D = dense.size(1);
  int F1 = sparse.size(1) + 1;
  int P = grad_out.size(1) - D;
  TORCH_CHECK(D % 32 == 0 && D <= 256 && F1 <= 32);
  TORCH_CHECK(grad_out.scalar_type() == dense.scalar_type());
  auto d_dense = at::empty_like(dense);
  auto d_sparse = at::empty_like(sparse);
  if (B == 0) return {d_dense, d_sparse};
  constexpr int kStrideE = 40;
  int lds_bytes = 32 * 32 * 4 + 32 * kStrideE * 2 + D * kStrideE * 2;
  int grid = std::min<int64_t>(B, kNumCU * 8);
  auto stream = imf_stream();
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kHalf, at::kBFloat16, dense.scalar_type(),
                                  "interaction_mfma_bwd", [&] {
    if constexpr (std::is_same_v<scalar_t, double>) {
      TORCH_CHECK(false, "fp64 unsupported");
    } else {
      using dev_t = typename DevType<scalar_t>::type;
      hipLaunchKernelGGL((interaction_mfma_bwd_kernel<dev_t>), dim3(grid),
                         dim3(kIWaves * kWaveSize), lds_bytes, stream,
                         reinterpret_cast<const dev_t*>(grad_out.contiguous().data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(dense.contiguous().data_ptr<scalar_t>()),
                         reinterpret_cast<const dev_t*>(sparse.contiguous().data_ptr<scalar_t>()),
                         pair_col.data_ptr<int32_t>(), B, F1, D, P,
                         reinterpret_cast<dev_t*>(d_dense.data_ptr<scalar_t>()),
                         reinterpret_cast<dev_t*>(d_sparse.data_ptr<scalar_t>()));
    }
  });
  return {d_dense, d_sparse};
}

}  // namespace trec_amd
