// INT8 rowwise-quantized inference TBE for MI355X (gfx950).
//
// MI355X-native equivalent of the reference's IntNBitTableBatchedEmbeddingBags
// (reference torchrec/distributed/quant_embedding_kernel.py:237; kernel spec:
// triton_tbe _nbit_TBE_forward_kernel :3051 with per-row scale/bias and
// cacheline-aligned rows :3429).
//
// Row format: [uint8 x D][fp16 scale][fp16 bias], padded to 16 B so rows load
// as uchar4/uint4. Dequant: w = q * scale + bias.

#include <torch/extension.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_fp16.h>

#include "common.h"

namespace trec_amd {

// UVM-capable pointer: host-pinned packed tables are addressed by the
// kernels over PCIe (QUANT_UVM compute kernel)
template <typename T>
static T* uvm_ptr(const at::Tensor& t) {
  if (t.numel() == 0) return nullptr;
  if (t.is_cuda()) return t.data_ptr<T>();
  TORCH_CHECK(t.is_pinned(), "quant TBE host-resident tensors must be pinned");
  void* dp = nullptr;
  TREC_HIP_CHECK(hipHostGetDevicePointer(&dp, t.data_ptr(), 0));
  return static_cast<T*>(dp);
}


static inline hipStream_t q_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

__host__ __device__ inline int64_t int8_row_stride(int D) {
  return ((D + 4 /*scale+bias*/ + 15) / 16) * 16;
}

// ---------------------------------------------------------------------------
// fp32 [R, D] -> int8 rowwise quantized [R, row_stride] bytes
// ---------------------------------------------------------------------------

__global__ void quantize_rowwise_int8_kernel(const float* __restrict__ w, int64_t R, int D,
                                             int64_t stride, uint8_t* __restrict__ out) {
  // one wave per row
  int64_t wave = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / kWaveSize;
  int64_t n_waves = (static_cast<int64_t>(gridDim.x) * blockDim.x) / kWaveSize;
  int l = lane_id();
  for (int64_t r = wave; r < R; r += n_waves) {
    const float* row = w + r * D;
    float mn = INFINITY, mx = -INFINITY;
    for (int d = l; d < D; d += kWaveSize) {
      float v = row[d];
      mn = fminf(mn, v);
      mx = fmaxf(mx, v);
    }
#pragma unroll
    for (int off = kWaveSize / 2; off > 0; off >>= 1) {
      mn = fminf(mn, __shfl_xor(mn, off, kWaveSize));
      mx = fmaxf(mx, __shfl_xor(mx, off, kWaveSize));
    }
    float scale = (mx - mn) / 255.f;
    float inv = scale > 0.f ? 1.f / scale : 0.f;
    uint8_t* orow = out + r * stride;
    for (int d = l; d < D; d += kWaveSize) {
      float q = (row[d] - mn) * inv;
      orow[d] = static_cast<uint8_t>(fminf(fmaxf(q + 0.5f, 0.f), 255.f));
    }
    if (l == 0) {
      __half* sb = reinterpret_cast<__half*>(orow + D);
      sb[0] = __float2half(scale);
      sb[1] = __float2half(mn);
    }
  }
}

at::Tensor quantize_rowwise_int8(const at::Tensor& weights) {
  TORCH_CHECK(weights.is_cuda() && weights.dim() == 2 && weights.scalar_type() == at::kFloat);
  int64_t R = weights.size(0);
  int D = weights.size(1);
  int64_t stride = int8_row_stride(D);
  auto out = at::empty({R, stride}, weights.options().dtype(at::kByte));
  if (R == 0) return out;
  auto w = weights.contiguous();
  hipLaunchKernelGGL(quantize_rowwise_int8_kernel,
                     dim3(grid_for(R * kWaveSize, kBlockThreads)), dim3(kBlockThreads), 0,
                     q_stream(), w.data_ptr<float>(), R, D, stride,
                     out.data_ptr<uint8_t>());
  return out;
}

// ---------------------------------------------------------------------------
// pooled int8 forward: same slot scheme as the fp32 TBE
// ---------------------------------------------------------------------------

template <int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_fwd_pooled_int8_kernel(
    const uint8_t* __restrict__ qweights,
    const int64_t* __restrict__ table_byte_offsets,  // [T]
    const int32_t* __restrict__ dims,                // [T]
    const int32_t* __restrict__ feat_table,          // [F]
    const int64_t* __restrict__ d_out_offsets,       // [F+1]
    const int64_t* __restrict__ indices,
    const int64_t* __restrict__ offsets,  // [F*B+1]
    const float* __restrict__ psw, int F, int B, int64_t total_D, bool mean_pool,
    float* __restrict__ out) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  int64_t n_bags = static_cast<int64_t>(F) * B;
  for (int64_t bag = slot; bag < n_bags; bag += n_slots) {
    int f = bag / B;
    int b = bag - static_cast<int64_t>(f) * B;
    int t = feat_table[f];
    int D = dims[t];
    int64_t stride = int8_row_stride(D);
    const uint8_t* tab = qweights + table_byte_offsets[t];
    float4 acc[CHUNKS];
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) acc[c] = make_float4(0.f, 0.f, 0.f, 0.f);
    int64_t i0 = offsets[bag], i1 = offsets[bag + 1];
    for (int64_t i = i0; i < i1; ++i) {
      const uint8_t* row = tab + indices[i] * stride;
      const __half* sb = reinterpret_cast<const __half*>(row + D);
      float scale = __half2float(sb[0]);
      float bias = __half2float(sb[1]);
      float w = psw ? psw[i] : 1.f;
#pragma unroll
      for (int c = 0; c < CHUNKS; ++c) {
        int col4 = c * LPS + sl;
        if (col4 * 4 < D) {
          uchar4 q = reinterpret_cast<const uchar4*>(row)[col4];
          acc[c].x += w * (q.x * scale + bias);
          acc[c].y += w * (q.y * scale + bias);
          acc[c].z += w * (q.z * scale + bias);
          acc[c].w += w * (q.w * scale + bias);
        }
      }
    }
    float s = 1.f;
    if (mean_pool && i1 > i0) s = 1.f / static_cast<float>(i1 - i0);
    float4* orow =
        reinterpret_cast<float4*>(out + static_cast<int64_t>(b) * total_D + d_out_offsets[f]);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int col4 = c * LPS + sl;
      if (col4 * 4 < D)
        orow[col4] = make_float4(acc[c].x * s, acc[c].y * s, acc[c].z * s, acc[c].w * s);
    }
  }
}

at::Tensor tbe_forward_pooled_int8(
    const at::Tensor& qweights, const at::Tensor& table_byte_offsets, const at::Tensor& dims,
    const at::Tensor& feat_table, const at::Tensor& d_out_offsets, const at::Tensor& indices,
    const at::Tensor& offsets, const at::Tensor& per_sample_weights, int64_t B,
    int64_t total_D, int64_t max_D, bool mean_pool) {
  TORCH_CHECK((qweights.is_cuda() || qweights.is_pinned()) && qweights.scalar_type() == at::kByte);
  TORCH_CHECK(max_D % 4 == 0 && max_D <= 2048);
  int F = feat_table.numel();
  auto out = at::empty({B, total_D}, indices.options().dtype(at::kFloat));
  if (B == 0 || F == 0) return out;
  const float* psw_ptr =
      per_sample_weights.numel() > 0 ? per_sample_weights.data_ptr<float>() : nullptr;
  auto stream = q_stream();
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  int grid = grid_for(static_cast<int64_t>(F) * B * lps, kBlockThreads);
#define TBE_Q_LAUNCH(LPS, CHUNKS)                                                         \
  hipLaunchKernelGGL((tbe_fwd_pooled_int8_kernel<LPS, CHUNKS>), dim3(grid),               \
                     dim3(kBlockThreads), 0, stream, uvm_ptr<uint8_t>(qweights),           \
                     table_byte_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(),    \
                     feat_table.data_ptr<int32_t>(), d_out_offsets.data_ptr<int64_t>(),   \
                     indices.data_ptr<int64_t>(), offsets.data_ptr<int64_t>(), psw_ptr,   \
                     F, (int)B, total_D, mean_pool, out.data_ptr<float>())
  if (lps == 16) TBE_Q_LAUNCH(16, 1);
  else if (lps == 32) TBE_Q_LAUNCH(32, 1);
  else switch (chunks) {
    case 1: TBE_Q_LAUNCH(64, 1); break;
    case 2: TBE_Q_LAUNCH(64, 2); break;
    case 3: case 4: TBE_Q_LAUNCH(64, 4); break;
    default: TBE_Q_LAUNCH(64, 8); break;
  }
#undef TBE_Q_LAUNCH
  return out;
}

// sequence int8 forward: out[n] = dequant(row(idx[n]))
template <int LPS, int CHUNKS>
__global__ void __launch_bounds__(kBlockThreads) tbe_fwd_seq_int8_kernel(
    const uint8_t* __restrict__ qweights, const int64_t* __restrict__ table_byte_offsets,
    const int32_t* __restrict__ dims, const int32_t* __restrict__ feat_table,
    const int64_t* __restrict__ feat_val_offsets, const int64_t* __restrict__ indices, int F,
    int64_t N, int64_t D_out, float* __restrict__ out) {
  int sl = threadIdx.x % LPS;
  int64_t slot = (static_cast<int64_t>(blockIdx.x) * blockDim.x + threadIdx.x) / LPS;
  int64_t n_slots = (static_cast<int64_t>(gridDim.x) * blockDim.x) / LPS;
  for (int64_t n = slot; n < N; n += n_slots) {
    int f = upper_bound_segment(feat_val_offsets, F, n);
    int t = feat_table[f];
    int D = dims[t];
    int64_t stride = int8_row_stride(D);
    const uint8_t* row = qweights + table_byte_offsets[t] + indices[n] * stride;
    const __half* sb = reinterpret_cast<const __half*>(row + D);
    float scale = __half2float(sb[0]);
    float bias = __half2float(sb[1]);
    float4* orow = reinterpret_cast<float4*>(out + n * D_out);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      int col4 = c * LPS + sl;
      if (col4 * 4 < D) {
        uchar4 q = reinterpret_cast<const uchar4*>(row)[col4];
        orow[col4] =
            make_float4(q.x * scale + bias, q.y * scale + bias, q.z * scale + bias,
                        q.w * scale + bias);
      }
    }
  }
}

at::Tensor tbe_forward_seq_int8(
    const at::Tensor& qweights, const at::Tensor& table_byte_offsets, const at::Tensor& dims,
    const at::Tensor& feat_table, const at::Tensor& feat_val_offsets, const at::Tensor& indices,
    int64_t D_out, int64_t max_D) {
  TORCH_CHECK((qweights.is_cuda() || qweights.is_pinned()) && max_D % 4 == 0 && max_D <= 2048);
  int64_t N = indices.numel();
  auto out = at::empty({N, D_out}, indices.options().dtype(at::kFloat));
  if (N == 0) return out;
  int F = feat_table.numel();
  auto stream = q_stream();
  int lps = (max_D <= 64) ? 16 : (max_D <= 128 ? 32 : 64);
  int chunks = (int)((max_D + lps * 4 - 1) / (lps * 4));
  int grid = grid_for(N * lps, kBlockThreads);
#define TBE_QS_LAUNCH(LPS, CHUNKS)                                                        \
  hipLaunchKernelGGL((tbe_fwd_seq_int8_kernel<LPS, CHUNKS>), dim3(grid),                  \
                     dim3(kBlockThreads), 0, stream, uvm_ptr<uint8_t>(qweights),           \
                     table_byte_offsets.data_ptr<int64_t>(), dims.data_ptr<int32_t>(),    \
                     feat_table.data_ptr<int32_t>(), feat_val_offsets.data_ptr<int64_t>(),\
                     indices.data_ptr<int64_t>(), F, N, D_out, out.data_ptr<float>())
  if (lps == 16) TBE_QS_LAUNCH(16, 1);
  else if (lps == 32) TBE_QS_LAUNCH(32, 1);
  else switch (chunks) {
    case 1: TBE_QS_LAUNCH(64, 1); break;
    case 2: TBE_QS_LAUNCH(64, 2); break;
    case 3: case 4: TBE_QS_LAUNCH(64, 4); break;
    default: TBE_QS_LAUNCH(64, 8); break;
  }
#undef TBE_QS_LAUNCH
  return out;
}

}  // namespace trec_amd
