// torch.ops.trec_amd.* registration + Python module init.

#include <torch/extension.h>

#include <tuple>

namespace trec_amd {

// jagged_ops.hip
at::Tensor complete_cumsum(const at::Tensor& lengths);
at::Tensor lengths_range(const at::Tensor& offsets);
std::tuple<at::Tensor, at::Tensor, at::Tensor> permute_2d_sparse_data(
    const at::Tensor& permute, const at::Tensor& lengths, const at::Tensor& values,
    const at::Tensor& weights, int64_t out_size_hint);
at::Tensor jagged_to_padded_dense(const at::Tensor& values, const at::Tensor& offsets,
                                  int64_t max_length, double padding_value);
at::Tensor dense_to_jagged(const at::Tensor& dense, const at::Tensor& offsets);
at::Tensor segment_sum_csr(const at::Tensor& csr, const at::Tensor& values);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor, at::Tensor>
block_bucketize_sparse_features(const at::Tensor& lengths, const at::Tensor& indices,
                                const at::Tensor& block_sizes, int64_t num_buckets,
                                bool bucketize_pos, bool sequence, const at::Tensor& weights,
                                const at::Tensor& bag_feature_bounds);
at::Tensor permute_pooled_embs(const at::Tensor& values, const at::Tensor& in_offsets,
                               const at::Tensor& out_offsets, const at::Tensor& order);

// tbe.hip
at::Tensor tbe_forward_pooled(const at::Tensor& weights, const at::Tensor& table_elem_offsets,
                              const at::Tensor& dims, const at::Tensor& feat_table,
                              const at::Tensor& d_out_offsets, const at::Tensor& indices,
                              const at::Tensor& offsets, const at::Tensor& per_sample_weights,
                              int64_t B, int64_t total_D, int64_t max_D, bool mean_pool,
                              const at::Tensor& cache_weights, const at::Tensor& cache_loc,
                              int64_t out_dtype);
at::Tensor tbe_forward_seq(const at::Tensor& weights, const at::Tensor& table_elem_offsets,
                           const at::Tensor& dims, const at::Tensor& feat_table,
                           const at::Tensor& feat_val_offsets, const at::Tensor& indices,
                           int64_t D_out, int64_t max_D, int64_t out_dtype);
at::Tensor tbe_forward_pooled_vbe(const at::Tensor& weights,
                                  const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                                  const at::Tensor& feat_table, const at::Tensor& bag_offsets,
                                  const at::Tensor& out_offsets, const at::Tensor& indices,
                                  const at::Tensor& offsets,
                                  const at::Tensor& per_sample_weights, int64_t n_bags,
                                  int64_t out_numel, int64_t max_D, bool mean_pool);
std::tuple<at::Tensor, at::Tensor> sort_pairs(const at::Tensor& keys, int64_t end_bit);
std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs(
    const at::Tensor& linear, const at::Tensor& offsets, int64_t B, int64_t F,
    int64_t end_bit, int64_t capacity);
std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs_2level(
    const at::Tensor& linear, const at::Tensor& offsets, int64_t B, int64_t F,
    int64_t end_bit, int64_t capacity);
std::tuple<at::Tensor, at::Tensor, at::Tensor> seg_sort_pairs_large(
    const at::Tensor& linear, const at::Tensor& feat_bounds, int64_t F, int64_t end_bit);
std::tuple<at::Tensor, at::Tensor> tbe_backward_prep(const at::Tensor& sorted_linear);
std::tuple<at::Tensor, at::Tensor, at::Tensor> tbe_bag_metadata(
    const at::Tensor& offsets, const at::Tensor& indices,
    const at::Tensor& feat_d_out, const at::Tensor& feat_row_offset, int64_t B);
void tbe_backward_fused(at::Tensor weights, at::Tensor momentum, const at::Tensor& grad,
                        const at::Tensor& sorted_linear, const at::Tensor& sort_perm,
                        const at::Tensor& seg_offsets, const at::Tensor& num_runs,
                        const at::Tensor& pos_row, const at::Tensor& pos_col,
                        const at::Tensor& pos_scale, const at::Tensor& table_row_offsets,
                        const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                        int64_t max_D, double lr, double eps, int64_t mode,
                        at::Tensor grad_weights, at::Tensor cache_weights,
                        const at::Tensor& cache_loc,
                        at::Tensor m1, at::Tensor m2, double beta1, double beta2,
                        const at::Tensor& iter_t, const at::Tensor& rng_state,
                        bool stochastic);
at::Tensor gather_run_heads(const at::Tensor& sorted_linear, const at::Tensor& seg_offsets,
                            const at::Tensor& num_runs);
at::Tensor tbe_grad_per_sample_weights(const at::Tensor& weights,
                                       const at::Tensor& table_elem_offsets,
                                       const at::Tensor& dims, const at::Tensor& grad,
                                       const at::Tensor& indices, const at::Tensor& pos_row,
                                       const at::Tensor& pos_col, const at::Tensor& pos_table,
                                       int64_t max_D);
at::Tensor bounds_check_indices(at::Tensor indices, const at::Tensor& feat_val_offsets,
                                const at::Tensor& rows, const at::Tensor& feat_table);

// cache.hip
void lxu_cache_populate(at::Tensor host_weights, const at::Tensor& table_row_offsets,
                        const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                        const at::Tensor& sorted_uniq_ids, const at::Tensor& seg_offsets,
                        const at::Tensor& num_runs, at::Tensor cache_weights,
                        at::Tensor cache_tags, at::Tensor cache_lru, int64_t max_D,
                        int64_t timestamp);
at::Tensor lxu_cache_lookup(const at::Tensor& ids, const at::Tensor& cache_tags);
at::Tensor hash_zch_remap(const at::Tensor& ids, at::Tensor identity,
                          at::Tensor metadata, int64_t max_probe, int64_t stamp,
                          bool train);
at::Tensor hash_zch_evict(at::Tensor identity, at::Tensor metadata, int64_t older_than);
void lxu_cache_flush(at::Tensor host_weights, const at::Tensor& table_row_offsets,
                     const at::Tensor& table_elem_offsets, const at::Tensor& dims,
                     const at::Tensor& cache_weights, const at::Tensor& cache_tags,
                     int64_t max_D);

// quant_tbe.hip
at::Tensor quantize_rowwise_int8(const at::Tensor& weights);
at::Tensor tbe_forward_pooled_int8(const at::Tensor& qweights,
                                   const at::Tensor& table_byte_offsets, const at::Tensor& dims,
                                   const at::Tensor& feat_table, const at::Tensor& d_out_offsets,
                                   const at::Tensor& indices, const at::Tensor& offsets,
                                   const at::Tensor& per_sample_weights, int64_t B,
                                   int64_t total_D, int64_t max_D, bool mean_pool);
at::Tensor tbe_forward_seq_int8(const at::Tensor& qweights,
                                const at::Tensor& table_byte_offsets, const at::Tensor& dims,
                                const at::Tensor& feat_table,
                                const at::Tensor& feat_val_offsets, const at::Tensor& indices,
                                int64_t D_out, int64_t max_D);

// interaction.hip
at::Tensor col_sum(const at::Tensor& input);
// mlp_ops.hip
at::Tensor lt_linear_relu_fwd(const at::Tensor& x, const at::Tensor& w,
                              const at::Tensor& b);
std::tuple<at::Tensor, at::Tensor> lt_wgrad_bgrad(const at::Tensor& g,
                                                  const at::Tensor& x);
at::Tensor relu_bwd_mask(const at::Tensor& grad_out, const at::Tensor& y);
at::Tensor bce_with_logits_fwd(const at::Tensor& logits, const at::Tensor& labels);
at::Tensor bce_with_logits_bwd(const at::Tensor& logits, const at::Tensor& labels,
                               const at::Tensor& grad_out);
std::tuple<at::Tensor, at::Tensor> relu_bwd_col_sum(const at::Tensor& grad_out,
                                                    const at::Tensor& y);
// interaction_mfma.hip
at::Tensor interaction_mfma_forward(const at::Tensor& dense, const at::Tensor& sparse,
                                    const at::Tensor& pi, const at::Tensor& pj);
std::tuple<at::Tensor, at::Tensor> interaction_mfma_backward(const at::Tensor& grad_out,
                                                             const at::Tensor& dense,
                                                             const at::Tensor& sparse,
                                                             const at::Tensor& pair_col);
at::Tensor interaction_forward(const at::Tensor& dense, const at::Tensor& sparse,
                               const at::Tensor& pi, const at::Tensor& pj);
std::tuple<at::Tensor, at::Tensor> interaction_backward(const at::Tensor& grad_out,
                                                        const at::Tensor& dense,
                                                        const at::Tensor& sparse,
                                                        const at::Tensor& pair_col);

}  // namespace trec_amd

TORCH_LIBRARY(trec_amd, m) {
  m.def("complete_cumsum(Tensor lengths) -> Tensor");
  m.def("lengths_range(Tensor offsets) -> Tensor");
  m.def(
      "permute_2d_sparse_data(Tensor permute, Tensor lengths, Tensor values, Tensor weights,"
      " int out_size_hint) -> (Tensor, Tensor, Tensor)");
  m.def(
      "jagged_to_padded_dense(Tensor values, Tensor offsets, int max_length,"
      " float padding_value) -> Tensor");
  m.def("dense_to_jagged(Tensor dense, Tensor offsets) -> Tensor");
  m.def("segment_sum_csr(Tensor csr, Tensor values) -> Tensor");
  m.def(
      "block_bucketize_sparse_features(Tensor lengths, Tensor indices, Tensor block_sizes,"
      " int num_buckets, bool bucketize_pos, bool sequence, Tensor weights,"
      " Tensor bag_feature_bounds) -> (Tensor, Tensor, Tensor, Tensor, Tensor)");
  m.def(
      "permute_pooled_embs(Tensor values, Tensor in_offsets, Tensor out_offsets,"
      " Tensor order) -> Tensor");
  m.def(
      "tbe_forward_pooled(Tensor weights, Tensor table_elem_offsets, Tensor dims,"
      " Tensor feat_table, Tensor d_out_offsets, Tensor indices, Tensor offsets,"
      " Tensor per_sample_weights, int B, int total_D, int max_D, bool mean_pool,"
      " Tensor cache_weights, Tensor cache_loc, int out_dtype) -> Tensor");
  m.def(
      "tbe_forward_seq(Tensor weights, Tensor table_elem_offsets, Tensor dims,"
      " Tensor feat_table, Tensor feat_val_offsets, Tensor indices, int D_out, int max_D,"
      " int out_dtype) -> Tensor");
  m.def(
      "tbe_forward_pooled_vbe(Tensor weights, Tensor table_elem_offsets, Tensor dims,"
      " Tensor feat_table, Tensor bag_offsets, Tensor out_offsets, Tensor indices,"
      " Tensor offsets, Tensor per_sample_weights, int n_bags, int out_numel, int max_D,"
      " bool mean_pool) -> Tensor");
  m.def("sort_pairs(Tensor keys, int end_bit) -> (Tensor, Tensor)");
  m.def(
      "seg_sort_pairs(Tensor linear, Tensor offsets, int B, int F, int end_bit, "
      "int capacity) -> (Tensor, Tensor, Tensor)");
  m.def(
      "seg_sort_pairs_2level(Tensor linear, Tensor offsets, int B, int F, int end_bit, "
      "int capacity) -> (Tensor, Tensor, Tensor)");
  m.def(
      "seg_sort_pairs_large(Tensor linear, Tensor feat_bounds, int F, int end_bit)"
      " -> (Tensor, Tensor, Tensor)");
  m.def("tbe_backward_prep(Tensor sorted_linear) -> (Tensor, Tensor)");
  m.def(
      "tbe_bag_metadata(Tensor offsets, Tensor indices, Tensor feat_d_out,"
      " Tensor feat_row_offset, int B) -> (Tensor, Tensor, Tensor)");
  m.def(
      "tbe_backward_fused(Tensor(a!) weights, Tensor(b!) momentum, Tensor grad,"
      " Tensor sorted_linear, Tensor sort_perm, Tensor seg_offsets, Tensor num_runs,"
      " Tensor pos_row, Tensor pos_col, Tensor pos_scale, Tensor table_row_offsets,"
      " Tensor table_elem_offsets, Tensor dims, int max_D, float lr, float eps, int mode,"
      " Tensor(c!) grad_weights, Tensor(d!) cache_weights, Tensor cache_loc,"
      " Tensor(e!) m1, Tensor(f!) m2, float beta1, float beta2, Tensor iter_t,"
      " Tensor rng_state, bool stochastic) -> ()");
  m.def("gather_run_heads(Tensor sorted_linear, Tensor seg_offsets, Tensor num_runs) -> Tensor");
  m.def(
      "lxu_cache_populate(Tensor(a!) host_weights, Tensor table_row_offsets,"
      " Tensor table_elem_offsets, Tensor dims, Tensor sorted_uniq_ids, Tensor seg_offsets,"
      " Tensor num_runs, Tensor(b!) cache_weights, Tensor(c!) cache_tags,"
      " Tensor(d!) cache_lru, int max_D, int timestamp) -> ()");
  m.def("lxu_cache_lookup(Tensor ids, Tensor cache_tags) -> Tensor");
  m.def(
      "hash_zch_remap(Tensor ids, Tensor(a!) identity, Tensor(b!) metadata, "
      "int max_probe, int stamp, bool train) -> Tensor");
  m.def("hash_zch_evict(Tensor(a!) identity, Tensor(b!) metadata, int older_than) -> Tensor");
  m.def(
      "lxu_cache_flush(Tensor(a!) host_weights, Tensor table_row_offsets,"
      " Tensor table_elem_offsets, Tensor dims, Tensor cache_weights, Tensor cache_tags,"
      " int max_D) -> ()");
  m.def(
      "tbe_grad_per_sample_weights(Tensor weights, Tensor table_elem_offsets, Tensor dims,"
      " Tensor grad, Tensor indices, Tensor pos_row, Tensor pos_col, Tensor pos_table,"
      " int max_D) -> Tensor");
  m.def(
      "bounds_check_indices(Tensor(a!) indices, Tensor feat_val_offsets, Tensor rows,"
      " Tensor feat_table) -> Tensor");
  m.def("quantize_rowwise_int8(Tensor weights) -> Tensor");
  m.def(
      "tbe_forward_pooled_int8(Tensor qweights, Tensor table_byte_offsets, Tensor dims,"
      " Tensor feat_table, Tensor d_out_offsets, Tensor indices, Tensor offsets,"
      " Tensor per_sample_weights, int B, int total_D, int max_D, bool mean_pool) -> Tensor");
  m.def(
      "tbe_forward_seq_int8(Tensor qweights, Tensor table_byte_offsets, Tensor dims,"
      " Tensor feat_table, Tensor feat_val_offsets, Tensor indices, int D_out, int max_D)"
      " -> Tensor");
  m.def("col_sum(Tensor input) -> Tensor");
  m.def("interaction_forward(Tensor dense, Tensor sparse, Tensor pi, Tensor pj) -> Tensor");
  m.def("relu_bwd_col_sum(Tensor grad_out, Tensor y) -> (Tensor, Tensor)");
  m.def("lt_linear_relu_fwd(Tensor x, Tensor w, Tensor b) -> Tensor");
  m.def("lt_wgrad_bgrad(Tensor g, Tensor x) -> (Tensor, Tensor)");
  m.def("relu_bwd_mask(Tensor grad_out, Tensor y) -> Tensor");
  m.def("bce_with_logits_fwd(Tensor logits, Tensor labels) -> Tensor");
  m.def("bce_with_logits_bwd(Tensor logits, Tensor labels, Tensor grad_out) -> Tensor");
  m.def("interaction_mfma_forward(Tensor dense, Tensor sparse, Tensor pi, Tensor pj) -> Tensor");
  m.def(
      "interaction_mfma_backward(Tensor grad_out, Tensor dense, Tensor sparse, Tensor pair_col)"
      " -> (Tensor, Tensor)");
  m.def(
      "interaction_backward(Tensor grad_out, Tensor dense, Tensor sparse, Tensor pair_col)"
      " -> (Tensor, Tensor)");
}

TORCH_LIBRARY_IMPL(trec_amd, CUDA, m) {
  m.impl("complete_cumsum", trec_amd::complete_cumsum);
  m.impl("lengths_range", trec_amd::lengths_range);
  m.impl("permute_2d_sparse_data", trec_amd::permute_2d_sparse_data);
  m.impl("jagged_to_padded_dense", trec_amd::jagged_to_padded_dense);
  m.impl("dense_to_jagged", trec_amd::dense_to_jagged);
  m.impl("segment_sum_csr", trec_amd::segment_sum_csr);
  m.impl("block_bucketize_sparse_features", trec_amd::block_bucketize_sparse_features);
  m.impl("permute_pooled_embs", trec_amd::permute_pooled_embs);
  m.impl("tbe_forward_pooled", trec_amd::tbe_forward_pooled);
  m.impl("tbe_forward_seq", trec_amd::tbe_forward_seq);
  m.impl("tbe_forward_pooled_vbe", trec_amd::tbe_forward_pooled_vbe);
  m.impl("sort_pairs", trec_amd::sort_pairs);
  m.impl("seg_sort_pairs", trec_amd::seg_sort_pairs);
  m.impl("seg_sort_pairs_2level", trec_amd::seg_sort_pairs_2level);
  m.impl("seg_sort_pairs_large", trec_amd::seg_sort_pairs_large);
  m.impl("tbe_backward_prep", trec_amd::tbe_backward_prep);
  m.impl("tbe_bag_metadata", trec_amd::tbe_bag_metadata);
  m.impl("tbe_backward_fused", trec_amd::tbe_backward_fused);
  m.impl("tbe_grad_per_sample_weights", trec_amd::tbe_grad_per_sample_weights);
  m.impl("gather_run_heads", trec_amd::gather_run_heads);
  m.impl("lxu_cache_populate", trec_amd::lxu_cache_populate);
  m.impl("lxu_cache_lookup", trec_amd::lxu_cache_lookup);
  m.impl("hash_zch_remap", trec_amd::hash_zch_remap);
  m.impl("hash_zch_evict", trec_amd::hash_zch_evict);
  m.impl("lxu_cache_flush", trec_amd::lxu_cache_flush);
  m.impl("bounds_check_indices", trec_amd::bounds_check_indices);
  m.impl("quantize_rowwise_int8", trec_amd::quantize_rowwise_int8);
  m.impl("tbe_forward_pooled_int8", trec_amd::tbe_forward_pooled_int8);
  m.impl("tbe_forward_seq_int8", trec_amd::tbe_forward_seq_int8);
  m.impl("col_sum", trec_amd::col_sum);
  m.impl("interaction_forward", trec_amd::interaction_forward);
  m.impl("relu_bwd_col_sum", trec_amd::relu_bwd_col_sum);
  m.impl("lt_linear_relu_fwd", trec_amd::lt_linear_relu_fwd);
  m.impl("lt_wgrad_bgrad", trec_amd::lt_wgrad_bgrad);
  m.impl("relu_bwd_mask", trec_amd::relu_bwd_mask);
  m.impl("bce_with_logits_fwd", trec_amd::bce_with_logits_fwd);
  m.impl("bce_with_logits_bwd", trec_amd::bce_with_logits_bwd);
  m.impl("interaction_mfma_forward", trec_amd::interaction_mfma_forward);
  m.impl("interaction_mfma_backward", trec_amd::interaction_mfma_backward);
  m.impl("interaction_backward", trec_amd::interaction_backward);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("loaded", []() { return true; });
  m.attr("__hip_arch__") = "gfx950";
}
