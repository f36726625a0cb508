// Internal launcher API between embedding_ops.hip (kernels) and bindings.cpp
// (torch glue).  gfx950-only; no CUDA-compat paths.
#pragma once

#include <hip/hip_runtime.h>

#include <cstddef>
#include <cstdint>

void launch_csr_lookup_forward(const void* params, bool params_bf16,
                               const int64_t* values, const int64_t* splits,
                               const float* per_id_w, void* out, bool out_bf16,
                               int64_t num_rows, int64_t nnz, int64_t vocab,
                               int width, bool mean, int64_t* long_rows,
                               int32_t* long_count, int64_t* work_items,
                               int32_t* n_work, hipStream_t stream);

void launch_row_to_split(const int64_t* rows, int64_t nnz, int64_t num_rows,
                         int64_t* splits, hipStream_t stream);

size_t csr_backward_temp_bytes(int64_t nnz, int64_t vocab);

void launch_expand_row_ids(const int64_t* splits, int64_t num_rows,
                           int64_t nnz, int32_t* row_ids, float* w, bool mean,
                           hipStream_t stream);

hipError_t run_inclusive_scan_i32(void* temp, size_t temp_bytes,
                                  const int32_t* in, int32_t* out, int64_t n,
                                  hipStream_t stream);

void launch_mark_heads(const int64_t* sorted_ids, int64_t n, int64_t vocab,
                       int32_t* head, hipStream_t stream);

void launch_scatter_unique(const int64_t* sorted_ids, const int32_t* head,
                           const int32_t* pos, int64_t n, int64_t vocab,
                           int64_t* unique_ids, int64_t* seg_offsets,
                           int32_t* num_unique, hipStream_t stream);

void launch_gather_sorted(const int32_t* perm, const int32_t* row_ids,
                          const float* w, int64_t n, int64_t* srow, float* sw,
                          hipStream_t stream);

void launch_set_seg_end(int64_t* seg_offsets, const int32_t* num_unique,
                        const int64_t* bounds, hipStream_t stream);

void launch_find_valid_bounds(const int64_t* sorted_ids, int64_t n,
                              int64_t vocab, int64_t* bounds,
                              hipStream_t stream);

void launch_integer_lookup(const int64_t* keys, int64_t n, int64_t* tkeys,
                           int64_t* tvals, int64_t capacity, int32_t* counts,
                           int64_t max_tokens, void* temp, size_t temp_bytes,
                           int32_t* scratch_flags, int32_t* scratch_pos,
                           int64_t* avail, int32_t* navail, int32_t* next_avail,
                           int64_t* out, hipStream_t stream);

size_t integer_lookup_temp_bytes(int64_t max_tokens);

void launch_hash_reinsert(const int64_t* old_keys, const int64_t* old_vals,
                          int64_t old_cap, int64_t* tkeys, int64_t* tvals,
                          int64_t capacity, hipStream_t stream);

void launch_pad_seg_offsets(int64_t* seg, int64_t n, const int32_t* num_unique,
                            const int64_t* bounds, hipStream_t stream);

void launch_sorted_optimizer_update(void* weight, bool weight_bf16,
                                    float* state, float eps,
                                    const int64_t* sorted_ids,
                                    const int64_t* seg, const int64_t* srow,
                                    const float* sw, const void* grad_out,
                                    bool grad_bf16,
                                    const float* lr, const int32_t* nu_ptr,
                                    int64_t max_segs, int width,
                                    int64_t* long_rows, int32_t* long_count,
                                    int64_t* work_items, int32_t* n_work,
                                    float* long_scratch, int64_t scratch_rows,
                                    bool adagrad, hipStream_t stream);

void launch_sparse_row_update(void* weight, bool weight_bf16, float* state,
                              const int64_t* ids, const float* grad,
                              int64_t num_rows, int width, float lr, float eps,
                              bool adagrad, hipStream_t stream);

void launch_dot_interact_fwd_packed(const void* bottom, const void* packed,
                                    const int* perm, void* out, int64_t B,
                                    int F, int D, int out_w, int tri_n,
                                    int64_t sb, int64_t sp,
                                    hipStream_t stream);
void launch_dot_interact_bwd_packed(const void* gout, const void* bottom,
                                    const void* packed, const int* perm,
                                    void* gbottom, void* gpacked, int64_t B,
                                    int F, int D, int out_w, int tri_n,
                                    int64_t sb, int64_t sp,
                                    hipStream_t stream);
void launch_dot_interact_fwd(const void* feats, void* out, int64_t B, int F,
                             int D, int out_w, int tri_n, hipStream_t stream);
void launch_dot_interact_bwd(const void* gout, const void* feats, void* gfeats,
                             int64_t B, int F, int D, int out_w, int tri_n,
                             hipStream_t stream);

size_t custom_radix_sort_hist_elems(int64_t n);

void custom_radix_sort_keys(const uint64_t* keys_in, uint64_t* keys_out,
                            uint64_t* keys_tmp, int32_t* hist,
                            int32_t* scan_sums, int64_t n, int begin_bit,
                            int end_bit, hipStream_t stream);

void launch_mask_oob_pack(const int64_t* ids, int64_t n, int64_t vocab,
                          uint64_t* packed, hipStream_t stream);
void launch_unpack_sorted(const uint64_t* packed, int64_t n,
                          int64_t* sorted_ids, int32_t* sorted_pos,
                          hipStream_t stream);
hipError_t run_sort_keys_u64(void* temp, size_t temp_bytes,
                             const uint64_t* keys_in, uint64_t* keys_out,
                             int64_t n, int begin_bit, int end_bit,
                             hipStream_t stream);
size_t rocprim_sort_keys_temp_bytes(int64_t n);
