// Torch bindings for the gfx950 HIP kernels (in-tree extension
// distributed_embeddings_amd._hip_ops).  Host-side orchestration of the
// backward pipeline lives here (the one deliberate D2H sync: num_unique —
// parity with the reference GradFunctor, embedding_lookup_kernels.cu:663-667).

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>

#include "ops_api.h"

namespace {

#define CHECK_CUDA(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")

inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

int log2_ceil(int64_t v) {
  int b = 0;
  while ((int64_t(1) << b) < v) ++b;
  return b;
}


// Sort dispatch: hand-written LSD radix sort (csrc/radix_sort.hip) by
// default; DE_USE_ROCPRIM_SORT=1 switches to the rocPRIM bring-up path for
// comparison.
bool use_rocprim_sort() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("DE_USE_ROCPRIM_SORT");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}

// Sorts ids (masking OOB to the `vocab` sentinel) by packing
// (id << 32 | position) into one u64 array; returns sorted_ids + the sorted
// original positions.  Hand-written sort by default, rocPRIM via env.
void sort_ids_dispatch(torch::Tensor values, int64_t vocab, int64_t nnz,
                       int end_bit, torch::Tensor sorted_ids,
                       torch::Tensor sorted_pos, hipStream_t stream) {
  TORCH_CHECK(end_bit <= 32, "fused table vocab must fit 32 bits");
  TORCH_CHECK(nnz < (int64_t(1) << 32),
              "sort packs positions into 32 bits: nnz must be < 2^32");
  auto u64 = values.options().dtype(torch::kUInt64);
  auto i32 = values.options().dtype(torch::kInt32);
  auto packed = torch::empty({nnz}, u64);
  launch_mask_oob_pack(values.data_ptr<int64_t>(), nnz, vocab,
                       (uint64_t*)packed.data_ptr(), stream);
  auto packed_out = torch::empty({nnz}, u64);
  if (use_rocprim_sort()) {
    size_t temp_bytes = rocprim_sort_keys_temp_bytes(nnz);
    auto temp = torch::empty({(int64_t)temp_bytes},
                             values.options().dtype(torch::kUInt8));
    auto err = run_sort_keys_u64(temp.data_ptr(), temp_bytes,
                                 (const uint64_t*)packed.data_ptr(),
                                 (uint64_t*)packed_out.data_ptr(), nnz, 32,
                                 32 + end_bit, stream);
    TORCH_CHECK(err == hipSuccess, "radix_sort_keys failed");
  } else {
    auto keys_tmp = torch::empty({nnz}, u64);
    const int64_t hist_elems = (int64_t)custom_radix_sort_hist_elems(nnz);
    auto hist = torch::empty({hist_elems}, i32);
    auto scan_sums = torch::empty({4096}, i32);
    custom_radix_sort_keys((const uint64_t*)packed.data_ptr(),
                           (uint64_t*)packed_out.data_ptr(),
                           (uint64_t*)keys_tmp.data_ptr(),
                           hist.data_ptr<int32_t>(),
                           scan_sums.data_ptr<int32_t>(), nnz, 32,
                           32 + end_bit, stream);
  }
  launch_unpack_sorted((const uint64_t*)packed_out.data_ptr(), nnz,
                       sorted_ids.data_ptr<int64_t>(),
                       sorted_pos.data_ptr<int32_t>(), stream);
}

torch::Tensor csr_lookup_forward(torch::Tensor params, torch::Tensor values,
                                 torch::Tensor row_splits, bool mean,
                                 bool out_bf16) {
  CHECK_CUDA(params); CHECK_CUDA(values); CHECK_CUDA(row_splits);
  CHECK_CONTIG(params); CHECK_CONTIG(values); CHECK_CONTIG(row_splits);
  TORCH_CHECK(params.dtype() == torch::kFloat32 ||
                  params.dtype() == torch::kBFloat16,
              "params must be fp32 or bf16");
  TORCH_CHECK(values.dtype() == torch::kInt64, "values must be int64");
  TORCH_CHECK(row_splits.dtype() == torch::kInt64, "row_splits must be int64");
  const int64_t num_rows = row_splits.numel() - 1;
  const int64_t vocab = params.size(0);
  const int width = (int)params.size(1);
  // accumulation is fp32 regardless of table/output storage dtype; bf16 out
  // stores RNE-rounded results directly (no separate cast kernel).  The
  // long-segment split needs fp32 atomics, so bf16-out launches reduce
  // every row in-register instead (fine for bounded forward hotness).
  auto out = torch::empty({num_rows, width},
                          params.options().dtype(
                              out_bf16 ? torch::kBFloat16 : torch::kFloat32));
  if (num_rows > 0) {
    const int64_t nnz_in = values.numel();
    auto long_rows = torch::empty({num_rows}, values.options());
    auto long_count = torch::empty({1}, values.options().dtype(torch::kInt32));
    auto work_items = torch::empty({nnz_in / 64 + 64}, values.options());
    auto n_work = torch::empty({1}, values.options().dtype(torch::kInt32));
    launch_csr_lookup_forward(params.data_ptr(),
                              params.dtype() == torch::kBFloat16,
                              values.data_ptr<int64_t>(),
                              row_splits.data_ptr<int64_t>(), nullptr,
                              out.data_ptr(), out_bf16, num_rows, nnz_in,
                              vocab, width, mean,
                              long_rows.data_ptr<int64_t>(),
                              long_count.data_ptr<int32_t>(),
                              work_items.data_ptr<int64_t>(),
                              n_work.data_ptr<int32_t>(), current_stream());
  }
  return out;
}

torch::Tensor row_to_split(torch::Tensor rows, int64_t num_rows) {
  CHECK_CUDA(rows); CHECK_CONTIG(rows);
  TORCH_CHECK(rows.dtype() == torch::kInt64);
  auto splits = torch::empty({num_rows + 1}, rows.options());
  launch_row_to_split(rows.data_ptr<int64_t>(), rows.numel(), num_rows,
                      splits.data_ptr<int64_t>(), current_stream());
  return splits;
}

std::vector<torch::Tensor> csr_lookup_backward(torch::Tensor grad_out,
                                               torch::Tensor values,
                                               torch::Tensor row_splits,
                                               int64_t vocab, bool mean) {
  CHECK_CUDA(grad_out); CHECK_CUDA(values); CHECK_CUDA(row_splits);
  CHECK_CONTIG(grad_out); CHECK_CONTIG(values); CHECK_CONTIG(row_splits);
  TORCH_CHECK(grad_out.dtype() == torch::kFloat32 ||
                  grad_out.dtype() == torch::kBFloat16,
              "grad_out must be fp32 or bf16");
  const int64_t nnz = values.numel();
  const int64_t num_rows = row_splits.numel() - 1;
  const int width = (int)grad_out.size(1);
  auto stream = current_stream();
  auto i64 = values.options();
  auto i32 = values.options().dtype(torch::kInt32);
  auto f32 = grad_out.options().dtype(torch::kFloat32);

  if (nnz == 0) {
    return {torch::empty({0}, i64), torch::empty({0, width}, f32)};
  }

  // 1+3 fused below: mask OOB -> sentinel and sort packed (id, pos).

  // 2. per-id row ids (+ mean weights keyed by original position).
  auto row_ids = torch::empty({nnz}, i32);
  torch::Tensor w;
  float* w_ptr = nullptr;
  if (mean) {
    w = torch::empty({nnz}, f32);
    w_ptr = w.data_ptr<float>();
  }
  launch_expand_row_ids(row_splits.data_ptr<int64_t>(), num_rows, nnz,
                        row_ids.data_ptr<int32_t>(), w_ptr, mean, stream);

  // 3. radix sort (ids, position) — end_bit covers [0, vocab] inclusive.
  auto sorted_ids = torch::empty({nnz}, i64);
  auto sorted_pos = torch::empty({nnz}, i32);
  sort_ids_dispatch(values, vocab, nnz, log2_ceil(vocab + 1), sorted_ids,
                    sorted_pos, stream);
  size_t temp_bytes = csr_backward_temp_bytes(nnz, vocab);
  auto temp = torch::empty({(int64_t)temp_bytes},
                           f32.dtype(torch::kUInt8));

  // 4. permute row ids (+ weights) into sorted order, widened to i64.
  auto srow = torch::empty({nnz}, i64);
  torch::Tensor sw;
  float* sw_ptr = nullptr;
  if (mean) {
    sw = torch::empty({nnz}, f32);
    sw_ptr = sw.data_ptr<float>();
  }
  launch_gather_sorted(sorted_pos.data_ptr<int32_t>(),
                       row_ids.data_ptr<int32_t>(), w_ptr, nnz,
                       srow.data_ptr<int64_t>(), sw_ptr, stream);

  // 5. unique-by-key via head flags + scan.
  auto head = torch::empty({nnz}, i32);
  auto pos = torch::empty({nnz}, i32);
  launch_mark_heads(sorted_ids.data_ptr<int64_t>(), nnz, vocab,
                    head.data_ptr<int32_t>(), stream);
  auto err = run_inclusive_scan_i32(temp.data_ptr(), temp_bytes,
                               head.data_ptr<int32_t>(),
                               pos.data_ptr<int32_t>(), nnz, stream);
  TORCH_CHECK(err == hipSuccess, "inclusive_scan failed");

  auto unique_tmp = torch::empty({nnz}, i64);
  auto seg_tmp = torch::empty({nnz + 1}, i64);
  auto num_unique_dev = torch::zeros({1}, i32);
  launch_scatter_unique(sorted_ids.data_ptr<int64_t>(),
                        head.data_ptr<int32_t>(), pos.data_ptr<int32_t>(), nnz,
                        vocab, unique_tmp.data_ptr<int64_t>(),
                        seg_tmp.data_ptr<int64_t>(),
                        num_unique_dev.data_ptr<int32_t>(), stream);
  auto bounds = torch::empty({2}, i64);
  launch_find_valid_bounds(sorted_ids.data_ptr<int64_t>(), nnz, vocab,
                           bounds.data_ptr<int64_t>(), stream);
  launch_set_seg_end(seg_tmp.data_ptr<int64_t>(),
                     num_unique_dev.data_ptr<int32_t>(),
                     bounds.data_ptr<int64_t>(), stream);

  // 6. the one host sync: number of unique ids (output allocation).
  const int64_t nu = num_unique_dev.to(torch::kCPU).item<int32_t>();

  auto unique_ids = unique_tmp.narrow(0, 0, nu).contiguous();
  auto unique_grad = torch::empty({nu, width}, f32);
  if (nu > 0) {
    // 7. segmented sum == forward gather-reduce over grad_out rows.
    auto long_rows = torch::empty({nu}, i64);
    auto long_count = torch::empty({1}, i32);
    auto work_items = torch::empty({nnz / 64 + 64}, i64);
    auto n_work = torch::empty({1}, i32);
    launch_csr_lookup_forward(grad_out.data_ptr(),
                              grad_out.dtype() == torch::kBFloat16,
                              srow.data_ptr<int64_t>(),
                              seg_tmp.data_ptr<int64_t>(), sw_ptr,
                              unique_grad.data_ptr<float>(), false, nu, nnz,
                              grad_out.size(0), width, /*mean=*/false,
                              long_rows.data_ptr<int64_t>(),
                              long_count.data_ptr<int32_t>(),
                              work_items.data_ptr<int64_t>(),
                              n_work.data_ptr<int32_t>(), stream);
  }
  return {unique_ids, unique_grad};
}

torch::Tensor integer_lookup(torch::Tensor keys, torch::Tensor table_keys,
                             torch::Tensor table_values, torch::Tensor counts,
                             int64_t max_tokens) {
  CHECK_CUDA(keys); CHECK_CUDA(table_keys); CHECK_CUDA(table_values);
  CHECK_CUDA(counts);
  CHECK_CONTIG(keys); CHECK_CONTIG(table_keys); CHECK_CONTIG(table_values);
  CHECK_CONTIG(counts);
  const int64_t n = keys.numel();
  const int64_t capacity = table_keys.numel();
  auto i32 = keys.options().dtype(torch::kInt32);
  auto i64 = keys.options();
  auto out = torch::empty({n}, i64);
  if (n == 0) return out;
  const int64_t nvals = max_tokens + 1;
  size_t temp_bytes = integer_lookup_temp_bytes(max_tokens);
  auto temp = torch::empty({(int64_t)temp_bytes}, i32.dtype(torch::kUInt8));
  auto flags = torch::empty({nvals}, i32);
  auto pos = torch::empty({nvals}, i32);
  auto avail = torch::empty({nvals}, i64);
  auto navail = torch::zeros({1}, i32);
  auto next_avail = torch::zeros({1}, i32);
  launch_integer_lookup(keys.data_ptr<int64_t>(), n,
                        table_keys.data_ptr<int64_t>(),
                        table_values.data_ptr<int64_t>(), capacity,
                        counts.data_ptr<int32_t>(), max_tokens,
                        temp.data_ptr(), temp_bytes, flags.data_ptr<int32_t>(),
                        pos.data_ptr<int32_t>(), avail.data_ptr<int64_t>(),
                        navail.data_ptr<int32_t>(),
                        next_avail.data_ptr<int32_t>(),
                        out.data_ptr<int64_t>(), current_stream());
  return out;
}

void hash_rehash(torch::Tensor old_keys, torch::Tensor old_values,
                 torch::Tensor new_keys, torch::Tensor new_values) {
  CHECK_CUDA(old_keys); CHECK_CUDA(old_values);
  CHECK_CUDA(new_keys); CHECK_CUDA(new_values);
  CHECK_CONTIG(old_keys); CHECK_CONTIG(old_values);
  CHECK_CONTIG(new_keys); CHECK_CONTIG(new_values);
  TORCH_CHECK(new_keys.numel() >= old_keys.numel(),
              "rehash target must not be smaller");
  launch_hash_reinsert(old_keys.data_ptr<int64_t>(),
                       old_values.data_ptr<int64_t>(), old_keys.numel(),
                       new_keys.data_ptr<int64_t>(),
                       new_values.data_ptr<int64_t>(), new_keys.numel(),
                       current_stream());
}

void sparse_row_update(torch::Tensor weight, torch::Tensor state,
                       torch::Tensor ids, torch::Tensor grad, double lr,
                       double eps, bool adagrad) {
  CHECK_CUDA(weight); CHECK_CUDA(ids); CHECK_CUDA(grad);
  CHECK_CONTIG(weight); CHECK_CONTIG(ids); CHECK_CONTIG(grad);
  TORCH_CHECK(weight.dtype() == torch::kFloat32 ||
              weight.dtype() == torch::kBFloat16);
  TORCH_CHECK(grad.dtype() == torch::kFloat32);
  TORCH_CHECK(ids.dtype() == torch::kInt64);
  const int64_t n = ids.numel();
  if (n == 0) return;
  float* state_ptr = nullptr;
  if (adagrad) {
    CHECK_CUDA(state); CHECK_CONTIG(state);
    TORCH_CHECK(state.sizes() == weight.sizes(), "adagrad state shape mismatch");
    TORCH_CHECK(state.dtype() == torch::kFloat32, "adagrad state must be fp32");
    state_ptr = state.data_ptr<float>();
  }
  launch_sparse_row_update(weight.data_ptr(),
                           weight.dtype() == torch::kBFloat16, state_ptr,
                           ids.data_ptr<int64_t>(), grad.data_ptr<float>(), n,
                           (int)weight.size(1), (float)lr, (float)eps, adagrad,
                           current_stream());
}

void csr_fused_optimizer_apply(torch::Tensor weight, torch::Tensor state,
                               torch::Tensor values, torch::Tensor row_splits,
                               torch::Tensor grad_out, torch::Tensor lr,
                               bool mean, bool adagrad, double eps) {
  // All-device sorted pipeline (no host sync; hipGraph-capturable):
  // sort (id, pos) -> permute row-ids -> head-flag unique -> segment pad ->
  // one direct update per unique row (long segments chunked with one atomic
  // per partial).  Replaces both torch sparse grads and the naive atomic
  // scatter (hot-row same-address chains).
  CHECK_CUDA(weight); CHECK_CUDA(values); CHECK_CUDA(row_splits);
  CHECK_CUDA(grad_out); CHECK_CUDA(lr);
  CHECK_CONTIG(weight); CHECK_CONTIG(values); CHECK_CONTIG(row_splits);
  CHECK_CONTIG(grad_out);
  TORCH_CHECK(weight.dtype() == torch::kFloat32 ||
              weight.dtype() == torch::kBFloat16);
  TORCH_CHECK((grad_out.dtype() == torch::kFloat32 ||
               grad_out.dtype() == torch::kBFloat16) &&
              lr.dtype() == torch::kFloat32);
  const bool wbf16 = weight.dtype() == torch::kBFloat16;
  const int64_t num_rows = row_splits.numel() - 1;
  const int64_t nnz = values.numel();
  const int64_t vocab = weight.size(0);
  const int width = (int)weight.size(1);
  if (num_rows <= 0 || nnz <= 0) return;
  auto stream = current_stream();
  auto i64 = values.options();
  auto i32 = values.options().dtype(torch::kInt32);
  auto f32 = grad_out.options().dtype(torch::kFloat32);

  auto row_ids = torch::empty({nnz}, i32);
  torch::Tensor w;
  float* w_ptr = nullptr;
  if (mean) {
    w = torch::empty({nnz}, f32);
    w_ptr = w.data_ptr<float>();
  }
  launch_expand_row_ids(row_splits.data_ptr<int64_t>(), num_rows, nnz,
                        row_ids.data_ptr<int32_t>(), w_ptr, mean, stream);
  auto sorted_ids = torch::empty({nnz}, i64);
  auto sorted_pos = torch::empty({nnz}, i32);
  sort_ids_dispatch(values, vocab, nnz, log2_ceil(vocab + 1), sorted_ids,
                    sorted_pos, stream);
  size_t temp_bytes = csr_backward_temp_bytes(nnz, vocab);
  auto temp = torch::empty({(int64_t)temp_bytes}, f32.dtype(torch::kUInt8));
  auto srow = torch::empty({nnz}, i64);
  torch::Tensor sw;
  float* sw_ptr = nullptr;
  if (mean) {
    sw = torch::empty({nnz}, f32);
    sw_ptr = sw.data_ptr<float>();
  }
  launch_gather_sorted(sorted_pos.data_ptr<int32_t>(),
                       row_ids.data_ptr<int32_t>(), w_ptr, nnz,
                       srow.data_ptr<int64_t>(), sw_ptr, stream);
  auto head = torch::empty({nnz}, i32);
  auto pos = torch::empty({nnz}, i32);
  launch_mark_heads(sorted_ids.data_ptr<int64_t>(), nnz, vocab,
                    head.data_ptr<int32_t>(), stream);
  auto err = run_inclusive_scan_i32(temp.data_ptr(), temp_bytes,
                               head.data_ptr<int32_t>(),
                               pos.data_ptr<int32_t>(), nnz, stream);
  TORCH_CHECK(err == hipSuccess, "inclusive_scan failed");
  auto unique_tmp = torch::empty({nnz}, i64);
  auto seg_tmp = torch::empty({nnz + 1}, i64);
  auto nu_dev = torch::zeros({1}, i32);
  launch_scatter_unique(sorted_ids.data_ptr<int64_t>(),
                        head.data_ptr<int32_t>(), pos.data_ptr<int32_t>(), nnz,
                        vocab, unique_tmp.data_ptr<int64_t>(),
                        seg_tmp.data_ptr<int64_t>(), nu_dev.data_ptr<int32_t>(),
                        stream);
  auto bounds = torch::empty({2}, i64);
  launch_find_valid_bounds(sorted_ids.data_ptr<int64_t>(), nnz, vocab,
                           bounds.data_ptr<int64_t>(), stream);
  launch_pad_seg_offsets(seg_tmp.data_ptr<int64_t>(), nnz,
                         nu_dev.data_ptr<int32_t>(), bounds.data_ptr<int64_t>(),
                         stream);
  auto long_rows = torch::empty({nnz}, i64);
  auto long_count = torch::empty({1}, i32);
  auto work_items = torch::empty({nnz / 64 + 64}, i64);
  auto n_work = torch::empty({1}, i32);
  float* state_ptr = nullptr;
  torch::Tensor scratch;
  float* scratch_ptr = nullptr;
  int64_t scratch_rows = 0;
  if (adagrad) {
    CHECK_CUDA(state); CHECK_CONTIG(state);
    TORCH_CHECK(state.sizes() == weight.sizes(), "adagrad state shape mismatch");
    TORCH_CHECK(state.dtype() == torch::kFloat32, "adagrad state must be fp32");
    state_ptr = state.data_ptr<float>();
  }
  if (adagrad || wbf16) {
    scratch_rows = nnz / (128 + 1) + 1;  // max possible long segments
    scratch = torch::empty({scratch_rows, (int64_t)width}, f32);
    scratch_ptr = scratch.data_ptr<float>();
  }
  launch_sorted_optimizer_update(weight.data_ptr(), wbf16, state_ptr,
                                 (float)eps, sorted_ids.data_ptr<int64_t>(),
                                 seg_tmp.data_ptr<int64_t>(),
                                 srow.data_ptr<int64_t>(), sw_ptr,
                                 grad_out.data_ptr(),
                                 grad_out.dtype() == torch::kBFloat16,
                                 lr.data_ptr<float>(),
                                 nu_dev.data_ptr<int32_t>(), nnz, width,
                                 long_rows.data_ptr<int64_t>(),
                                 long_count.data_ptr<int32_t>(),
                                 work_items.data_ptr<int64_t>(),
                                 n_work.data_ptr<int32_t>(), scratch_ptr,
                                 scratch_rows, adagrad, stream);
}

torch::Tensor dot_interact_fwd(torch::Tensor feats, int64_t out_w) {
  CHECK_CUDA(feats); CHECK_CONTIG(feats);
  TORCH_CHECK(feats.dtype() == torch::kBFloat16, "feats must be bf16");
  TORCH_CHECK(feats.dim() == 3, "feats must be [B, F, D]");
  const int64_t B = feats.size(0);
  const int F = (int)feats.size(1), D = (int)feats.size(2);
  TORCH_CHECK(F <= 32 && D % 32 == 0, "F<=32 and D%32==0 required");
  const int tri_n = F * (F - 1) / 2;
  TORCH_CHECK(out_w >= tri_n + D, "out width too small");
  auto out = torch::empty({B, out_w}, feats.options());
  launch_dot_interact_fwd(feats.data_ptr(), out.data_ptr(), B, F, D,
                          (int)out_w, tri_n, current_stream());
  return out;
}

// sample_major=false: packed is [P, B, D] (a2a recv layout, world>1);
// sample_major=true: packed is [B, P, D] (world==1 zero-copy lookup output —
// each sample's feature rows are adjacent, so reads/writes stay local).
torch::Tensor dot_interact_fwd_packed(torch::Tensor bottom,
                                      torch::Tensor packed, torch::Tensor perm,
                                      int64_t out_w, bool sample_major) {
  CHECK_CUDA(bottom); CHECK_CUDA(packed); CHECK_CUDA(perm);
  CHECK_CONTIG(bottom); CHECK_CONTIG(packed); CHECK_CONTIG(perm);
  TORCH_CHECK(bottom.dtype() == torch::kBFloat16 &&
              packed.dtype() == torch::kBFloat16, "bf16 required");
  TORCH_CHECK(perm.dtype() == torch::kInt32, "perm must be int32");
  TORCH_CHECK(bottom.dim() == 2 && packed.dim() == 3, "bottom [B,D], packed 3-D");
  const int64_t B = bottom.size(0);
  const int D = (int)bottom.size(1);
  const int P = (int)(sample_major ? packed.size(1) : packed.size(0));
  TORCH_CHECK(packed.size(sample_major ? 0 : 1) == B && packed.size(2) == D,
              "shape mismatch");
  TORCH_CHECK(perm.numel() == P, "perm must have P entries");
  const int F = P + 1;
  TORCH_CHECK(F <= 32 && D % 32 == 0, "F<=32 and D%32==0 required");
  const int tri_n = F * (F - 1) / 2;
  TORCH_CHECK(out_w >= tri_n + D, "out width too small");
  const int64_t sb = sample_major ? P : 1;
  const int64_t sp = sample_major ? 1 : B;
  auto out = torch::empty({B, out_w}, bottom.options());
  launch_dot_interact_fwd_packed(bottom.data_ptr(), packed.data_ptr(),
                                 perm.data_ptr<int>(), out.data_ptr(), B, F, D,
                                 (int)out_w, tri_n, sb, sp, current_stream());
  return out;
}

std::vector<torch::Tensor> dot_interact_bwd_packed(torch::Tensor gout,
                                                   torch::Tensor bottom,
                                                   torch::Tensor packed,
                                                   torch::Tensor perm,
                                                   bool sample_major) {
  CHECK_CUDA(gout); CHECK_CUDA(bottom); CHECK_CUDA(packed); CHECK_CUDA(perm);
  CHECK_CONTIG(gout); CHECK_CONTIG(bottom); CHECK_CONTIG(packed);
  CHECK_CONTIG(perm);
  TORCH_CHECK(gout.dtype() == torch::kBFloat16 &&
              bottom.dtype() == torch::kBFloat16 &&
              packed.dtype() == torch::kBFloat16, "bf16 required");
  TORCH_CHECK(perm.dtype() == torch::kInt32, "perm must be int32");
  const int64_t B = bottom.size(0);
  const int D = (int)bottom.size(1);
  const int P = (int)(sample_major ? packed.size(1) : packed.size(0));
  TORCH_CHECK(perm.numel() == P, "perm must have P entries");
  TORCH_CHECK(packed.size(sample_major ? 0 : 1) == B && packed.size(2) == D,
              "shape mismatch");
  const int F = P + 1;
  const int tri_n = F * (F - 1) / 2;
  const int64_t sb = sample_major ? P : 1;
  const int64_t sp = sample_major ? 1 : B;
  auto gbottom = torch::empty_like(bottom);
  auto gpacked = torch::empty_like(packed);
  launch_dot_interact_bwd_packed(gout.data_ptr(), bottom.data_ptr(),
                                 packed.data_ptr(), perm.data_ptr<int>(),
                                 gbottom.data_ptr(), gpacked.data_ptr(), B, F,
                                 D, (int)gout.size(1), tri_n, sb, sp,
                                 current_stream());
  return {gbottom, gpacked};
}

torch::Tensor dot_interact_bwd(torch::Tensor gout, torch::Tensor feats) {
  CHECK_CUDA(gout); CHECK_CUDA(feats);
  CHECK_CONTIG(gout); CHECK_CONTIG(feats);
  TORCH_CHECK(gout.dtype() == torch::kBFloat16 &&
              feats.dtype() == torch::kBFloat16);
  const int64_t B = feats.size(0);
  const int F = (int)feats.size(1), D = (int)feats.size(2);
  const int tri_n = F * (F - 1) / 2;
  auto gfeats = torch::empty_like(feats);
  launch_dot_interact_bwd(gout.data_ptr(), feats.data_ptr(),
                          gfeats.data_ptr(), B, F, D, (int)gout.size(1), tri_n,
                          current_stream());
  return gfeats;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("csr_lookup_forward", &csr_lookup_forward,
        "CSR segmented gather-reduce forward (gfx950)",
        pybind11::arg("params"), pybind11::arg("values"),
        pybind11::arg("row_splits"), pybind11::arg("mean"),
        pybind11::arg("out_bf16") = false);
  m.def("csr_lookup_backward", &csr_lookup_backward,
        "sparse backward: sort + unique + segmented sum (gfx950)");
  m.def("row_to_split", &row_to_split, "COO rows -> CSR splits (gfx950)");
  m.def("hash_rehash", &hash_rehash,
        "re-insert occupied (key,value) pairs into a larger hash (gfx950)");
  m.def("integer_lookup", &integer_lookup,
        "open-addressing hash vocab build + lookup (gfx950)");
  m.def("sparse_row_update", &sparse_row_update,
        "fused sparse SGD/Adagrad row update (gfx950)");
  m.def("csr_fused_optimizer_apply", &csr_fused_optimizer_apply,
        "in-backward fused SGD/Adagrad update (gfx950)");
  m.def("dot_interact_fwd_packed", &dot_interact_fwd_packed,
        "pairwise-dot interaction on (bottom, packed, perm) (MFMA, gfx950)");
  m.def("dot_interact_bwd_packed", &dot_interact_bwd_packed,
        "packed interaction backward -> (gbottom, gpacked) (MFMA, gfx950)");
  m.def("dot_interact_fwd", &dot_interact_fwd,
        "fused DLRM pairwise-dot interaction forward (MFMA bf16, gfx950)");
  m.def("dot_interact_bwd", &dot_interact_bwd,
        "fused DLRM pairwise-dot interaction backward (MFMA bf16, gfx950)");
}
