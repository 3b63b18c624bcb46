// Torch-extension bindings for the CDNA4 relational kernels
// (fugue_amd/hip/csrc/relational.hip).
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <cstdio>
#include <cstring>
#include <limits>
#include <vector>

extern "C" {
void launch_hash_col_i64(const int64_t*, const bool*, uint64_t*, int64_t, int,
                         hipStream_t);
void launch_hash_col_i32(const int32_t*, const bool*, uint64_t*, int64_t, int,
                         hipStream_t);
void launch_hash_col_f64(const double*, const bool*, uint64_t*, int64_t, int,
                         hipStream_t);
void launch_hash_col_f32(const float*, const bool*, uint64_t*, int64_t, int,
                         hipStream_t);
void launch_hash_col_i8(const int8_t*, const bool*, uint64_t*, int64_t, int,
                        hipStream_t);
void launch_bucket_of(const uint64_t*, int32_t*, int64_t, int32_t,
                      hipStream_t);
void launch_histogram(const int32_t*, int64_t*, int64_t, int32_t, hipStream_t);
void launch_scatter(const int32_t*, int64_t*, int64_t*, int64_t, hipStream_t);
void launch_gb_aggregate(const int64_t*, const double*, const bool*,
                         const int32_t*, int, int64_t, int64_t*, double*,
                         int64_t*, int64_t, int, hipStream_t);
void launch_join_emit_unique(const int64_t*, int64_t, const int64_t*,
                             const int64_t*, const int64_t*,
                             const int32_t*, const int32_t*, int64_t, int,
                             int64_t*, int64_t*, int64_t*, bool*, int,
                             hipStream_t);
void launch_join_build(const int64_t*, int64_t, int32_t*, int32_t*, int64_t,
                       int32_t*,
                       hipStream_t);
void launch_gb_part_hist(const int64_t*, int64_t, int, int64_t*, int,
                         int64_t*, hipStream_t);
void launch_reduce_cols(const double*, const bool*, const int32_t*, int,
                        int64_t, double*, int64_t*, hipStream_t);
void launch_gb_part_scatter(const int64_t*, const double*, int, int64_t, int,
                            int64_t*, int64_t*, double*, int, int32_t*,
                            hipStream_t);
void launch_scatter_by_pos(const double*, const int32_t*, int64_t, double*,
                           hipStream_t);
void launch_gb_aggregate_part(const int64_t*, const double*, const int32_t*,
                              int, int64_t, const int64_t*, int64_t, int64_t*,
                              double*, int64_t*, int64_t, hipStream_t);
void launch_gb_part_scatter_staged(const int64_t*, const double*, int64_t,
                                   int, int64_t*, void*, double*, int64_t,
                                   int, int, int*, int, hipStream_t);
void launch_gb_aggregate_part_big(const void*, const double*,
                                  const int32_t*, int64_t, int64_t*,
                                  double*, int64_t*, int64_t, int64_t,
                                  int, int, int, hipStream_t);
void launch_gather_cols(const uint64_t*, const uint64_t*, int, int,
                        const int64_t*, int64_t, hipStream_t);
void launch_expr_filter(const void*, int64_t, bool*, hipStream_t);
void launch_expr_value(const void*, int64_t, int, void*, bool*,
                       hipStream_t);
void launch_join_count(const int64_t*, int64_t, const int64_t*,
                       const int64_t*, const int64_t*, const int32_t*,
                       const int32_t*, int64_t, int32_t*, hipStream_t);
void launch_join_total(const int64_t*, int64_t, const int64_t*,
                       const int64_t*, const int64_t*, const int32_t*,
                       const int32_t*, int64_t, int, int64_t*, hipStream_t);
void launch_join_emit_chunked(const int64_t*, int64_t, const int64_t*,
                              const int64_t*, const int64_t*, const int32_t*,
                              const int32_t*, int64_t, int64_t*, int64_t*,
                              int64_t*, int, hipStream_t);
void launch_join_emit(const int64_t*, int64_t, const int64_t*, const int64_t*,
                      const int64_t*, const int32_t*, const int32_t*, int64_t,
                      const int64_t*, int64_t*, int64_t*, int, hipStream_t);
void launch_join_mark_build(const int64_t*, int64_t, const int64_t*,
                            const int64_t*, const int64_t*, const int32_t*,
                            const int32_t*, int64_t, bool*, hipStream_t);
void launch_hash_string_col(const int64_t*, const uint8_t*, const bool*,
                            uint64_t*, int64_t, int, hipStream_t);
void launch_hash_seed(uint64_t*, int64_t, uint64_t, hipStream_t);
void launch_gb_mark_reps(const int64_t*, const int64_t*, const int64_t*,
                         int64_t, int64_t*, int64_t*, int64_t*, int64_t,
                         hipStream_t);
void launch_compact8(const bool*, int64_t, int64_t*, const uint64_t**,
                     uint64_t**, int, hipStream_t);
void launch_gb_key_stats(const void**, const bool**, const int*, int,
                         int64_t, int64_t, int64_t*, int32_t*, int, int,
                         uint64_t*, hipStream_t);
void launch_pack_cols(const void**, const bool**, const int*, const int64_t*,
                      const int*, int, int64_t, int64_t*, hipStream_t);
void launch_unpack_col(const int64_t*, int64_t, int, int64_t, int64_t, int,
                       int, void*, bool*, hipStream_t);
void launch_joinoa_build(const int64_t*, int64_t, int64_t*, int64_t,
                         int64_t*, hipStream_t);
void launch_joinoa_probe(const int64_t*, int64_t, const int64_t*, int64_t,
                         int64_t*, int64_t*, bool*, int, hipStream_t);
void launch_topk(const void*, int, int, int64_t, int64_t, void*, int64_t*,
                 void*, int64_t*, hipStream_t);
void launch_eq2_mask(const int64_t*, const int64_t*, const int64_t*,
                     const int64_t*, const bool*, int, int64_t, bool*,
                     hipStream_t);
void launch_cmp_imm(const void*, const bool*, int, int64_t, double, int,
                    int, int64_t, bool*, hipStream_t);
void launch_cmp_col(const void*, const bool*, const void*, const bool*, int,
                    int, int64_t, bool*, hipStream_t);
void launch_gb_compact(const int64_t*, const int64_t*, const double**,
                       double**, int, int64_t, int64_t*, int64_t*, int64_t*,
                       int64_t*, const int64_t*, int64_t*, hipStream_t);
}

namespace {

hipStream_t current_stream() {
  return (hipStream_t)c10::hip::getCurrentHIPStream().stream();
}

void check_gpu(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on the GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

const bool* opt_valid_ptr(const c10::optional<at::Tensor>& v) {
  if (!v.has_value()) return nullptr;
  return v->data_ptr<bool>();
}

}  // namespace


// Combine one column into the running row hash (out int64 viewed as u64).
void hash_column(at::Tensor data, c10::optional<at::Tensor> valid,
                 at::Tensor out, bool is_first) {
  check_gpu(data, "data");
  check_gpu(out, "out");
  int64_t n = data.numel();
  TORCH_CHECK(out.numel() == n, "out size mismatch");
  auto stream = current_stream();
  uint64_t* optr = reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>());
  const bool* vptr = opt_valid_ptr(valid);
  switch (data.scalar_type()) {
    case at::kLong:
      launch_hash_col_i64(data.data_ptr<int64_t>(), vptr, optr, n, is_first,
                          stream);
      break;
    case at::kInt:
      launch_hash_col_i32(data.data_ptr<int32_t>(), vptr, optr, n, is_first,
                          stream);
      break;
    case at::kDouble:
      launch_hash_col_f64(data.data_ptr<double>(), vptr, optr, n, is_first,
                          stream);
      break;
    case at::kFloat:
      launch_hash_col_f32(data.data_ptr<float>(), vptr, optr, n, is_first,
                          stream);
      break;
    case at::kChar:
    case at::kBool:
      launch_hash_col_i8((const int8_t*)data.data_ptr(), vptr, optr, n,
                         is_first, stream);
      break;
    default:
      TORCH_CHECK(false, "unsupported dtype for hash_column");
  }
}

at::Tensor bucket_of(at::Tensor hashes, int64_t num_buckets) {
  check_gpu(hashes, "hashes");
  int64_t n = hashes.numel();
  auto buckets = at::empty({n}, hashes.options().dtype(at::kInt));
  launch_bucket_of(
      reinterpret_cast<const uint64_t*>(hashes.data_ptr<int64_t>()),
      buckets.data_ptr<int32_t>(), n, (int32_t)num_buckets, current_stream());
  return buckets;
}

at::Tensor bucket_histogram(at::Tensor buckets, int64_t num_buckets) {
  check_gpu(buckets, "buckets");
  auto hist = at::zeros({num_buckets}, buckets.options().dtype(at::kLong));
  launch_histogram(buckets.data_ptr<int32_t>(), hist.data_ptr<int64_t>(),
                   buckets.numel(), (int32_t)num_buckets, current_stream());
  return hist;
}

at::Tensor bucket_scatter(at::Tensor buckets, at::Tensor offsets) {
  // offsets: exclusive prefix sum of the histogram (int64, on device);
  // mutated in-place as the scatter cursor.
  check_gpu(buckets, "buckets");
  check_gpu(offsets, "offsets");
  int64_t n = buckets.numel();
  auto perm = at::empty({n}, buckets.options().dtype(at::kLong));
  launch_scatter(buckets.data_ptr<int32_t>(), offsets.data_ptr<int64_t>(),
                 perm.data_ptr<int64_t>(), n, current_stream());
  return perm;
}

std::vector<at::Tensor> gb_aggregate(at::Tensor keys, at::Tensor vals,
                                     c10::optional<at::Tensor> valids,
                                     at::Tensor ops, int64_t tsize,
                                     bool use_lds) {
  check_gpu(keys, "keys");
  check_gpu(vals, "vals");
  check_gpu(ops, "ops");
  TORCH_CHECK((tsize & (tsize - 1)) == 0, "tsize must be a power of 2");
  int64_t n = keys.numel();
  int n_aggs = (int)vals.size(0);
  TORCH_CHECK(vals.dim() == 2 && vals.size(1) == n, "vals must be [A, n]");
  auto tkeys = at::full({tsize}, (int64_t)0x8000000000000000LL,
                        keys.options());
  // init aggs to op-appropriate identity: sum/count -> 0; min -> +inf; max -> -inf
  auto gaggs = at::zeros({n_aggs, tsize}, vals.options());
  auto ops_cpu = ops.to(at::kCPU);
  for (int a = 0; a < n_aggs; ++a) {
    int op = ops_cpu[a].item<int32_t>();
    if (op == 1) gaggs[a].fill_(std::numeric_limits<double>::infinity());
    if (op == 2) gaggs[a].fill_(-std::numeric_limits<double>::infinity());
  }
  auto gcount = at::zeros({tsize}, keys.options());
  launch_gb_aggregate(keys.data_ptr<int64_t>(), vals.data_ptr<double>(),
                      opt_valid_ptr(valids), ops.data_ptr<int32_t>(), n_aggs,
                      n, tkeys.data_ptr<int64_t>(), gaggs.data_ptr<double>(),
                      gcount.data_ptr<int64_t>(), tsize, use_lds ? 1 : 0,
                      current_stream());
  return {tkeys, gaggs, gcount};
}

std::vector<at::Tensor> gb_aggregate_partitioned(
    at::Tensor keys, at::Tensor vals, at::Tensor ops, int64_t num_parts,
    int64_t tsize, int64_t scatter_chunk, int64_t agg_chunk, int64_t nt,
    int64_t narrow, bool record_layout, bool force_simple) {
  check_gpu(keys, "keys");
  check_gpu(vals, "vals");
  TORCH_CHECK((tsize & (tsize - 1)) == 0, "tsize must be a power of 2");
  TORCH_CHECK((num_parts & (num_parts - 1)) == 0,
              "num_parts must be a power of 2");
  int64_t n = keys.numel();
  int n_aggs = (int)vals.size(0);
  auto stream = current_stream();
  int shift = 64;
  {
    int64_t p = num_parts;
    while (p > 1) { --shift; p >>= 1; }
  }
  // phase 1: histogram + scan (also reduces key min/max when the
  // caller asks for narrow auto-detection: narrow < 0)
  auto hist = at::zeros({num_parts}, keys.options());
  TORCH_CHECK(num_parts <= 4096, "num_parts must be <= 4096");
  bool staged =
      (n_aggs == 1 &&
       (num_parts == 512 || num_parts == 1024 || num_parts == 2048));
  // layout recording goes through the simple scatter (it has each
  // row's spill position in hand); replay restores the staged-path
  // spill format (narrow keys done by the caller)
  record_layout = record_layout && n_aggs == 1 && n < (int64_t(1) << 31);
  if (record_layout || force_simple) staged = false;
  // phase-3 LDS table sized for ~0.5 load at the partition granularity:
  // 512 parts -> 4096 slots, 1024 -> 2048, 2048 -> 1024
  int slots = num_parts == 1024 ? 2048 : (num_parts == 2048 ? 1024 : 4096);
  // narrow < 0 = speculative: run the int32 path with an in-kernel
  // overflow flag; the CALLER checks the returned flag at its existing
  // sync point and re-runs wide if set (no mid-pipeline sync here)
  bool speculative = staged && narrow < 0;
  if (speculative) narrow = 1;
  launch_gb_part_hist(keys.data_ptr<int64_t>(), n, shift,
                      hist.data_ptr<int64_t>(), (int)num_parts, nullptr,
                      stream);
  auto offsets = at::zeros({num_parts + 1}, keys.options());
  offsets.narrow(0, 1, num_parts).copy_(at::cumsum(hist, 0));
  auto cursor = offsets.narrow(0, 0, num_parts).clone();
  // phase 2: scatter into partitioned order
  bool use_narrow = staged && narrow > 0;
  auto pkeys = use_narrow
                   ? at::empty({n}, keys.options().dtype(at::kInt))
                   : at::empty({n}, keys.options());
  auto pvals = at::empty({n_aggs, n}, vals.options());
  auto ovf = at::zeros({1}, keys.options().dtype(at::kInt));
  auto pos = at::empty({record_layout ? n : 0},
                       keys.options().dtype(at::kInt));
  if (staged) {
    launch_gb_part_scatter_staged(
        keys.data_ptr<int64_t>(), vals.data_ptr<double>(), n, shift,
        cursor.data_ptr<int64_t>(), pkeys.data_ptr(),
        pvals.data_ptr<double>(), scatter_chunk, (int)nt,
        use_narrow ? 1 : 0,
        (use_narrow && speculative) ? ovf.data_ptr<int32_t>() : nullptr,
        (int)num_parts, stream);
  } else {
    launch_gb_part_scatter(keys.data_ptr<int64_t>(), vals.data_ptr<double>(),
                           n_aggs, n, shift, cursor.data_ptr<int64_t>(),
                           pkeys.data_ptr<int64_t>(), pvals.data_ptr<double>(),
                           (int)num_parts,
                           record_layout ? pos.data_ptr<int32_t>() : nullptr,
                           stream);
  }
  // phase 3: per-partition LDS aggregation into the global table
  auto tkeys = at::full({tsize}, (int64_t)0x8000000000000000LL,
                        keys.options());
  auto gaggs = at::zeros({n_aggs, tsize}, vals.options());
  auto gcount = at::zeros({tsize}, keys.options());
  if (staged) {
    launch_gb_aggregate_part_big(
        pkeys.data_ptr(), pvals.data_ptr<double>(),
        ops.data_ptr<int32_t>(), n, tkeys.data_ptr<int64_t>(),
        gaggs.data_ptr<double>(), gcount.data_ptr<int64_t>(), tsize,
        agg_chunk, (int)nt, use_narrow ? 1 : 0, slots, stream);
  } else {
    launch_gb_aggregate_part(
        pkeys.data_ptr<int64_t>(), pvals.data_ptr<double>(),
        ops.data_ptr<int32_t>(), n_aggs, n, offsets.data_ptr<int64_t>(),
        num_parts, tkeys.data_ptr<int64_t>(), gaggs.data_ptr<double>(),
        gcount.data_ptr<int64_t>(), tsize, stream);
  }
  return {tkeys, gaggs, gcount, ovf, pkeys, pos};
}

// Phase 2+3 only, with a recorded layout: place the fresh values at
// the recorded spill positions and aggregate against the cached
// partitioned keys (phase 1 + key scatter skipped entirely).
std::vector<at::Tensor> gb_aggregate_replay(at::Tensor pkeys_cached,
                                            at::Tensor pos, at::Tensor vals,
                                            at::Tensor ops, int64_t tsize,
                                            int64_t agg_chunk, int64_t nt,
                                            int64_t slots) {
  check_gpu(pkeys_cached, "pkeys");
  check_gpu(pos, "pos");
  check_gpu(vals, "vals");
  TORCH_CHECK(vals.size(0) == 1, "replay supports single-agg spills");
  int64_t n = pos.numel();
  auto stream = current_stream();
  auto pvals = at::empty({1, n}, vals.options());
  launch_scatter_by_pos(vals.data_ptr<double>(), pos.data_ptr<int32_t>(), n,
                        pvals.data_ptr<double>(), stream);
  auto tkeys = at::full({tsize}, (int64_t)0x8000000000000000LL,
                        pos.options().dtype(at::kLong));
  auto gaggs = at::zeros({1, tsize}, vals.options());
  auto gcount = at::zeros({tsize}, pos.options().dtype(at::kLong));
  launch_gb_aggregate_part_big(
      pkeys_cached.data_ptr(), pvals.data_ptr<double>(),
      ops.data_ptr<int32_t>(), n, tkeys.data_ptr<int64_t>(),
      gaggs.data_ptr<double>(), gcount.data_ptr<int64_t>(), tsize,
      agg_chunk, (int)nt, pkeys_cached.element_size() == 4 ? 1 : 0,
      (int)slots, stream);
  return {tkeys, gaggs, gcount};
}

std::vector<at::Tensor> join_build(at::Tensor keys, int64_t tsize) {
  check_gpu(keys, "keys");
  TORCH_CHECK((tsize & (tsize - 1)) == 0, "tsize must be a power of 2");
  int64_t n = keys.numel();
  auto heads = at::full({tsize}, -1, keys.options().dtype(at::kInt));
  auto next = at::empty({std::max<int64_t>(n, 1)},
                        keys.options().dtype(at::kInt));
  auto dup = at::zeros({1}, keys.options().dtype(at::kInt));
  if (n > 0) {
    launch_join_build(keys.data_ptr<int64_t>(), n, heads.data_ptr<int32_t>(),
                      next.data_ptr<int32_t>(), tsize,
                      dup.data_ptr<int32_t>(), current_stream());
  }
  return {heads, next, dup};
}


namespace {
const int64_t* opt_i64_ptr(const c10::optional<at::Tensor>& v) {
  if (!v.has_value()) return nullptr;
  return v->data_ptr<int64_t>();
}
}  // namespace

std::vector<at::Tensor> join_emit_unique(
    at::Tensor pkeys, at::Tensor bkeys, c10::optional<at::Tensor> ph2,
    c10::optional<at::Tensor> bh2, at::Tensor heads, at::Tensor next,
    int64_t mode, bool want_pi, bool want_mask, bool mask_neg) {
  check_gpu(pkeys, "pkeys");
  int64_t np = pkeys.numel();
  int64_t tsize = heads.numel();
  // positional mode (1) with want_pi=false skips the pi write — the
  // compaction kernel emits row indices itself; want_mask emits the
  // matched mask in the same pass (no separate compare kernel)
  auto out_pi = at::empty({(mode == 1 && !want_pi) ? 0 : np},
                          pkeys.options());
  auto out_bi = at::empty({np}, pkeys.options());
  auto cursor = at::zeros({1}, pkeys.options());
  auto out_mask = at::empty({want_mask && mode == 1 ? np : 0},
                            pkeys.options().dtype(at::kBool));
  if (np > 0) {
    launch_join_emit_unique(
        pkeys.data_ptr<int64_t>(), np, bkeys.data_ptr<int64_t>(),
        opt_i64_ptr(ph2), opt_i64_ptr(bh2), heads.data_ptr<int32_t>(),
        next.data_ptr<int32_t>(), tsize, (int)mode,
        out_pi.numel() > 0 ? out_pi.data_ptr<int64_t>() : nullptr,
        out_bi.data_ptr<int64_t>(), cursor.data_ptr<int64_t>(),
        out_mask.numel() > 0 ? out_mask.data_ptr<bool>() : nullptr,
        mask_neg ? 1 : 0, current_stream());
  }
  return {out_pi, out_bi, cursor, out_mask};
}

at::Tensor join_count(at::Tensor pkeys, at::Tensor bkeys,
                      c10::optional<at::Tensor> ph2,
                      c10::optional<at::Tensor> bh2, at::Tensor heads,
                      at::Tensor next, int64_t tsize) {
  check_gpu(pkeys, "pkeys");
  int64_t np = pkeys.numel();
  auto counts = at::zeros({np}, pkeys.options().dtype(at::kInt));
  if (np > 0) {
    launch_join_count(pkeys.data_ptr<int64_t>(), np,
                      bkeys.data_ptr<int64_t>(), opt_i64_ptr(ph2),
                      opt_i64_ptr(bh2), heads.data_ptr<int32_t>(),
                      next.data_ptr<int32_t>(), tsize,
                      counts.data_ptr<int32_t>(), current_stream());
  }
  return counts;
}

std::vector<at::Tensor> join_pairs(at::Tensor pkeys, at::Tensor bkeys,
                                   c10::optional<at::Tensor> ph2,
                                   c10::optional<at::Tensor> bh2,
                                   at::Tensor heads, at::Tensor next,
                                   int64_t tsize, int64_t mode) {
  // one-scalar total + chunked single-reservation emit (no per-row
  // counts array / prefix sum)
  check_gpu(pkeys, "pkeys");
  int64_t np = pkeys.numel();
  auto stream = current_stream();
  auto total_t = at::zeros({1}, pkeys.options());
  if (np > 0) {
    launch_join_total(pkeys.data_ptr<int64_t>(), np,
                      bkeys.data_ptr<int64_t>(), opt_i64_ptr(ph2),
                      opt_i64_ptr(bh2), heads.data_ptr<int32_t>(),
                      next.data_ptr<int32_t>(), tsize, (int)mode,
                      total_t.data_ptr<int64_t>(), stream);
  }
  int64_t out_n = total_t.cpu().item<int64_t>();
  auto out_p = at::empty({out_n}, pkeys.options());
  auto out_b = at::empty({out_n}, pkeys.options());
  if (np > 0 && out_n > 0) {
    auto cursor = at::zeros({1}, pkeys.options());
    launch_join_emit_chunked(
        pkeys.data_ptr<int64_t>(), np, bkeys.data_ptr<int64_t>(),
        opt_i64_ptr(ph2), opt_i64_ptr(bh2), heads.data_ptr<int32_t>(),
        next.data_ptr<int32_t>(), tsize, cursor.data_ptr<int64_t>(),
        out_p.data_ptr<int64_t>(), out_b.data_ptr<int64_t>(), (int)mode,
        stream);
  }
  return {out_p, out_b};
}

std::vector<at::Tensor> join_emit(at::Tensor pkeys, at::Tensor bkeys,
                                  c10::optional<at::Tensor> ph2,
                                  c10::optional<at::Tensor> bh2,
                                  at::Tensor heads, at::Tensor next,
                                  int64_t tsize, at::Tensor offsets,
                                  int64_t out_n, int64_t mode) {
  check_gpu(pkeys, "pkeys");
  int64_t np = pkeys.numel();
  auto out_p = at::empty({out_n}, pkeys.options());
  auto out_b = at::empty({out_n}, pkeys.options());
  if (np > 0 && out_n > 0) {
    launch_join_emit(pkeys.data_ptr<int64_t>(), np, bkeys.data_ptr<int64_t>(),
                     opt_i64_ptr(ph2), opt_i64_ptr(bh2),
                     heads.data_ptr<int32_t>(), next.data_ptr<int32_t>(),
                     tsize, offsets.data_ptr<int64_t>(),
                     out_p.data_ptr<int64_t>(), out_b.data_ptr<int64_t>(),
                     (int)mode, current_stream());
  }
  return {out_p, out_b};
}

at::Tensor join_mark_build(at::Tensor pkeys, at::Tensor bkeys,
                           c10::optional<at::Tensor> ph2,
                           c10::optional<at::Tensor> bh2, at::Tensor heads,
                           at::Tensor next, int64_t tsize, int64_t n_build) {
  check_gpu(pkeys, "pkeys");
  auto matched = at::zeros({n_build}, pkeys.options().dtype(at::kBool));
  if (pkeys.numel() > 0 && n_build > 0) {
    launch_join_mark_build(pkeys.data_ptr<int64_t>(), pkeys.numel(),
                           bkeys.data_ptr<int64_t>(), opt_i64_ptr(ph2),
                           opt_i64_ptr(bh2), heads.data_ptr<int32_t>(),
                           next.data_ptr<int32_t>(), tsize,
                           matched.data_ptr<bool>(), current_stream());
  }
  return matched;
}

void hash_string_column(at::Tensor offsets, at::Tensor bytes,
                        c10::optional<at::Tensor> valid, at::Tensor out,
                        bool is_first) {
  check_gpu(offsets, "offsets");
  check_gpu(out, "out");
  int64_t n = out.numel();
  launch_hash_string_col(
      offsets.data_ptr<int64_t>(), bytes.data_ptr<uint8_t>(),
      opt_valid_ptr(valid),
      reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()), n, is_first,
      current_stream());
}

void hash_seed(at::Tensor out, int64_t seed) {
  check_gpu(out, "out");
  launch_hash_seed(reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()),
                   out.numel(), (uint64_t)seed, current_stream());
}

std::vector<at::Tensor> gb_mark_reps(at::Tensor h1, at::Tensor h2,
                                     at::Tensor tkeys, int64_t tsize) {
  check_gpu(h1, "h1");
  auto rep = at::full({tsize}, -1, h1.options());
  auto th2 = at::full({tsize}, (int64_t)0x8000000000000000LL, h1.options());
  auto conflict = at::zeros({1}, h1.options());
  if (h1.numel() > 0) {
    launch_gb_mark_reps(h1.data_ptr<int64_t>(), h2.data_ptr<int64_t>(),
                        tkeys.data_ptr<int64_t>(), tsize,
                        rep.data_ptr<int64_t>(), th2.data_ptr<int64_t>(),
                        conflict.data_ptr<int64_t>(), h1.numel(),
                        current_stream());
  }
  return {rep, th2, conflict};
}

// Compact up to 8 eight-byte columns by mask in one pass.
// Returns the output tensors (sized by mask.sum(), computed on device
// and read back once).
std::vector<at::Tensor> compact_columns(at::Tensor mask,
                                        std::vector<at::Tensor> cols,
                                        int64_t out_n) {
  check_gpu(mask, "mask");
  TORCH_CHECK(cols.size() >= 1 && cols.size() <= 8, "1..8 columns");
  int64_t n = mask.numel();
  auto cursor = at::zeros({1}, mask.options().dtype(at::kLong));
  const uint64_t* srcs[8];
  uint64_t* dsts[8];
  std::vector<at::Tensor> outs;
  for (size_t c = 0; c < cols.size(); ++c) {
    check_gpu(cols[c], "col");
    TORCH_CHECK(cols[c].element_size() == 8, "8-byte columns only");
    auto out = at::empty({out_n}, cols[c].options());
    srcs[c] = reinterpret_cast<const uint64_t*>(cols[c].data_ptr());
    dsts[c] = reinterpret_cast<uint64_t*>(out.data_ptr());
    outs.push_back(out);
  }
  launch_compact8(mask.data_ptr<bool>(), n, cursor.data_ptr<int64_t>(),
                  srcs, dsts, (int)cols.size(), current_stream());
  return outs;
}

// Capacity-mode compaction: outputs are allocated at mask length and the
// caller narrows using the returned cursor (total) — skips the separate
// mask.sum() reduction + its host sync (profiles/NOTES.md r02).
std::vector<at::Tensor> compact_columns_cap(
    at::Tensor mask, std::vector<c10::optional<at::Tensor>> cols) {
  check_gpu(mask, "mask");
  TORCH_CHECK(cols.size() >= 1 && cols.size() <= 8, "1..8 columns");
  int64_t n = mask.numel();
  auto cursor = at::zeros({1}, mask.options().dtype(at::kLong));
  const uint64_t* srcs[8];
  uint64_t* dsts[8];
  std::vector<at::Tensor> outs;
  for (size_t c = 0; c < cols.size(); ++c) {
    at::Tensor out;
    if (cols[c].has_value()) {  // nullopt -> emit the row index
      check_gpu(*cols[c], "col");
      TORCH_CHECK(cols[c]->element_size() == 8, "8-byte columns only");
      srcs[c] = reinterpret_cast<const uint64_t*>(cols[c]->data_ptr());
      out = at::empty({n}, cols[c]->options());
    } else {
      srcs[c] = nullptr;
      out = at::empty({n}, mask.options().dtype(at::kLong));
    }
    dsts[c] = reinterpret_cast<uint64_t*>(out.data_ptr());
    outs.push_back(out);
  }
  launch_compact8(mask.data_ptr<bool>(), n, cursor.data_ptr<int64_t>(),
                  srcs, dsts, (int)cols.size(), current_stream());
  outs.push_back(cursor);
  return outs;
}


namespace {
int cmp_dtype_code(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kLong: return 0;
    case at::kInt: return 1;
    case at::kShort: return 2;
    case at::kDouble: return 3;
    case at::kFloat: return 4;
    default: TORCH_CHECK(false, "unsupported compare dtype"); return -1;
  }
}
}  // namespace




// Open-addressed unique join for int64 keys (no h2): one 16B (key,idx)
// entry per slot — ~1 random cache line per probe vs ~3 for the
// chained layout.  flags = [dup_build_key, sentinel_key_seen]; either
// nonzero means the caller must fall back to the chained join.
std::vector<at::Tensor> join_open_unique(at::Tensor pkeys, at::Tensor bkeys,
                                         bool want_pi, bool want_mask,
                                         bool mask_neg) {
  check_gpu(pkeys, "pkeys");
  check_gpu(bkeys, "bkeys");
  int64_t np = pkeys.numel();
  int64_t nb = bkeys.numel();
  int64_t tsize = 16;
  while (tsize < nb * 2) tsize <<= 1;
  auto table = at::empty({2 * tsize}, bkeys.options());
  auto flags = at::zeros({2}, bkeys.options());
  launch_joinoa_build(nb > 0 ? bkeys.data_ptr<int64_t>() : nullptr, nb,
                      table.data_ptr<int64_t>(), tsize,
                      flags.data_ptr<int64_t>(), current_stream());
  auto out_pi = at::empty({want_pi ? np : 0}, pkeys.options());
  auto out_bi = at::empty({np}, pkeys.options());
  auto out_mask =
      at::empty({want_mask ? np : 0}, pkeys.options().dtype(at::kBool));
  if (np > 0) {
    launch_joinoa_probe(
        pkeys.data_ptr<int64_t>(), np, table.data_ptr<int64_t>(), tsize,
        want_pi ? out_pi.data_ptr<int64_t>() : nullptr,
        out_bi.data_ptr<int64_t>(),
        want_mask ? out_mask.data_ptr<bool>() : nullptr, mask_neg ? 1 : 0,
        current_stream());
  }
  return {out_pi, out_bi, out_mask, flags};
}

// Own top-k select for k <= 16 (see relational.hip); returns
// (values[k], indices[k]) sorted, index-tiebroken (deterministic).
std::vector<at::Tensor> topk_select(at::Tensor vals, int64_t k,
                                    bool largest) {
  check_gpu(vals, "vals");
  TORCH_CHECK(k >= 1 && k <= 16, "k must be in [1, 16]");
  int dt;
  if (vals.scalar_type() == at::kLong) {
    dt = 0;
  } else if (vals.scalar_type() == at::kDouble) {
    dt = 1;
  } else {
    TORCH_CHECK(false, "topk_select supports int64/float64");
  }
  int64_t n = vals.numel();
  TORCH_CHECK(n >= 1, "empty input");
  auto cand_v = at::empty({512 * k}, vals.options());
  auto cand_i = at::empty({512 * k}, vals.options().dtype(at::kLong));
  auto out_v = at::empty({k}, vals.options());
  auto out_i = at::empty({k}, vals.options().dtype(at::kLong));
  launch_topk(vals.data_ptr(), dt, largest ? 1 : 0, n, k,
              cand_v.data_ptr(), cand_i.data_ptr<int64_t>(),
              out_v.data_ptr(), out_i.data_ptr<int64_t>(),
              current_stream());
  return {out_v, out_i};
}

at::Tensor eq2_mask(at::Tensor h1, at::Tensor h2, at::Tensor l1,
                    at::Tensor l2, c10::optional<at::Tensor> valid,
                    bool neg) {
  check_gpu(h1, "h1");
  check_gpu(h2, "h2");
  auto out = at::empty({h1.numel()}, h1.options().dtype(at::kBool));
  launch_eq2_mask(h1.data_ptr<int64_t>(), h2.data_ptr<int64_t>(),
                  l1.data_ptr<int64_t>(), l2.data_ptr<int64_t>(),
                  opt_valid_ptr(valid), neg ? 1 : 0, h1.numel(),
                  out.data_ptr<bool>(), current_stream());
  return out;
}

// Fast path for single-comparison WHERE filters (see relational.hip).
at::Tensor cmp_imm(at::Tensor a, c10::optional<at::Tensor> valid,
                   int64_t imm_i, double imm_d, bool use_int, int64_t op) {
  check_gpu(a, "a");
  int dt = cmp_dtype_code(a);
  bool ui = use_int && dt <= 2;
  auto out = at::empty({a.numel()}, a.options().dtype(at::kBool));
  launch_cmp_imm(a.data_ptr(), opt_valid_ptr(valid), dt, imm_i, imm_d,
                 ui ? 1 : 0, (int)op, a.numel(), out.data_ptr<bool>(),
                 current_stream());
  return out;
}

at::Tensor cmp_col(at::Tensor a, c10::optional<at::Tensor> va, at::Tensor b,
                   c10::optional<at::Tensor> vb, int64_t op) {
  check_gpu(a, "a");
  check_gpu(b, "b");
  TORCH_CHECK(a.scalar_type() == b.scalar_type(), "same dtype required");
  int dt = cmp_dtype_code(a);
  auto out = at::empty({a.numel()}, a.options().dtype(at::kBool));
  launch_cmp_col(a.data_ptr(), opt_valid_ptr(va), b.data_ptr(),
                 opt_valid_ptr(vb), dt, (int)op, a.numel(),
                 out.data_ptr<bool>(), current_stream());
  return out;
}

// host mirror of ExprProg in relational.hip (layout must match)
struct ExprProgHost {
  int n_ops;
  unsigned char op[48];
  signed char aux[48];
  unsigned long long imm[12];
  unsigned long long col_data[12];
  unsigned long long col_valid[12];
  unsigned char col_dt[12];
};

static ExprProgHost build_expr_prog(at::Tensor& ops, at::Tensor& aux,
                                    at::Tensor& imm,
                                    std::vector<at::Tensor>& col_data,
                                    std::vector<at::Tensor>& col_valid,
                                    at::Tensor& col_dt) {
  TORCH_CHECK(ops.numel() <= 48, "program too long");
  TORCH_CHECK(col_data.size() <= 12, "too many columns");
  TORCH_CHECK(imm.numel() <= 12, "too many immediates");
  ExprProgHost prog;
  std::memset(&prog, 0, sizeof(prog));
  prog.n_ops = (int)ops.numel();
  auto ops_c = ops.to(at::kByte).cpu();
  auto aux_c = aux.to(at::kChar).cpu();
  auto imm_c = imm.cpu();
  auto dt_c = col_dt.to(at::kByte).cpu();
  for (int i = 0; i < prog.n_ops; ++i) {
    prog.op[i] = ops_c[i].item<uint8_t>();
    prog.aux[i] = aux_c[i].item<int8_t>();
  }
  for (int i = 0; i < (int)imm_c.numel(); ++i) {
    prog.imm[i] = (unsigned long long)imm_c[i].item<int64_t>();
  }
  for (size_t c = 0; c < col_data.size(); ++c) {
    check_gpu(col_data[c], "col");
    TORCH_CHECK(col_data[c].is_contiguous(), "columns must be contiguous");
    prog.col_data[c] = reinterpret_cast<unsigned long long>(
        col_data[c].data_ptr());
    prog.col_dt[c] = dt_c[c].item<uint8_t>();
    if (col_valid[c].defined() && col_valid[c].numel() > 0) {
      check_gpu(col_valid[c], "valid");
      prog.col_valid[c] = reinterpret_cast<unsigned long long>(
          col_valid[c].data_ptr());
    }
  }
  return prog;
}

std::vector<at::Tensor> expr_value(at::Tensor ops, at::Tensor aux,
                                   at::Tensor imm,
                                   std::vector<at::Tensor> col_data,
                                   std::vector<at::Tensor> col_valid,
                                   at::Tensor col_dt, int64_t n,
                                   int64_t out_int) {
  TORCH_CHECK(col_data.size() >= 1, "needs at least one column");
  auto prog = build_expr_prog(ops, aux, imm, col_data, col_valid, col_dt);
  auto opts = col_data[0].options();
  auto out = at::empty({n}, opts.dtype(out_int ? at::kLong : at::kDouble));
  auto valid = at::empty({n}, opts.dtype(at::kBool));
  if (n > 0) {
    launch_expr_value(&prog, n, (int)out_int, out.data_ptr(),
                      valid.data_ptr<bool>(), current_stream());
  }
  return {out, valid};
}

at::Tensor expr_filter(at::Tensor ops, at::Tensor aux, at::Tensor imm,
                       std::vector<at::Tensor> col_data,
                       std::vector<at::Tensor> col_valid,
                       at::Tensor col_dt, int64_t n) {
  TORCH_CHECK(col_data.size() >= 1, "needs at least one column");
  auto prog = build_expr_prog(ops, aux, imm, col_data, col_valid, col_dt);
  auto out = at::empty({n}, col_data[0].options().dtype(at::kBool));
  if (n > 0) {
    launch_expr_filter(&prog, n, out.data_ptr<bool>(), current_stream());
  }
  return out;
}

std::vector<at::Tensor> gather_columns(at::Tensor idx,
                                       std::vector<at::Tensor> cols) {
  check_gpu(idx, "idx");
  TORCH_CHECK(cols.size() >= 1 && cols.size() <= 16, "1..16 columns");
  int64_t n = idx.numel();
  // group by element width: one fused launch per width class
  std::vector<at::Tensor> outs(cols.size());
  for (int width : {8, 4, 2, 1}) {
    uint64_t srcs[16];
    uint64_t dsts[16];
    int k = 0;
    for (size_t c = 0; c < cols.size(); ++c) {
      if (cols[c].element_size() != width) continue;
      check_gpu(cols[c], "col");
      TORCH_CHECK(cols[c].is_contiguous(), "columns must be contiguous");
      auto out = at::empty({n}, cols[c].options());
      srcs[k] = reinterpret_cast<uint64_t>(cols[c].data_ptr());
      dsts[k] = reinterpret_cast<uint64_t>(out.data_ptr());
      outs[c] = out;
      ++k;
    }
    if (k > 0 && n > 0) {
      launch_gather_cols(srcs, dsts, k, width, idx.data_ptr<int64_t>(), n,
                         current_stream());
    } else if (k > 0) {
      // n == 0: outputs already sized 0
    }
  }
  return outs;
}

std::vector<at::Tensor> reduce_columns(at::Tensor vals,
                                       c10::optional<at::Tensor> valids,
                                       at::Tensor ops) {
  check_gpu(vals, "vals");
  int n_aggs = (int)vals.size(0);
  int64_t n = vals.size(1);
  auto out = at::zeros({n_aggs}, vals.options());
  auto ops_cpu = ops.cpu();
  for (int a = 0; a < n_aggs; ++a) {
    int op = (int)ops_cpu[a].item<int32_t>();
    if (op == 1) out[a] = std::numeric_limits<double>::infinity();
    if (op == 2) out[a] = -std::numeric_limits<double>::infinity();
  }
  auto cnt = at::zeros({n_aggs}, vals.options().dtype(at::kLong));
  const bool* vptr = nullptr;
  if (valids.has_value()) {
    vptr = valids.value().data_ptr<bool>();
  }
  if (n > 0) {
    launch_reduce_cols(vals.data_ptr<double>(), vptr,
                       ops.data_ptr<int32_t>(), n_aggs, n,
                       out.data_ptr<double>(), cnt.data_ptr<int64_t>(),
                       current_stream());
  }
  return {out, cnt};
}


// ---- fused group-by key preparation --------------------------------- //

// One launch set + ONE host readback for everything the group-by sizing
// needs: per-column min/max (2*ncols leading slots, signed order
// preserved via the s2u bias) and the sampled distinct estimate
// (d, f1, f2 in the last 3 slots).  Replaces per-column
// min().item()/max().item() and the torch.unique Chao83 sampling
// (profiles/NOTES.md r02 host-sync findings).
at::Tensor gb_key_stats(std::vector<at::Tensor> cols,
                        std::vector<c10::optional<at::Tensor>> valids,
                        int64_t nsamples, int64_t tsize, bool do_minmax) {
  int k = (int)cols.size();
  TORCH_CHECK(k >= 1 && k <= 8, "1..8 key columns");
  TORCH_CHECK((tsize & (tsize - 1)) == 0, "tsize must be a power of 2");
  int64_t n = cols[0].numel();
  const void* data[8];
  const bool* valid[8];
  int dwidth[8];
  for (int c = 0; c < k; ++c) {
    check_gpu(cols[c], "key col");
    int es = (int)cols[c].element_size();
    TORCH_CHECK(es == 8 || es == 4 || es == 2, "2/4/8-byte key columns");
    data[c] = cols[c].data_ptr();
    valid[c] = opt_valid_ptr(valids[c]);
    dwidth[c] = es;
  }
  auto out = at::empty({2 * k + 3}, cols[0].options().dtype(at::kLong));
  auto slots = at::empty({tsize}, cols[0].options().dtype(at::kLong));
  auto counts = at::empty({tsize}, cols[0].options().dtype(at::kInt));
  launch_gb_key_stats(data, valid, dwidth, k, n, nsamples,
                      slots.data_ptr<int64_t>(), counts.data_ptr<int32_t>(),
                      (int)tsize, do_minmax ? 1 : 0,
                      reinterpret_cast<uint64_t*>(out.data_ptr<int64_t>()),
                      current_stream());
  return out;
}

// Pack up to 8 integer key columns into one int64 group key in a single
// pass (code 0 = NULL; valid values offset by 1 from the column min).
at::Tensor pack_columns(std::vector<at::Tensor> cols,
                        std::vector<c10::optional<at::Tensor>> valids,
                        std::vector<int64_t> mins,
                        std::vector<int64_t> shifts) {
  int k = (int)cols.size();
  TORCH_CHECK(k >= 1 && k <= 8, "1..8 key columns");
  int64_t n = cols[0].numel();
  const void* data[8];
  const bool* valid[8];
  int dwidth[8];
  int shift_i[8];
  for (int c = 0; c < k; ++c) {
    check_gpu(cols[c], "key col");
    int es = (int)cols[c].element_size();
    TORCH_CHECK(es == 8 || es == 4 || es == 2, "2/4/8-byte key columns");
    data[c] = cols[c].data_ptr();
    valid[c] = opt_valid_ptr(valids[c]);
    dwidth[c] = es;
    shift_i[c] = (int)shifts[c];
  }
  auto out = at::empty({n}, cols[0].options().dtype(at::kLong));
  launch_pack_cols(data, valid, dwidth, mins.data(), shift_i, k, n,
                   out.data_ptr<int64_t>(), current_stream());
  return out;
}

// Unpack one column from packed group keys; returns (data, valid) where
// valid is an empty tensor when the source column had no nulls.
std::vector<at::Tensor> unpack_column(at::Tensor packed, int64_t shift,
                                      int64_t width, int64_t lo,
                                      int64_t dwidth, bool has_nulls) {
  check_gpu(packed, "packed");
  int64_t n = packed.numel();
  auto dt = dwidth == 8 ? at::kLong : (dwidth == 4 ? at::kInt : at::kShort);
  auto out = at::empty({n}, packed.options().dtype(dt));
  at::Tensor valid;
  if (has_nulls) {
    valid = at::empty({n}, packed.options().dtype(at::kBool));
  } else {
    valid = at::empty({0}, packed.options().dtype(at::kBool));
  }
  int64_t mask = width >= 64 ? -1 : ((int64_t(1) << width) - 1);
  launch_unpack_col(packed.data_ptr<int64_t>(), n, (int)shift, mask, lo,
                    (int)dwidth, has_nulls ? 1 : 0, out.data_ptr(),
                    has_nulls ? valid.data_ptr<bool>() : nullptr,
                    current_stream());
  return {out, valid};
}

// Deterministic compaction of the group hash table: drops EMPTY slots
// from tkeys/gcount/gaggs in slot order.  Returns capacity-sized outputs
// plus the block-base scan whose last element is the total (single host
// read; caller narrows).
std::vector<at::Tensor> gb_compact(at::Tensor tkeys, at::Tensor gcount,
                                   at::Tensor gaggs,
                                   c10::optional<at::Tensor> extra) {
  check_gpu(tkeys, "tkeys");
  check_gpu(gcount, "gcount");
  int64_t tsize = tkeys.numel();
  int n_aggs = gaggs.numel() > 0 ? (int)gaggs.size(0) : 0;
  TORCH_CHECK(n_aggs <= 6, "at most 6 aggregate columns");
  const int64_t chunk = 256 * 32;
  int64_t nblocks = (tsize + chunk - 1) / chunk;
  auto bcounts = at::empty({nblocks}, tkeys.options());
  auto bases = at::empty({nblocks + 1}, tkeys.options());
  auto out_keys = at::empty({tsize}, tkeys.options());
  auto out_count = at::empty({tsize}, gcount.options());
  at::Tensor out_aggs = at::empty({n_aggs, tsize}, gaggs.options());
  const double* asrc[6];
  double* adst[6];
  for (int a = 0; a < n_aggs; ++a) {
    asrc[a] = gaggs[a].data_ptr<double>();
    adst[a] = out_aggs[a].data_ptr<double>();
  }
  at::Tensor out_extra;
  const int64_t* esrc = nullptr;
  int64_t* edst = nullptr;
  if (extra.has_value()) {
    check_gpu(*extra, "extra");
    out_extra = at::empty({tsize}, extra->options());
    esrc = extra->data_ptr<int64_t>();
    edst = out_extra.data_ptr<int64_t>();
  } else {
    out_extra = at::empty({0}, tkeys.options());
  }
  launch_gb_compact(tkeys.data_ptr<int64_t>(), gcount.data_ptr<int64_t>(),
                    asrc, adst, n_aggs, tsize, bcounts.data_ptr<int64_t>(),
                    bases.data_ptr<int64_t>(), out_keys.data_ptr<int64_t>(),
                    out_count.data_ptr<int64_t>(), esrc, edst,
                    current_stream());
  return {out_keys, out_count, out_aggs, out_extra, bases};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("hash_column", &hash_column,
        "combine a column into the running row hash");
  m.def("bucket_of", &bucket_of, "hash -> bucket id");
  m.def("bucket_histogram", &bucket_histogram, "bucket histogram");
  m.def("bucket_scatter", &bucket_scatter,
        "scatter row indices into bucket-contiguous order");
  m.def("gb_aggregate", &gb_aggregate, "hash group-by aggregation");
  m.def("gb_aggregate_replay", &gb_aggregate_replay,
        "re-aggregate with a recorded shuffle layout");
  m.def("gb_aggregate_partitioned", &gb_aggregate_partitioned,
        "partitioned (2-phase) hash group-by aggregation");
  m.def("join_build", &join_build, "build chained hash table");
  m.def("join_emit_unique", &join_emit_unique,
        "single-pass join emit for unique build keys");
  m.def("hash_string_column", &hash_string_column,
        "combine a string column into the running row hash");
  m.def("hash_seed", &hash_seed, "seed a row-hash buffer");
  m.def("gb_mark_reps", &gb_mark_reps,
        "representative rows + h2 verification for hashed group-by");
  m.def("expr_value", &expr_value,
        "fused value-expression interpreter");
  m.def("expr_filter", &expr_filter,
        "fused filter-predicate interpreter");
  m.def("gather_columns", &gather_columns,
        "fused multi-column row gather");
  m.def("gb_key_stats", &gb_key_stats,
        "fused key min/max + sampled distinct estimate");
  m.def("pack_columns", &pack_columns, "pack key columns into int64");
  m.def("unpack_column", &unpack_column, "unpack one key column");
  m.def("gb_compact", &gb_compact,
        "deterministic group-table compaction");
  m.def("compact_columns_cap", &compact_columns_cap,
        "mask compaction, capacity outputs + cursor");
  m.def("join_open_unique", &join_open_unique,
        "open-addressed unique int join (build+probe)");
  m.def("topk_select", &topk_select, "own top-k (k<=16) select");
  m.def("eq2_mask", &eq2_mask, "128-bit string equality mask");
  m.def("cmp_imm", &cmp_imm, "single col-vs-literal comparison mask");
  m.def("cmp_col", &cmp_col, "single col-vs-col comparison mask");
  m.def("compact_columns", &compact_columns,
        "fused masked compaction of 8-byte columns");
  m.def("join_count", &join_count, "count matches per probe row");
  m.def("join_emit", &join_emit, "emit join pairs");
  m.def("reduce_columns", &reduce_columns,
        "global column reductions (MFMA-reduced sums)");
  m.def("join_pairs", &join_pairs,
        "total + chunked-reservation join pair emission");
  m.def("join_mark_build", &join_mark_build, "mark matched build rows");
}
