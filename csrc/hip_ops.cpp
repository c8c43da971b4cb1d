// Torch bindings for greptimedb_amd._hip_ops (see kernels.hip).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <cstdint>
#include <vector>

namespace gdb_hip {
void launch_ts_bucket_agg(
    const int64_t*, const int32_t*, const double*, int64_t, const int32_t*, int,
    const int32_t*, int, int64_t, int64_t, int64_t, int64_t, int, int, int64_t,
    double*, unsigned long long*, unsigned long long*, unsigned long long*,
    unsigned long long*, hipStream_t);
void launch_decode_minmax(
    const unsigned long long*, const unsigned long long*, const unsigned long long*,
    double*, double*, int64_t, hipStream_t);
void launch_filter_series_time(
    const int64_t*, const int32_t*, const int32_t*, int, int64_t, int64_t,
    int64_t, bool*, hipStream_t);
void launch_dedup_mark_last(const int32_t*, const int64_t*, int64_t, bool*, hipStream_t);
void launch_bucket_agg2(
    const int64_t*, const int32_t*, const double*, int64_t, const int32_t*, int,
    const int32_t*, int, int64_t, int64_t, int64_t, int64_t, int, int, int64_t,
    int32_t*, double*, unsigned long long*, unsigned long long*,
    unsigned long long*, unsigned long long*, hipStream_t);
void launch_series_last(
    const int64_t*, const int32_t*, const int32_t*, int, int64_t, int64_t,
    int64_t, unsigned long long*, hipStream_t);
void launch_series_last_row(
    const int64_t*, const int32_t*, const int32_t*, int, int64_t, int64_t,
    int64_t, const unsigned long long*, unsigned long long,
    unsigned long long*, hipStream_t);
void launch_prom_range_eval(
    const int64_t*, const double*, const int64_t*, const int64_t*, int, int,
    int64_t, int64_t, int64_t, int64_t, double, int, double*, hipStream_t);
void launch_scatter_append(
    const int64_t*, const int32_t*, const double*, const int32_t*,
    const int64_t*, int64_t* const*, int32_t* const*, double* const*,
    const int64_t*, int, int64_t, hipStream_t);
void launch_rle_expand_indices(const int64_t*, int64_t, const uint8_t*, int,
                               int32_t*, int64_t, hipStream_t);
void launch_gorilla_decode(const uint8_t*, const int64_t*, const int64_t*,
                           int64_t, int64_t*, double*, int64_t, hipStream_t);
}  // namespace gdb_hip

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

// Fused filter + time-bucket aggregate (K1+K2+K5).
// ts: i64[n] (ms), series: i32[n], fields: f64[nf_total, field_stride]
// field_idx: i32[nf] rows of `fields` to aggregate, slot_lut: i32[lut_size].
// Returns (sum f64, count i64, min f64, max f64), each [nf, n_slots, n_buckets].
// When `acc` (5 tensors from a prior call) is passed, accumulation continues
// into the same buffers — multi-source scans make ONE set of outputs with no
// per-source combine (atomics make cross-launch accumulation safe).
std::vector<torch::Tensor> ts_bucket_agg(
    torch::Tensor ts, torch::Tensor series, torch::Tensor fields,
    torch::Tensor field_idx, torch::Tensor slot_lut,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int64_t n_slots, int64_t n_buckets,
    std::vector<torch::Tensor> acc) {
  CHECK_GPU(ts); CHECK_GPU(series); CHECK_GPU(fields);
  CHECK_GPU(field_idx); CHECK_GPU(slot_lut);
  CHECK_CONTIG(ts); CHECK_CONTIG(series); CHECK_CONTIG(fields);
  CHECK_CONTIG(field_idx); CHECK_CONTIG(slot_lut);
  TORCH_CHECK(ts.scalar_type() == torch::kInt64);
  TORCH_CHECK(series.scalar_type() == torch::kInt32);
  TORCH_CHECK(fields.scalar_type() == torch::kFloat64);
  TORCH_CHECK(fields.dim() == 2);
  const int64_t n = ts.numel();
  TORCH_CHECK(series.numel() == n);
  TORCH_CHECK(fields.size(1) >= n, "fields stride shorter than rows");
  const int nf = (int)field_idx.numel();
  auto opts_f64 = ts.options().dtype(torch::kFloat64);
  auto opts_i64 = ts.options().dtype(torch::kInt64);
  const int64_t cells = nf * n_slots * n_buckets;
  torch::Tensor sum, cnt, rows, minmax_init_min, minmax_init_max;
  if (!acc.empty()) {
    TORCH_CHECK(acc.size() == 5, "acc must be the 5 raw accumulators");
    sum = acc[0]; cnt = acc[1]; minmax_init_min = acc[2];
    minmax_init_max = acc[3]; rows = acc[4];
  } else {
    sum = torch::zeros({nf, n_slots, n_buckets}, opts_f64);
    cnt = torch::zeros({nf, n_slots, n_buckets}, opts_i64);
    rows = torch::zeros({n_slots, n_buckets}, opts_i64);
    // min keys init to u64 max, max keys to 0
    minmax_init_min = torch::full({nf, n_slots, n_buckets}, -1, opts_i64);
    minmax_init_max = torch::zeros({nf, n_slots, n_buckets}, opts_i64);
  }
  auto stream = at::cuda::getCurrentHIPStream().stream();
  auto cell = torch::empty({n}, ts.options().dtype(torch::kInt32));
  gdb_hip::launch_bucket_agg2(
      ts.data_ptr<int64_t>(), series.data_ptr<int32_t>(), fields.data_ptr<double>(),
      fields.size(1), field_idx.data_ptr<int32_t>(), nf,
      slot_lut.data_ptr<int32_t>(), (int)slot_lut.numel(),
      ts_lo, ts_hi, origin, bucket_ms, (int)n_slots, (int)n_buckets, n,
      cell.data_ptr<int32_t>(),
      sum.data_ptr<double>(),
      reinterpret_cast<unsigned long long*>(cnt.data_ptr<int64_t>()),
      reinterpret_cast<unsigned long long*>(minmax_init_min.data_ptr<int64_t>()),
      reinterpret_cast<unsigned long long*>(minmax_init_max.data_ptr<int64_t>()),
      reinterpret_cast<unsigned long long*>(rows.data_ptr<int64_t>()),
      stream);
  return {sum, cnt, minmax_init_min, minmax_init_max, rows};
}

// Decode raw accumulators (min/max u64 keys) → final (sum, cnt, min f64,
// max f64, rows). Call once after the last ts_bucket_agg of a scan.
std::vector<torch::Tensor> ts_bucket_agg_finish(std::vector<torch::Tensor> acc) {
  TORCH_CHECK(acc.size() == 5);
  auto sum = acc[0];
  auto cnt = acc[1];
  const int64_t cells = cnt.numel();
  auto minv = torch::empty_like(sum);
  auto maxv = torch::empty_like(sum);
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_decode_minmax(
      reinterpret_cast<unsigned long long*>(acc[2].data_ptr<int64_t>()),
      reinterpret_cast<unsigned long long*>(acc[3].data_ptr<int64_t>()),
      reinterpret_cast<unsigned long long*>(cnt.data_ptr<int64_t>()),
      minv.data_ptr<double>(), maxv.data_ptr<double>(), cells, stream);
  return {sum, cnt, minv, maxv, acc[4]};
}

torch::Tensor filter_series_time(
    torch::Tensor ts, torch::Tensor series, torch::Tensor slot_lut,
    int64_t ts_lo, int64_t ts_hi) {
  CHECK_GPU(ts); CHECK_CONTIG(ts);
  const int64_t n = ts.numel();
  auto keep = torch::empty({n}, ts.options().dtype(torch::kBool));
  const int32_t* lut = nullptr;
  int lut_size = 0;
  if (slot_lut.numel() > 0) {
    CHECK_GPU(slot_lut); CHECK_CONTIG(slot_lut);
    lut = slot_lut.data_ptr<int32_t>();
    lut_size = (int)slot_lut.numel();
  }
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_filter_series_time(
      ts.data_ptr<int64_t>(), series.data_ptr<int32_t>(), lut, lut_size,
      ts_lo, ts_hi, n, keep.data_ptr<bool>(), stream);
  return keep;
}

torch::Tensor dedup_mark_last(torch::Tensor series, torch::Tensor ts) {
  CHECK_GPU(series); CHECK_GPU(ts); CHECK_CONTIG(series); CHECK_CONTIG(ts);
  const int64_t n = ts.numel();
  auto keep = torch::empty({n}, ts.options().dtype(torch::kBool));
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_dedup_mark_last(
      series.data_ptr<int32_t>(), ts.data_ptr<int64_t>(), n,
      keep.data_ptr<bool>(), stream);
  return keep;
}

// Accumulate per-slot max-ts keys across sources (call once per source with
// a shared best_key tensor, int64 viewed as u64 keys, init 0).
void series_last_ts(torch::Tensor ts, torch::Tensor series, torch::Tensor slot_lut,
                    int64_t ts_lo, int64_t ts_hi, torch::Tensor best_key) {
  CHECK_GPU(ts); CHECK_CONTIG(ts); CHECK_GPU(best_key); CHECK_CONTIG(best_key);
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_series_last(
      ts.data_ptr<int64_t>(), series.data_ptr<int32_t>(),
      slot_lut.data_ptr<int32_t>(), (int)slot_lut.numel(), ts_lo, ts_hi,
      ts.numel(),
      reinterpret_cast<unsigned long long*>(best_key.data_ptr<int64_t>()),
      stream);
}

void series_last_row(torch::Tensor ts, torch::Tensor series, torch::Tensor slot_lut,
                     int64_t ts_lo, int64_t ts_hi, torch::Tensor best_key,
                     int64_t src_tag, torch::Tensor best_row) {
  CHECK_GPU(ts); CHECK_CONTIG(ts);
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_series_last_row(
      ts.data_ptr<int64_t>(), series.data_ptr<int32_t>(),
      slot_lut.data_ptr<int32_t>(), (int)slot_lut.numel(), ts_lo, ts_hi,
      ts.numel(),
      reinterpret_cast<const unsigned long long*>(best_key.data_ptr<int64_t>()),
      (unsigned long long)src_tag,
      reinterpret_cast<unsigned long long*>(best_row.data_ptr<int64_t>()),
      stream);
}

// PromQL range/instant evaluation over (slot, ts)-sorted samples.
torch::Tensor prom_range_eval(
    torch::Tensor ts, torch::Tensor vals, torch::Tensor seg_lo, torch::Tensor seg_hi,
    int64_t T, int64_t t0, int64_t step_ms, int64_t range_ms, int64_t offset_ms,
    double param, int64_t mode) {
  CHECK_GPU(ts); CHECK_CONTIG(ts); CHECK_GPU(vals); CHECK_CONTIG(vals);
  CHECK_GPU(seg_lo); CHECK_CONTIG(seg_lo); CHECK_GPU(seg_hi); CHECK_CONTIG(seg_hi);
  const int S = (int)seg_lo.numel();
  auto out = torch::empty({S, T}, vals.options());
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_prom_range_eval(
      ts.data_ptr<int64_t>(), vals.data_ptr<double>(),
      seg_lo.data_ptr<int64_t>(), seg_hi.data_ptr<int64_t>(),
      S, (int)T, t0, step_ms, range_ms, offset_ms, param, (int)mode,
      out.data_ptr<double>(), stream);
  return out;
}

// K16 bulk append: scatter one routed batch into multiple region memtables.
// `dests` = per-region (ts, series, fields) destination tensors.
void scatter_append(
    torch::Tensor ts, torch::Tensor series, torch::Tensor fields,
    torch::Tensor region_of, torch::Tensor dst_off,
    std::vector<torch::Tensor> dst_ts, std::vector<torch::Tensor> dst_se,
    std::vector<torch::Tensor> dst_fields) {
  CHECK_GPU(ts); CHECK_CONTIG(ts); CHECK_GPU(fields); CHECK_CONTIG(fields);
  CHECK_GPU(region_of); CHECK_CONTIG(region_of);
  CHECK_GPU(dst_off); CHECK_CONTIG(dst_off);
  const int64_t n = ts.numel();
  const int nf = (int)fields.size(0);
  const int R = (int)dst_ts.size();
  std::vector<int64_t> hptrs(R * 3 + R);
  for (int r = 0; r < R; r++) {
    hptrs[r] = (int64_t)dst_ts[r].data_ptr<int64_t>();
    hptrs[R + r] = (int64_t)dst_se[r].data_ptr<int32_t>();
    hptrs[2 * R + r] = (int64_t)dst_fields[r].data_ptr<double>();
    hptrs[3 * R + r] = dst_fields[r].size(1);  // stride = cap
  }
  auto ptrs = torch::from_blob(hptrs.data(), {(int64_t)hptrs.size()},
                               torch::kInt64).to(ts.device());
  auto stream = at::cuda::getCurrentHIPStream().stream();
  const int64_t* pbase = ptrs.data_ptr<int64_t>();
  gdb_hip::launch_scatter_append(
      ts.data_ptr<int64_t>(), series.data_ptr<int32_t>(),
      fields.data_ptr<double>(), region_of.data_ptr<int32_t>(),
      dst_off.data_ptr<int64_t>(),
      reinterpret_cast<int64_t* const*>(pbase),
      reinterpret_cast<int32_t* const*>(pbase + R),
      reinterpret_cast<double* const*>(pbase + 2 * R),
      pbase + 3 * R, nf, n, stream);
}


// K11: parquet hybrid RLE/bit-packed dictionary indices → i32 on device.
// runs: i64[R,5] (host-built, csrc/pagedec.cpp), blob: u8 device tensor
// (pad >= 8 bytes past the last packed bit).
torch::Tensor rle_expand_indices(torch::Tensor runs, torch::Tensor blob,
                                 int64_t bw, int64_t n) {
  CHECK_GPU(runs); CHECK_CONTIG(runs); CHECK_GPU(blob); CHECK_CONTIG(blob);
  auto out = torch::empty({n}, blob.options().dtype(torch::kInt32));
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_rle_expand_indices(
      runs.data_ptr<int64_t>(), runs.size(0), blob.data_ptr<uint8_t>(),
      (int)bw, out.data_ptr<int32_t>(), n, stream);
  return out;
}

// K20: Gorilla/delta block decode → (ts i64[n], vals f64[n]) on device.
std::vector<torch::Tensor> gorilla_decode(torch::Tensor blob,
                                          torch::Tensor block_off,
                                          torch::Tensor out_off, int64_t n) {
  CHECK_GPU(blob); CHECK_CONTIG(blob);
  CHECK_GPU(block_off); CHECK_GPU(out_off);
  auto ts = torch::empty({n}, blob.options().dtype(torch::kInt64));
  auto vals = torch::empty({n}, blob.options().dtype(torch::kFloat64));
  auto stream = at::cuda::getCurrentHIPStream().stream();
  gdb_hip::launch_gorilla_decode(
      blob.data_ptr<uint8_t>(), block_off.data_ptr<int64_t>(),
      out_off.data_ptr<int64_t>(), block_off.numel(),
      ts.data_ptr<int64_t>(), vals.data_ptr<double>(), n, stream);
  return {ts, vals};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("ts_bucket_agg", &ts_bucket_agg, "fused filter + time-bucket aggregate",
        py::arg("ts"), py::arg("series"), py::arg("fields"), py::arg("field_idx"),
        py::arg("slot_lut"), py::arg("ts_lo"), py::arg("ts_hi"), py::arg("origin"),
        py::arg("bucket_ms"), py::arg("n_slots"), py::arg("n_buckets"),
        py::arg("acc") = std::vector<torch::Tensor>());
  m.def("ts_bucket_agg_finish", &ts_bucket_agg_finish, "decode raw accumulators");
  m.def("rle_expand_indices", &rle_expand_indices, "K11 hybrid RLE expand");
  m.def("gorilla_decode", &gorilla_decode, "K20 gorilla block decode");
  m.def("filter_series_time", &filter_series_time, "series/time filter mask");
  m.def("dedup_mark_last", &dedup_mark_last, "last-row dedup marker");
  m.def("series_last_ts", &series_last_ts, "per-slot max-ts accumulate (lastpoint)");
  m.def("series_last_row", &series_last_row, "per-slot winner row (lastpoint)");
  m.def("prom_range_eval", &prom_range_eval, "PromQL range-vector evaluator");
  m.def("scatter_append", &scatter_append, "bulk routed memtable append (K16)");
}
