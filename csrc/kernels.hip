// CDNA4 (gfx950 / MI355X) kernels for the mito-hip scan path.
//
// Kernel inventory (SURVEY.md §2.7 numbering):
//  - ts_bucket_agg_kernel : fused K1 (predicate filter via series LUT) +
//    K2 (time-range visibility) + K5 (time-bucket aggregate). One pass over
//    the (ts, series, fields) columns computing sum/count/min/max for every
//    requested field into [nf, n_slots, n_buckets] accumulators.
//    Memory-bound by design: ts+series loaded once per row and amortized
//    over all selected fields.
//  - filter_series_time_kernel : K1/K2 producing a keep-mask for raw scans.
//  - dedup_mark_last_kernel : K4 last_row marker on (series, ts)-sorted data.
//
// Design notes (cdna_hip_programming.md):
//  * 256-thread blocks (4 waves of 64), grid-stride, grid capped at
//    2048 blocks (G11: memory-bound ops).
//  * f64 min/max accumulate through the monotonic u64 key mapping so we can
//    use hardware atomicMin/Max on unsigned long long (no f64 min/max atomic).
//  * NaN field values are nulls (influx missing fields) and are skipped.
//  * Small aggregate tables (slots*buckets) stay hot in L2/LLC; per-LDS
//    privatization is a planned optimization once rocprof shows atomic
//    contention (expected only for >100k groups).

#include <hip/hip_runtime.h>
#include <cstdint>

#define DEV_INLINE __device__ __forceinline__

namespace gdb_hip {

DEV_INLINE uint64_t f64_to_key(double d) {
  uint64_t bits = __double_as_longlong(d);
  return (int64_t)bits < 0 ? ~bits : (bits | 0x8000000000000000ULL);
}

DEV_INLINE double key_to_f64(uint64_t key) {
  uint64_t bits = (key & 0x8000000000000000ULL) ? (key & 0x7FFFFFFFFFFFFFFFULL) : ~key;
  return __longlong_as_double(bits);
}

// ---------------------------------------------------------------- K1+K2+K5
// fields: column-major per field: fields[f * field_stride + row]
// out arrays: [nf, n_slots, n_buckets]
__global__ void ts_bucket_agg_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const double* __restrict__ fields,
    int64_t field_stride,
    const int32_t* __restrict__ field_idx, int nf,
    const int32_t* __restrict__ slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int n_slots, int n_buckets, int64_t n,
    double* __restrict__ out_sum,
    unsigned long long* __restrict__ out_cnt,
    unsigned long long* __restrict__ out_min,
    unsigned long long* __restrict__ out_max,
    unsigned long long* __restrict__ out_rows) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t t = ts[i];
    if (t < ts_lo || t >= ts_hi) continue;
    const int32_t s = series[i];
    if (s < 0 || s >= lut_size) continue;
    const int32_t slot = slot_lut[s];
    if (slot < 0) continue;
    int64_t b = (t - origin) / bucket_ms;
    if (b < 0 || b >= n_buckets) continue;
    const int64_t cell0 = (int64_t)slot * n_buckets + b;
    atomicAdd(&out_rows[cell0], 1ULL);
    for (int f = 0; f < nf; f++) {
      const double v = fields[(int64_t)field_idx[f] * field_stride + i];
      if (isnan(v)) continue;
      const int64_t cell = (int64_t)f * n_slots * n_buckets + cell0;
      atomicAdd(&out_sum[cell], v);
      atomicAdd(&out_cnt[cell], 1ULL);
      const uint64_t k = f64_to_key(v);
      atomicMin(&out_min[cell], (unsigned long long)k);
      atomicMax(&out_max[cell], (unsigned long long)k);
    }
  }
}

// decode min/max key arrays into f64 (count==0 → NaN)
__global__ void decode_minmax_kernel(
    const unsigned long long* __restrict__ minkey,
    const unsigned long long* __restrict__ maxkey,
    const unsigned long long* __restrict__ cnt,
    double* __restrict__ out_min, double* __restrict__ out_max, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    if (cnt[i] == 0) {
      out_min[i] = __longlong_as_double(0x7FF8000000000000LL);  // NaN
      out_max[i] = __longlong_as_double(0x7FF8000000000000LL);
    } else {
      out_min[i] = key_to_f64(minkey[i]);
      out_max[i] = key_to_f64(maxkey[i]);
    }
  }
}

// ---------------------------------------------------------------- K1/K2 mask
__global__ void filter_series_time_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const int32_t* __restrict__ slot_lut, int lut_size,  // lut==nullptr → all series
    int64_t ts_lo, int64_t ts_hi, int64_t n,
    bool* __restrict__ keep) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t t = ts[i];
    bool k = (t >= ts_lo) & (t < ts_hi);
    if (k && slot_lut != nullptr) {
      const int32_t s = series[i];
      k = (s >= 0) & (s < lut_size) && (slot_lut[s] >= 0);
    }
    keep[i] = k;
  }
}

// ---------------------------------------------------------------- K4 dedup
// Input sorted ascending by (series, ts, seq). Keep the LAST row of each
// (series, ts) group (reference read/dedup.rs LastRow semantics).
__global__ void dedup_mark_last_kernel(
    const int32_t* __restrict__ series,
    const int64_t* __restrict__ ts,
    int64_t n, bool* __restrict__ keep) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    keep[i] = (i == n - 1) || (series[i] != series[i + 1]) || (ts[i] != ts[i + 1]);
  }
}

// ------------------------------------------------- K5 two-phase (sorted-aware)
//
// The naive per-row atomic version above costs nf*4+1 atomics PER ROW —
// rocprof showed it running at ~1.4% of HBM bandwidth on the TSBS
// double-groupby shape (4.2B atomic RMWs over 104M rows). Scan sources are
// (series, ts)-sorted, so consecutive rows usually land in the SAME
// (slot, bucket) cell: phase A computes the cell id per row once (reused by
// every field); phase B gives each thread a contiguous row chunk per field
// and accumulates in REGISTERS, flushing atomics only when the cell changes
// (≈ once per run instead of once per row). Correct for unsorted input too
// (degenerates to per-row flushes).

__global__ void bucket_cell_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const int32_t* __restrict__ slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int n_buckets, int64_t n,
    int32_t* __restrict__ cell,                 // [n] slot*n_buckets+b or -1
    unsigned long long* __restrict__ out_rows,  // [n_slots*n_buckets]
    int rows_chunk) {
  // rows-count accumulation with chunked run detection
  const int64_t nchunks = (n + rows_chunk - 1) / rows_chunk;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; c < nchunks; c += stride) {
    const int64_t lo = c * rows_chunk;
    const int64_t hi = min(lo + rows_chunk, n);
    int32_t cur = -1;
    unsigned long long run = 0;
    for (int64_t i = lo; i < hi; i++) {
      const int64_t t = ts[i];
      const int32_t s = series[i];
      int32_t cl = -1;
      if (t >= ts_lo && t < ts_hi && s >= 0 && s < lut_size) {
        const int32_t slot = slot_lut[s];
        if (slot >= 0) {
          const int64_t b = (t - origin) / bucket_ms;
          if (b >= 0 && b < n_buckets) cl = slot * n_buckets + (int32_t)b;
        }
      }
      cell[i] = cl;
      if (cl == cur) {
        run++;
      } else {
        if (cur >= 0 && run) atomicAdd(&out_rows[cur], run);
        cur = cl;
        run = 1;
      }
    }
    if (cur >= 0 && run) atomicAdd(&out_rows[cur], run);
  }
}

__global__ void field_bucket_agg_kernel(
    const int32_t* __restrict__ cell,
    const double* __restrict__ fields, int64_t field_stride,
    const int32_t* __restrict__ field_idx, int nf,
    int64_t n, int64_t cells_per_field, int rows_chunk,
    double* __restrict__ out_sum,
    unsigned long long* __restrict__ out_cnt,
    unsigned long long* __restrict__ out_min,
    unsigned long long* __restrict__ out_max) {
  const int64_t nchunks = (n + rows_chunk - 1) / rows_chunk;
  const int64_t total = nchunks * nf;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; t < total; t += stride) {
    // consecutive threads take consecutive chunks of the SAME field
    const int f = (int)(t / nchunks);
    const int64_t c = t % nchunks;
    const double* __restrict__ col = fields + (int64_t)field_idx[f] * field_stride;
    double* __restrict__ o_sum = out_sum + (int64_t)f * cells_per_field;
    unsigned long long* __restrict__ o_cnt = out_cnt + (int64_t)f * cells_per_field;
    unsigned long long* __restrict__ o_min = out_min + (int64_t)f * cells_per_field;
    unsigned long long* __restrict__ o_max = out_max + (int64_t)f * cells_per_field;
    const int64_t lo = c * rows_chunk;
    const int64_t hi = min(lo + rows_chunk, n);
    int32_t cur = -1;
    double a_sum = 0.0;
    unsigned long long a_cnt = 0;
    uint64_t a_min = ~0ULL, a_max = 0ULL;
    for (int64_t i = lo; i < hi; i++) {
      const int32_t cl = cell[i];
      if (cl != cur) {
        if (cur >= 0 && a_cnt) {
          atomicAdd(&o_sum[cur], a_sum);
          atomicAdd(&o_cnt[cur], a_cnt);
          atomicMin(&o_min[cur], a_min);
          atomicMax(&o_max[cur], a_max);
        }
        cur = cl; a_sum = 0.0; a_cnt = 0; a_min = ~0ULL; a_max = 0ULL;
      }
      if (cl < 0) continue;
      const double v = col[i];
      if (isnan(v)) continue;
      a_sum += v;
      a_cnt++;
      const uint64_t k = f64_to_key(v);
      a_min = min(a_min, (uint64_t)k);
      a_max = max(a_max, (uint64_t)k);
    }
    if (cur >= 0 && a_cnt) {
      atomicAdd(&o_sum[cur], a_sum);
      atomicAdd(&o_cnt[cur], a_cnt);
      atomicMin(&o_min[cur], a_min);
      atomicMax(&o_max[cur], a_max);
    }
  }
}

// ---------------------------------------------- K5 v3: LDS-privatized tiles
//
// PMC (FETCH_SIZE) showed the chunked-run kernels fetching 8-10x the input
// bytes: per-thread contiguous chunks make every wavefront touch 64 cache
// lines per step (uncoalesced). v3 restores coalescing — threads stride the
// tile contiguously — and privatizes the accumulator table in LDS: a tile
// of sorted rows spans only a few (slot,bucket) cells, so per-element LDS
// atomics replace global ones, flushed once per tile. Tiles whose cell
// range exceeds the LDS table fall back to global atomics (still with
// coalesced loads).

__global__ void bucket_cell_coalesced_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const int32_t* __restrict__ slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int n_buckets, int64_t n,
    int32_t* __restrict__ cell) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t t = ts[i];
    const int32_t sId = series[i];
    int32_t cl = -1;
    if (t >= ts_lo && t < ts_hi && sId >= 0 && sId < lut_size) {
      const int32_t slot = slot_lut[sId];
      if (slot >= 0) {
        const int64_t b = (t - origin) / bucket_ms;
        if (b >= 0 && b < n_buckets) cl = slot * n_buckets + (int32_t)b;
      }
    }
    cell[i] = cl;
  }
}

#define AGG_TILE 8192
#define LDS_CELLS 1024

__global__ void field_bucket_agg_lds_kernel(
    const int32_t* __restrict__ cell,
    const double* __restrict__ fields, int64_t field_stride,
    const int32_t* __restrict__ field_idx, int nf,
    int64_t n, int64_t cells_per_field,
    double* __restrict__ out_sum,
    unsigned long long* __restrict__ out_cnt,
    unsigned long long* __restrict__ out_min,
    unsigned long long* __restrict__ out_max,
    unsigned long long* __restrict__ out_rows) {
  __shared__ int32_t s_lo, s_hi;
  __shared__ double s_sum[LDS_CELLS];
  __shared__ unsigned long long s_cnt[LDS_CELLS];
  __shared__ unsigned long long s_min[LDS_CELLS];
  __shared__ unsigned long long s_max[LDS_CELLS];
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const int64_t ntiles = (n + AGG_TILE - 1) / AGG_TILE;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * AGG_TILE;
    const int64_t t1 = min(t0 + (int64_t)AGG_TILE, n);
    if (tid == 0) { s_lo = INT32_MAX; s_hi = -1; }
    __syncthreads();
    int32_t my_lo = INT32_MAX, my_hi = -1;
    for (int64_t i = t0 + tid; i < t1; i += nthreads) {
      const int32_t c = cell[i];
      if (c >= 0) { my_lo = min(my_lo, c); my_hi = max(my_hi, c); }
    }
    if (my_hi >= 0) {
      atomicMin(&s_lo, my_lo);
      atomicMax(&s_hi, my_hi);
    }
    __syncthreads();
    const int32_t lo = s_lo, hi = s_hi;
    __syncthreads();
    if (hi < 0) continue;                       // tile fully filtered out
    const int range = hi - lo + 1;
    if (range <= LDS_CELLS) {
      // rows count once per tile (reuse s_cnt as the rows table)
      for (int k = tid; k < range; k += nthreads) s_cnt[k] = 0ULL;
      __syncthreads();
      for (int64_t i = t0 + tid; i < t1; i += nthreads) {
        const int32_t c = cell[i];
        if (c >= 0) atomicAdd(&s_cnt[c - lo], 1ULL);
      }
      __syncthreads();
      for (int k = tid; k < range; k += nthreads)
        if (s_cnt[k]) atomicAdd(&out_rows[lo + k], s_cnt[k]);
      __syncthreads();
      for (int f = 0; f < nf; f++) {
        const double* __restrict__ col =
            fields + (int64_t)field_idx[f] * field_stride;
        for (int k = tid; k < range; k += nthreads) {
          s_sum[k] = 0.0; s_cnt[k] = 0ULL; s_min[k] = ~0ULL; s_max[k] = 0ULL;
        }
        __syncthreads();
        for (int64_t i = t0 + tid; i < t1; i += nthreads) {
          const int32_t c = cell[i];
          if (c < 0) continue;
          const double v = col[i];
          if (isnan(v)) continue;
          const int k = c - lo;
          atomicAdd(&s_sum[k], v);
          atomicAdd(&s_cnt[k], 1ULL);
          const uint64_t key = f64_to_key(v);
          atomicMin(&s_min[k], (unsigned long long)key);
          atomicMax(&s_max[k], (unsigned long long)key);
        }
        __syncthreads();
        const int64_t base = (int64_t)f * cells_per_field;
        for (int k = tid; k < range; k += nthreads) {
          if (!s_cnt[k]) continue;
          atomicAdd(&out_sum[base + lo + k], s_sum[k]);
          atomicAdd(&out_cnt[base + lo + k], s_cnt[k]);
          atomicMin(&out_min[base + lo + k], s_min[k]);
          atomicMax(&out_max[base + lo + k], s_max[k]);
        }
        __syncthreads();
      }
    } else {
      // wide tile (unsorted data): coalesced loads, global atomics
      for (int64_t i = t0 + tid; i < t1; i += nthreads) {
        const int32_t c = cell[i];
        if (c >= 0) atomicAdd(&out_rows[c], 1ULL);
      }
      for (int f = 0; f < nf; f++) {
        const double* __restrict__ col =
            fields + (int64_t)field_idx[f] * field_stride;
        const int64_t base = (int64_t)f * cells_per_field;
        for (int64_t i = t0 + tid; i < t1; i += nthreads) {
          const int32_t c = cell[i];
          if (c < 0) continue;
          const double v = col[i];
          if (isnan(v)) continue;
          atomicAdd(&out_sum[base + c], v);
          atomicAdd(&out_cnt[base + c], 1ULL);
          const uint64_t key = f64_to_key(v);
          atomicMin(&out_min[base + c], (unsigned long long)key);
          atomicMax(&out_max[base + c], (unsigned long long)key);
        }
      }
    }
    __syncthreads();
  }
}

// ------------------------------------------------- K-lastpoint (series last)
// Monotonic i64→u64 key so unsigned atomicMax orders signed timestamps.
DEV_INLINE uint64_t i64_to_key(int64_t v) {
  return (uint64_t)v ^ 0x8000000000000000ULL;
}

// pass 1: per-slot max ts (key-mapped); best init 0
__global__ void series_last_ts_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const int32_t* __restrict__ slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t n,
    unsigned long long* __restrict__ best_key) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t t = ts[i];
    if (t < ts_lo || t >= ts_hi) continue;
    const int32_t s = series[i];
    if (s < 0 || s >= lut_size) continue;
    const int32_t slot = slot_lut[s];
    if (slot < 0) continue;
    atomicMax(&best_key[slot], (unsigned long long)i64_to_key(t));
  }
}

// pass 2: among rows whose ts matches the winner, pick (src_tag, row) max —
// later sources / later arrivals win ties (LastRow semantics)
__global__ void series_last_row_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const int32_t* __restrict__ slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t n,
    const unsigned long long* __restrict__ best_key,
    unsigned long long src_tag,                    // source index (recency order)
    unsigned long long* __restrict__ best_row) {   // (src_tag<<40)|row ; init 0
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int64_t t = ts[i];
    if (t < ts_lo || t >= ts_hi) continue;
    const int32_t s = series[i];
    if (s < 0 || s >= lut_size) continue;
    const int32_t slot = slot_lut[s];
    if (slot < 0) continue;
    if ((unsigned long long)i64_to_key(t) != best_key[slot]) continue;
    // +1 so "no row" (0) is distinguishable from src 0 row 0
    atomicMax(&best_row[slot], (src_tag << 40) | ((unsigned long long)i + 1));
  }
}

// ------------------------------------------------- K7/K8 PromQL evaluators
//
// One kernel evaluates a PromQL selector/range-function over a
// (slot, ts)-sorted sample array onto the query step grid. Thread =
// (slot, step): binary-search the window bounds inside the slot's segment,
// then reduce the window per `mode`. Rate/increase/delta follow Prometheus'
// extrapolation rules exactly (reference: promql/functions/extrapolate_rate.rs,
// itself a port of Prometheus' extrapolatedRate).

enum PromMode {
  PM_INSTANT = 0, PM_RATE = 1, PM_INCREASE = 2, PM_DELTA = 3,
  PM_AVG = 4, PM_SUM = 5, PM_MIN = 6, PM_MAX = 7, PM_COUNT = 8, PM_LAST = 9,
  PM_IDELTA = 10, PM_IRATE = 11, PM_DERIV = 12, PM_PREDICT = 13,
  PM_RESETS = 14, PM_CHANGES = 15, PM_STDDEV = 16, PM_STDVAR = 17,
  PM_ABSENT_OT = 18, PM_QUANTILE_OT = 19, PM_LAST_TS = 20,
};

DEV_INLINE double prom_nan() { return __longlong_as_double(0x7FF8000000000000LL); }

// IEEE-754 double ↔ totally-ordered u64 (sign-flip transform): key order ==
// numeric order, so order statistics can be selected bitwise.
DEV_INLINE uint64_t f64_sort_key(double v) {
  const uint64_t u = (uint64_t)__double_as_longlong(v);
  return (u & 0x8000000000000000ULL) ? ~u : (u | 0x8000000000000000ULL);
}
DEV_INLINE double sort_key_f64(uint64_t k) {
  const uint64_t u = (k & 0x8000000000000000ULL)
      ? (k ^ 0x8000000000000000ULL) : ~k;
  return __longlong_as_double((long long)u);
}

__global__ void prom_range_eval_kernel(
    const int64_t* __restrict__ ts,
    const double* __restrict__ vals,
    const int64_t* __restrict__ seg_lo,   // [S] first row of slot
    const int64_t* __restrict__ seg_hi,   // [S] one past last row
    int S, int T,
    int64_t t0, int64_t step_ms, int64_t range_ms, int64_t offset_ms,
    double param, int mode,
    double* __restrict__ out) {           // [S, T]
  const int64_t total = (int64_t)S * T;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const int s = (int)(i / T);
    const int t = (int)(i % T);
    const int64_t te = t0 + (int64_t)t * step_ms - offset_ms;   // window end (inclusive)
    const int64_t tb = te - range_ms;                           // window begin (exclusive)
    const int64_t lo = seg_lo[s], hi = seg_hi[s];
    // lower: first idx with ts > tb ; upper: last idx with ts <= te
    int64_t a = lo, b = hi;
    while (a < b) { int64_t m = (a + b) >> 1; if (ts[m] <= tb) a = m + 1; else b = m; }
    const int64_t w_lo = a;
    a = w_lo; b = hi;
    while (a < b) { int64_t m = (a + b) >> 1; if (ts[m] <= te) a = m + 1; else b = m; }
    const int64_t w_hi = a;                                     // exclusive
    const int64_t cnt = w_hi - w_lo;
    double r = prom_nan();
    if (mode == PM_INSTANT) {
      if (cnt > 0) r = vals[w_hi - 1];
    } else if (mode == PM_LAST) {
      if (cnt > 0) r = vals[w_hi - 1];
    } else if (mode == PM_LAST_TS) {
      // last sample timestamp as f64 (exact to 2^53 ms) — the cross-rank
      // last_value merge argmaxes on this plane
      if (cnt > 0) r = (double)ts[w_hi - 1];
    } else if (mode == PM_COUNT) {
      if (cnt > 0) r = (double)cnt;
    } else if (mode == PM_ABSENT_OT) {
      r = (cnt > 0) ? prom_nan() : 1.0;
    } else if (cnt > 0 && (mode == PM_SUM || mode == PM_AVG || mode == PM_MIN ||
                           mode == PM_MAX || mode == PM_STDDEV || mode == PM_STDVAR)) {
      double sum = 0, mn = vals[w_lo], mx = vals[w_lo];
      for (int64_t j = w_lo; j < w_hi; j++) {
        const double v = vals[j];
        sum += v; mn = fmin(mn, v); mx = fmax(mx, v);
      }
      if (mode == PM_SUM) r = sum;
      else if (mode == PM_AVG) r = sum / cnt;
      else if (mode == PM_MIN) r = mn;
      else if (mode == PM_MAX) r = mx;
      else {
        const double mean = sum / cnt;
        double aux = 0;
        for (int64_t j = w_lo; j < w_hi; j++) {
          const double d = vals[j] - mean;
          aux += d * d;
        }
        const double var = aux / cnt;
        r = (mode == PM_STDVAR) ? var : sqrt(var);
      }
    } else if (cnt >= 2 && (mode == PM_RATE || mode == PM_INCREASE || mode == PM_DELTA)) {
      const bool is_counter = (mode != PM_DELTA);
      const double first_v = vals[w_lo], last_v = vals[w_hi - 1];
      const int64_t first_t = ts[w_lo], last_t = ts[w_hi - 1];
      double total_incr = last_v - first_v;
      if (is_counter) {
        double prev = first_v;
        for (int64_t j = w_lo + 1; j < w_hi; j++) {
          const double v = vals[j];
          if (v < prev) total_incr += prev;   // counter reset correction
          prev = v;
        }
      }
      const double sampled = (double)(last_t - first_t) / 1000.0;
      const double range_s = (double)range_ms / 1000.0;
      const double avg_dur = sampled / (cnt - 1);
      double dur_to_start = (double)(first_t - tb) / 1000.0;
      const double dur_to_end = (double)(te - last_t) / 1000.0;
      if (is_counter && total_incr > 0 && first_v >= 0) {
        const double dur_to_zero = sampled * (first_v / total_incr);
        if (dur_to_zero < dur_to_start) dur_to_start = dur_to_zero;
      }
      const double thresh = avg_dur * 1.1;
      double ext = sampled;
      ext += (dur_to_start < thresh) ? dur_to_start : avg_dur / 2;
      ext += (dur_to_end < thresh) ? dur_to_end : avg_dur / 2;
      const double factor = (sampled > 0) ? ext / sampled : 1.0;
      double res = total_incr * factor;
      if (mode == PM_RATE) res /= range_s;
      r = res;
    } else if (cnt >= 2 && (mode == PM_IDELTA || mode == PM_IRATE)) {
      const double dv = vals[w_hi - 1] - vals[w_hi - 2];
      const double dt = (double)(ts[w_hi - 1] - ts[w_hi - 2]) / 1000.0;
      if (mode == PM_IDELTA) {
        r = dv;
      } else {
        double v2 = vals[w_hi - 1], v1 = vals[w_hi - 2];
        double d = (v2 < v1) ? v2 : dv;   // reset → use raw value
        r = (dt > 0) ? d / dt : prom_nan();
      }
    } else if (cnt >= 2 && (mode == PM_DERIV || mode == PM_PREDICT)) {
      // least-squares slope/intercept, x relative to window end (Prometheus
      // uses intercept at `te`)
      double sx = 0, sy = 0, sxx = 0, sxy = 0;
      for (int64_t j = w_lo; j < w_hi; j++) {
        const double x = (double)(ts[j] - te) / 1000.0;
        const double y = vals[j];
        sx += x; sy += y; sxx += x * x; sxy += x * y;
      }
      const double nn = (double)cnt;
      const double den = nn * sxx - sx * sx;
      if (den != 0) {
        const double slope = (nn * sxy - sx * sy) / den;
        const double intercept = (sy - slope * sx) / nn;
        r = (mode == PM_DERIV) ? slope : intercept + slope * param;
      }
    } else if (cnt >= 1 && (mode == PM_RESETS || mode == PM_CHANGES)) {
      double prev = vals[w_lo];
      int64_t k = 0;
      for (int64_t j = w_lo + 1; j < w_hi; j++) {
        const double v = vals[j];
        if (mode == PM_RESETS ? (v < prev) : (v != prev)) k++;
        prev = v;
      }
      r = (double)k;
    } else if (cnt >= 1 && mode == PM_QUANTILE_OT) {
      // Prometheus quantile: sorted linear interpolation (promql/quantile.go);
      // q outside [0,1] yields ∓Inf. Small windows (the common case: ≤128
      // samples) insertion-sort in scratch; larger windows select the two
      // needed order statistics by O(W²) rank counting — no extra storage.
      const double q = param;
      if (q < 0.0) {
        r = -INFINITY;
      } else if (q > 1.0) {
        r = INFINITY;
      } else {
        const double rank = q * (double)(cnt - 1);
        const int64_t k1 = (int64_t)rank;
        const int64_t k2 = (k1 + 1 < cnt) ? k1 + 1 : k1;
        double vk1 = NAN, vk2 = NAN;
        if (cnt <= 128) {
          double buf[128];
          int64_t m = 0;
          for (int64_t j = w_lo; j < w_hi; j++) {
            const double v = vals[j];
            int64_t p = m++;
            while (p > 0 && buf[p - 1] > v) { buf[p] = buf[p - 1]; p--; }
            buf[p] = v;
          }
          vk1 = buf[k1];
          vk2 = buf[k2];
        } else {
          // radix bisection on the sort-key transform: O(64·W) instead of
          // the old O(W²) rank counting. Register-only on purpose — each
          // thread owns its WHOLE ragged window (windows ≫ threads per
          // launch), so cooperative LDS staging would idle the rest of the
          // wavefront; per-thread bitwise selection is the CDNA4-shaped
          // answer for this layout.
          uint64_t prefix = 0;
          int64_t k = k1;
          for (int bit = 63; bit >= 0; bit--) {
            const uint64_t high_mask =
                (bit == 63) ? 0ULL : (~0ULL << (bit + 1));
            const uint64_t b = 1ULL << bit;
            int64_t cnt0 = 0;
            for (int64_t j = w_lo; j < w_hi; j++) {
              const uint64_t kj = f64_sort_key(vals[j]);
              cnt0 += ((kj & high_mask) == prefix) & !(kj & b);
            }
            if (k >= cnt0) { k -= cnt0; prefix |= b; }
          }
          const uint64_t kk = prefix;   // exact key of rank k1
          int64_t less = 0, eq = 0;
          uint64_t next = ~0ULL;
          bool has_next = false;
          for (int64_t j = w_lo; j < w_hi; j++) {
            const uint64_t kj = f64_sort_key(vals[j]);
            if (kj < kk) less++;
            else if (kj == kk) eq++;
            else if (kj < next) { next = kj; has_next = true; }
          }
          vk1 = sort_key_f64(kk);
          vk2 = (k2 < less + eq || !has_next) ? vk1 : sort_key_f64(next);
        }
        r = vk1 + (vk2 - vk1) * (rank - (double)k1);
      }
    }
    out[i] = r;
  }
}

// ------------------------------------------------- K16 scatter append
/// One launch appends a routed wire batch into MULTIPLE region memtables:
// row i goes to region region_of[i] at row dst_off[i]. Replaces per-region
// narrow+copy_ chains (two dozen small torch dispatches per batch) with a
// single kernel — the ingest hot path's device side.
__global__ void scatter_append_kernel(
    const int64_t* __restrict__ ts,
    const int32_t* __restrict__ series,
    const double* __restrict__ fields,        // [nf, n] row-major
    const int32_t* __restrict__ region_of,
    const int64_t* __restrict__ dst_off,
    int64_t* const* __restrict__ ts_ptrs,     // [R]
    int32_t* const* __restrict__ se_ptrs,
    double* const* __restrict__ f_ptrs,
    const int64_t* __restrict__ strides,      // [R] field stride (cap)
    int nf, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const int r = region_of[i];
    const int64_t o = dst_off[i];
    ts_ptrs[r][o] = ts[i];
    se_ptrs[r][o] = series[i];
    double* __restrict__ fp = f_ptrs[r];
    const int64_t st = strides[r];
    for (int f = 0; f < nf; f++) {
      fp[(int64_t)f * st + o] = fields[(int64_t)f * n + i];
    }
  }
}

// ---------------------------------------------------------------- launchers

static inline int grid_for(int64_t n, int block) {
  int64_t g = (n + block - 1) / block;
  if (g > 2048) g = 2048;  // G11: cap + grid-stride
  if (g < 1) g = 1;
  return (int)g;
}

void launch_ts_bucket_agg(
    const int64_t* ts, const int32_t* series, const double* fields,
    int64_t field_stride, const int32_t* field_idx, int nf,
    const int32_t* slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int n_slots, int n_buckets, int64_t n,
    double* out_sum, unsigned long long* out_cnt,
    unsigned long long* out_min, unsigned long long* out_max,
    unsigned long long* out_rows,
    hipStream_t stream) {
  hipLaunchKernelGGL(ts_bucket_agg_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, fields, field_stride, field_idx, nf, slot_lut, lut_size,
      ts_lo, ts_hi, origin, bucket_ms, n_slots, n_buckets, n,
      out_sum, out_cnt, out_min, out_max, out_rows);
}

void launch_decode_minmax(
    const unsigned long long* minkey, const unsigned long long* maxkey,
    const unsigned long long* cnt, double* out_min, double* out_max,
    int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(decode_minmax_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      minkey, maxkey, cnt, out_min, out_max, n);
}

void launch_filter_series_time(
    const int64_t* ts, const int32_t* series, const int32_t* slot_lut,
    int lut_size, int64_t ts_lo, int64_t ts_hi, int64_t n, bool* keep,
    hipStream_t stream) {
  hipLaunchKernelGGL(filter_series_time_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, slot_lut, lut_size, ts_lo, ts_hi, n, keep);
}

void launch_dedup_mark_last(
    const int32_t* series, const int64_t* ts, int64_t n, bool* keep,
    hipStream_t stream) {
  hipLaunchKernelGGL(dedup_mark_last_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      series, ts, n, keep);
}

void launch_bucket_agg2(
    const int64_t* ts, const int32_t* series, const double* fields,
    int64_t field_stride, const int32_t* field_idx, int nf,
    const int32_t* slot_lut, int lut_size,
    int64_t ts_lo, int64_t ts_hi, int64_t origin, int64_t bucket_ms,
    int n_slots, int n_buckets, int64_t n,
    int32_t* cell_buf,
    double* out_sum, unsigned long long* out_cnt,
    unsigned long long* out_min, unsigned long long* out_max,
    unsigned long long* out_rows, hipStream_t stream) {
  // v3 (PMC-guided): coalesced cell pass + LDS-privatized tile aggregation
  hipLaunchKernelGGL(bucket_cell_coalesced_kernel,
      dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, slot_lut, lut_size, ts_lo, ts_hi, origin, bucket_ms,
      n_buckets, n, cell_buf);
  {
    const int64_t ntiles = (n + AGG_TILE - 1) / AGG_TILE;
    const int grid = (int)min(ntiles, (int64_t)8192);
    hipLaunchKernelGGL(field_bucket_agg_lds_kernel,
        dim3(grid), dim3(256), 0, stream,
        cell_buf, fields, field_stride, field_idx, nf, n,
        (int64_t)n_slots * n_buckets,
        out_sum, out_cnt, out_min, out_max, out_rows);
  }
}

void launch_scatter_append(
    const int64_t* ts, const int32_t* series, const double* fields,
    const int32_t* region_of, const int64_t* dst_off,
    int64_t* const* ts_ptrs, int32_t* const* se_ptrs, double* const* f_ptrs,
    const int64_t* strides, int nf, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(scatter_append_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, fields, region_of, dst_off, ts_ptrs, se_ptrs, f_ptrs,
      strides, nf, n);
}

void launch_series_last(
    const int64_t* ts, const int32_t* series, const int32_t* slot_lut,
    int lut_size, int64_t ts_lo, int64_t ts_hi, int64_t n,
    unsigned long long* best_key, hipStream_t stream) {
  hipLaunchKernelGGL(series_last_ts_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, slot_lut, lut_size, ts_lo, ts_hi, n, best_key);
}

void launch_prom_range_eval(
    const int64_t* ts, const double* vals, const int64_t* seg_lo,
    const int64_t* seg_hi, int S, int T, int64_t t0, int64_t step_ms,
    int64_t range_ms, int64_t offset_ms, double param, int mode, double* out,
    hipStream_t stream) {
  const int64_t total = (int64_t)S * T;
  hipLaunchKernelGGL(prom_range_eval_kernel, dim3(grid_for(total, 256)), dim3(256), 0, stream,
      ts, vals, seg_lo, seg_hi, S, T, t0, step_ms, range_ms, offset_ms,
      param, mode, out);
}

void launch_series_last_row(
    const int64_t* ts, const int32_t* series, const int32_t* slot_lut,
    int lut_size, int64_t ts_lo, int64_t ts_hi, int64_t n,
    const unsigned long long* best_key, unsigned long long src_tag,
    unsigned long long* best_row, hipStream_t stream) {
  hipLaunchKernelGGL(series_last_row_kernel, dim3(grid_for(n, 256)), dim3(256), 0, stream,
      ts, series, slot_lut, lut_size, ts_lo, ts_hi, n, best_key, src_tag, best_row);
}



// ------------------------------------------------- K11 RLE/bit-packed expand
// Parquet hybrid RLE/bit-packed dictionary indices → i32, fully parallel:
// one thread per OUTPUT element; the run covering the element is found by
// binary search over the host-built run table (csrc/pagedec.cpp), then
// either the RLE constant is written or `bw` bits are extracted at the
// element's bit offset. Branchy page parsing stays on the host; the
// bandwidth-bound expansion runs at HBM speed here.
__global__ void rle_expand_indices_kernel(
    const int64_t* __restrict__ runs,   // [R,5] is_packed, off, val, out_start, count
    int64_t R,
    const uint8_t* __restrict__ blob,
    int bw,
    int32_t* __restrict__ out, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    // binary search: last run with out_start <= i
    int64_t lo = 0, hi = R - 1;
    while (lo < hi) {
      int64_t mid = (lo + hi + 1) >> 1;
      if (runs[mid * 5 + 3] <= i) lo = mid; else hi = mid - 1;
    }
    const int64_t* run = runs + lo * 5;
    if (run[0] == 0) {
      out[i] = (int32_t)run[2];
    } else {
      const int64_t rel = i - run[3];
      const int64_t bit = rel * bw;
      const uint8_t* p = blob + run[1] + (bit >> 3);
      // assemble up to 40 bits (bw <= 32) from unaligned bytes
      uint64_t v = (uint64_t)p[0] | ((uint64_t)p[1] << 8) |
                   ((uint64_t)p[2] << 16) | ((uint64_t)p[3] << 24) |
                   ((uint64_t)p[4] << 32);
      out[i] = (int32_t)((v >> (bit & 7)) & ((bw >= 32) ? 0xFFFFFFFFULL
                                                        : ((1ULL << bw) - 1)));
    }
  }
}

// ------------------------------------------------- K20 Gorilla / delta-delta
// Block format (engine/gorilla.py packs on host at spill time):
//   per block of BLK values: header {int64 first_ts, int64 first_delta,
//   u64 first_val_bits, int32 nbits_ts, int32 nbits_val...} — here the
//   simple fixed-width variant: delta-of-delta stored with per-block bit
//   width, XOR'd value bits with per-block width (CDNA4-friendly: fixed
//   widths per block → branchless bit extraction, one thread per value
//   with a per-block prefix scan done at pack time so decode is O(1)).
// Layout per block in `blob`:
//   [i64 base_ts][i64 base_dod_acc? not needed][u64 base_val]
//   [u8 ts_bits][u8 val_bits][u16 count][pad to 8B]
//   [packed ts dods (zigzag, ts_bits each)][packed val xors (val_bits each)]
// ts[i] = base_ts + prefix_sum(deltas); to keep decode parallel the packer
// stores ABSOLUTE deltas from base (ts[i] - base_ts) instead of
// delta-of-delta when that fits the same width; flag in high bit of ts_bits.
__global__ void gorilla_decode_kernel(
    const uint8_t* __restrict__ blob,
    const int64_t* __restrict__ block_off,   // [B] byte offset of each block
    const int64_t* __restrict__ out_off,     // [B] first output index
    int64_t B,
    int64_t* __restrict__ out_ts,
    double* __restrict__ out_val,
    int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t total_threads = stride;
  (void)total_threads;
  for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;; g += stride) {
    // map thread → (block, element) via flat index over n
    if (g >= n) return;
    // find block by binary search on out_off
    int64_t lo = 0, hi = B - 1;
    while (lo < hi) {
      int64_t mid = (lo + hi + 1) >> 1;
      if (out_off[mid] <= g) lo = mid; else hi = mid - 1;
    }
    const uint8_t* blk = blob + block_off[lo];
    int64_t base_ts;
    uint64_t base_val;
    memcpy(&base_ts, blk, 8);
    memcpy(&base_val, blk + 8, 8);
    const int ts_bits = blk[16];
    const int val_bits = blk[17];
    uint16_t count;
    memcpy(&count, blk + 18, 2);
    const int val_mode = blk[20];   // 0 = XOR f64, 1 = scaled int delta
    const int scale_k = blk[21];
    double scale = 1.0;                 // 10^k exact in f64 for k <= 4
    for (int q = 0; q < scale_k; q++) scale *= 10.0;
    const uint8_t* ts_data = blk + 24;
    const int64_t rel = g - out_off[lo];
    if (rel >= count) continue;
    if (rel == 0) {
      out_ts[g] = base_ts;
      out_val[g] = (val_mode == 1)
          ? (double)(int64_t)base_val / scale
          : __longlong_as_double((long long)base_val);
      continue;
    }
    const int64_t k = rel - 1;   // packed arrays hold values 1..count-1
    // absolute zigzagged delta from base, ts_bits each
    uint64_t tsv = 0;
    if (ts_bits) {
      const int64_t bit = k * ts_bits;
      const uint8_t* p = ts_data + (bit >> 3);
      const int sft = (int)(bit & 7);
      uint64_t w0 = 0;
      for (int b = 0; b < 8; b++) w0 |= (uint64_t)p[b] << (8 * b);
      uint64_t v = w0 >> sft;
      if (sft) v |= (uint64_t)p[8] << (64 - sft);   // spillover bits 64..70
      tsv = v & ((ts_bits >= 64) ? ~0ULL : ((1ULL << ts_bits) - 1));
    }
    const int64_t dz = (int64_t)(tsv >> 1) ^ -(int64_t)(tsv & 1);
    out_ts[g] = base_ts + dz;
    const int64_t ts_bytes = ((int64_t)(count - 1) * ts_bits + 7) / 8;
    const uint8_t* val_data = ts_data + ((ts_bytes + 7) / 8) * 8;
    uint64_t xv = 0;
    if (val_bits) {
      const int64_t bit = k * val_bits;
      const uint8_t* p = val_data + (bit >> 3);
      const int sft = (int)(bit & 7);
      uint64_t w0 = 0;
      for (int b = 0; b < 8; b++) w0 |= (uint64_t)p[b] << (8 * b);
      uint64_t v = w0 >> sft;
      if (sft) v |= (uint64_t)p[8] << (64 - sft);
      xv = v & ((val_bits >= 64) ? ~0ULL : ((1ULL << val_bits) - 1));
    }
    if (val_mode == 1) {
      const int64_t dvz = (int64_t)(xv >> 1) ^ -(int64_t)(xv & 1);
      out_val[g] = (double)((int64_t)base_val + dvz) / scale;
    } else {
      out_val[g] = __longlong_as_double((long long)(base_val ^ xv));
    }
  }
}

void launch_rle_expand_indices(
    const int64_t* runs, int64_t R, const uint8_t* blob, int bw,
    int32_t* out, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(rle_expand_indices_kernel, dim3(grid_for(n, 256)),
                     dim3(256), 0, stream, runs, R, blob, bw, out, n);
}

void launch_gorilla_decode(
    const uint8_t* blob, const int64_t* block_off, const int64_t* out_off,
    int64_t B, int64_t* out_ts, double* out_val, int64_t n,
    hipStream_t stream) {
  hipLaunchKernelGGL(gorilla_decode_kernel, dim3(grid_for(n, 256)),
                     dim3(256), 0, stream, blob, block_off, out_off, B,
                     out_ts, out_val, n);
}

}  // namespace gdb_hip
