// greptimedb_amd._native — host-side native ingest path.
//
// Two components, both on the ingest hot path:
//  1. LineParser — influx line protocol → columnar batch. Replaces the
//     reference's per-protocol row decoding (src/servers/src/influxdb.rs +
//     row protos). Tagsets are dict-encoded to dense int32 series refs with
//     an interned-string map so steady-state ingest never re-parses tags.
//  2. WalWriter — segmented group-commit WAL writer (reference:
//     src/log-store raft_engine backend + mito2 wal.rs:187 WalWriter).
//     Frame: [u32 len][u32 crc32][u64 region_id][u64 seq][payload].
//
// Deliberately torch-free: compiles in seconds, usable from any process.

#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <pybind11/stl.h>

#include <cerrno>
#include <cmath>
#include <cstdint>
#include <cstring>
#include <fcntl.h>
#include <stdexcept>
#include <string>
#include <unistd.h>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

// ------------------------------------- crc32 (IEEE, zlib-compatible, slice-by-8)
static uint32_t crc_table[8][256];
static bool crc_init_done = [] {
  for (uint32_t i = 0; i < 256; i++) {
    uint32_t c = i;
    for (int k = 0; k < 8; k++) c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
    crc_table[0][i] = c;
  }
  for (int t = 1; t < 8; t++)
    for (uint32_t i = 0; i < 256; i++)
      crc_table[t][i] = crc_table[0][crc_table[t - 1][i] & 0xFF] ^
                        (crc_table[t - 1][i] >> 8);
  return true;
}();

static uint32_t crc32_update(uint32_t crc, const uint8_t* buf, size_t len) {
  crc = ~crc;
  while (len >= 8) {
    uint32_t lo, hi;
    std::memcpy(&lo, buf, 4);
    std::memcpy(&hi, buf + 4, 4);
    lo ^= crc;
    crc = crc_table[7][lo & 0xFF] ^ crc_table[6][(lo >> 8) & 0xFF] ^
          crc_table[5][(lo >> 16) & 0xFF] ^ crc_table[4][lo >> 24] ^
          crc_table[3][hi & 0xFF] ^ crc_table[2][(hi >> 8) & 0xFF] ^
          crc_table[1][(hi >> 16) & 0xFF] ^ crc_table[0][hi >> 24];
    buf += 8;
    len -= 8;
  }
  for (size_t i = 0; i < len; i++)
    crc = crc_table[0][(crc ^ buf[i]) & 0xFF] ^ (crc >> 8);
  return ~crc;
}

// ---------------------------------------------------------------- LineParser

struct ParseResult {
  std::vector<int32_t> series;
  std::vector<int64_t> ts;
  // field columns, lazily NaN-padded
  std::vector<std::vector<double>> cols;
  std::vector<std::pair<int32_t, std::string>> new_tagsets;
};

// FNV-1a 64-bit over a byte range (tagset / field-name interning probe).
static inline uint64_t fnv1a(const char* p, size_t n) {
  uint64_t h = 1469598103934665603ULL;
  for (size_t i = 0; i < n; i++) { h ^= (uint8_t)p[i]; h *= 1099511628211ULL; }
  return h;
}

// Fast decimal parse for the common "[-]digits[.digits]" shape; falls back to
// strtod for exponents/inf/nan.
static inline double fast_atof(const char* p, const char* end) {
  bool neg = false;
  const char* s = p;
  if (s < end && (*s == '-' || *s == '+')) { neg = (*s == '-'); s++; }
  uint64_t ip = 0; int nd = 0;
  while (s < end && *s >= '0' && *s <= '9' && nd < 18) { ip = ip * 10 + (*s - '0'); s++; nd++; }
  double v = (double)ip;
  if (s < end && *s == '.') {
    s++;
    uint64_t fp = 0; int fd = 0;
    while (s < end && *s >= '0' && *s <= '9' && fd < 18) { fp = fp * 10 + (*s - '0'); s++; fd++; }
    static const double pow10[19] = {1e0,1e1,1e2,1e3,1e4,1e5,1e6,1e7,1e8,1e9,1e10,
                                     1e11,1e12,1e13,1e14,1e15,1e16,1e17,1e18};
    v += (double)fp / pow10[fd];
  }
  if (s < end && (*s == 'e' || *s == 'E')) return strtod(p, nullptr);  // rare
  return neg ? -v : v;
}

static inline int64_t fast_atoll(const char* p, const char* end) {
  bool neg = false;
  if (p < end && (*p == '-' || *p == '+')) { neg = (*p == '-'); p++; }
  int64_t v = 0;
  while (p < end && *p >= '0' && *p <= '9') { v = v * 10 + (*p - '0'); p++; }
  return neg ? -v : v;
}

class LineParser {
 public:
  // Parse a batch of lines. Returns (series i32[n], ts i64[n],
  // {field: f64[n]}, [(id, tagset_bytes)...new]).
  py::tuple parse(py::bytes data) {
    char* buf;
    Py_ssize_t len;
    if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
      throw std::runtime_error("expected bytes");
    ParseResult r;
    r.cols.resize(field_names_.size());
    size_t nrows = 0;
    {
      py::gil_scoped_release rel;  // pure C scan; enables threaded ingest workers
      // estimate rows for reserve
      size_t est = 0;
      for (const char* q = buf; (q = static_cast<const char*>(
               memchr(q, '\n', buf + len - q))) != nullptr; q++) est++;
      est += 1;
      r.series.reserve(est);
      r.ts.reserve(est);
      for (auto& c : r.cols) c.reserve(est);

      const char* p = buf;
      const char* end = buf + len;
      while (p < end) {
        const char* nl = static_cast<const char*>(memchr(p, '\n', end - p));
        const char* line_end = nl ? nl : end;
        if (line_end > p && *p != '#') {
          parse_line(p, line_end, r, nrows);
        }
        p = nl ? nl + 1 : end;
      }
    }

    // materialize numpy outputs
    py::array_t<int32_t> series(nrows);
    py::array_t<int64_t> ts(nrows);
    std::memcpy(series.mutable_data(), r.series.data(), nrows * 4);
    std::memcpy(ts.mutable_data(), r.ts.data(), nrows * 8);
    py::dict fields;
    for (size_t c = 0; c < field_names_.size(); c++) {
      auto& col = r.cols[c];
      col.resize(nrows, std::nan(""));
      py::array_t<double> a(nrows);
      std::memcpy(a.mutable_data(), col.data(), nrows * 8);
      fields[py::str(field_names_[c])] = std::move(a);
    }
    py::list newts;
    for (auto& [id, s] : r.new_tagsets)
      newts.append(py::make_tuple(id, py::bytes(s)));
    return py::make_tuple(std::move(series), std::move(ts), std::move(fields), std::move(newts));
  }

  size_t num_series() const { return tagset_ids_.size(); }
  std::vector<std::string> field_names() const { return field_names_; }

  py::bytes tagset_str(int32_t sid) const {
    if (sid < 0 || (size_t)sid >= tagset_strs_.size())
      throw std::out_of_range("unknown series ref");
    return py::bytes(tagset_strs_[sid]);
  }

  // Restore dictionary state (e.g. after WAL replay / reopen).
  void register_tagset(const std::string& s, int32_t id) {
    tagset_ids_.emplace(fnv1a(s.data(), s.size()), id);
    tagset_strs_.resize(std::max<size_t>(tagset_strs_.size(), id + 1));
    tagset_strs_[id] = s;
    next_id_ = std::max(next_id_, id + 1);
  }
  void register_field(const std::string& name) { field_idx(name.data(), name.size()); }

 private:
  void parse_line(const char* p, const char* end, ParseResult& r, size_t& nrows) {
    // measurement,tagset fieldset [timestamp]
    // NOTE: influx escape sequences (\,, \ , \=) are not handled — TSBS and
    // our writers never emit them; reject lines containing a backslash.
    const char* sp1 = static_cast<const char*>(memchr(p, ' ', end - p));
    if (!sp1) return;  // malformed
    const char* sp2 = static_cast<const char*>(memchr(sp1 + 1, ' ', end - sp1 - 1));

    // series key = "measurement,tagset" — interned via hash probe + byte
    // verify (no per-row allocation on the hit path)
    const size_t klen = sp1 - p;
    const uint64_t h = fnv1a(p, klen);
    int32_t sid = -1;
    auto range = tagset_ids_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it) {
      const std::string& s = tagset_strs_[it->second];
      if (s.size() == klen && memcmp(s.data(), p, klen) == 0) { sid = it->second; break; }
    }
    if (sid < 0) {
      sid = next_id_++;
      tagset_ids_.emplace(h, sid);
      tagset_strs_.resize(std::max<size_t>(tagset_strs_.size(), sid + 1));
      tagset_strs_[sid].assign(p, klen);
      r.new_tagsets.emplace_back(sid, tagset_strs_[sid]);
    }

    // timestamp (ns); missing timestamp → 0 (caller fills server time)
    int64_t tsv = 0;
    if (sp2) tsv = fast_atoll(sp2 + 1, end);

    size_t row = nrows++;
    r.series.push_back(sid);
    r.ts.push_back(tsv);

    // fieldset: k=v,k=v,...
    const char* f = sp1 + 1;
    const char* fend = sp2 ? sp2 : end;
    size_t fpos = 0;
    while (f < fend) {
      const char* eq = static_cast<const char*>(memchr(f, '=', fend - f));
      if (!eq) break;
      const char* comma = static_cast<const char*>(memchr(eq + 1, ',', fend - eq - 1));
      const char* vend = comma ? comma : fend;
      double val;
      const char* v = eq + 1;
      if (v < vend && (*v == '"')) {
        // string field value — not representable in the float column; store NaN.
        val = std::nan("");
      } else if (vend > v && (vend[-1] == 'i' || vend[-1] == 'u')) {
        val = static_cast<double>(fast_atoll(v, vend - 1));
      } else if (vend > v && (*v == 't' || *v == 'T' || *v == 'f' || *v == 'F')) {
        val = (*v == 't' || *v == 'T') ? 1.0 : 0.0;
      } else {
        val = fast_atof(v, vend);
      }
      // positional field cache: batches of lines share a fieldset layout
      // (TSBS and most collectors), so field j of each row is usually the
      // same name — memcmp against the cached name skips hash+map
      size_t c;
      const size_t flen = eq - f;
      if (fpos < fcache_.size() && fcache_[fpos].first == flen &&
          memcmp(field_names_[fcache_[fpos].second].data(), f, flen) == 0) {
        c = fcache_[fpos].second;
      } else {
        c = field_idx(f, flen);
        if (fpos >= fcache_.size()) fcache_.resize(fpos + 1);
        fcache_[fpos] = {flen, c};
      }
      fpos++;
      if (c >= r.cols.size()) r.cols.resize(c + 1);
      auto& col = r.cols[c];
      if (col.size() < row) col.resize(row, std::nan(""));
      if (col.size() == row) col.push_back(val); else col[row] = val;
      f = comma ? comma + 1 : fend;
    }
  }

  size_t field_idx(const char* name, size_t len) {
    const uint64_t h = fnv1a(name, len);
    auto range = fmap_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it) {
      const std::string& s = field_names_[it->second];
      if (s.size() == len && memcmp(s.data(), name, len) == 0) return it->second;
    }
    size_t idx = field_names_.size();
    fmap_.emplace(h, idx);
    field_names_.emplace_back(name, len);
    return idx;
  }

  std::unordered_multimap<uint64_t, int32_t> tagset_ids_;
  std::vector<std::string> tagset_strs_;
  int32_t next_id_ = 0;
  std::unordered_multimap<uint64_t, size_t> fmap_;
  std::vector<std::string> field_names_;
  std::vector<std::pair<size_t, size_t>> fcache_;  // positional (len, idx)
};

// ---------------------------------------------------------------- WalWriter

class WalWriter {
 public:
  WalWriter() = default;
  ~WalWriter() { close_segment(); }

  void open_segment(const std::string& path) {
    close_segment();
    fd_ = ::open(path.c_str(), O_CREAT | O_WRONLY | O_APPEND, 0644);
    if (fd_ < 0) throw std::runtime_error("wal open failed: " + path + ": " + strerror(errno));
    path_ = path;
    seg_bytes_ = ::lseek(fd_, 0, SEEK_END);
  }

  void close_segment() {
    if (fd_ >= 0) { ::close(fd_); fd_ = -1; }
  }

  // Buffer one entry (group commit happens in commit()). The crc over the
  // payload runs without the GIL — it is the ingest path's biggest
  // GIL-held C++ cost otherwise (callers serialize via the python Wal lock).
  void append(uint64_t region_id, uint64_t seq, py::bytes payload) {
    char* pbuf; Py_ssize_t plen;
    if (PyBytes_AsStringAndSize(payload.ptr(), &pbuf, &plen) != 0)
      throw std::runtime_error("payload must be bytes");
    py::gil_scoped_release rel;
    uint32_t body_len = 16 + static_cast<uint32_t>(plen);
    size_t off = buf_.size();
    buf_.resize(off + 8 + body_len);
    uint8_t* w = buf_.data() + off;
    std::memcpy(w, &body_len, 4);
    // crc over [region][seq][payload]
    uint8_t hdr[16];
    std::memcpy(hdr, &region_id, 8);
    std::memcpy(hdr + 8, &seq, 8);
    uint32_t crc = crc32_update(0, hdr, 16);
    crc = crc32_update(crc, reinterpret_cast<uint8_t*>(pbuf), plen);
    std::memcpy(w + 4, &crc, 4);
    std::memcpy(w + 8, hdr, 16);
    std::memcpy(w + 24, pbuf, plen);
  }

  // Flush buffered entries; optionally fdatasync. Returns segment size.
  uint64_t commit(bool sync) {
    if (fd_ < 0) throw std::runtime_error("wal: no open segment");
    size_t n = buf_.size();
    if (n) {
      py::gil_scoped_release rel;
      const uint8_t* p = buf_.data();
      size_t left = n;
      while (left) {
        ssize_t w = ::write(fd_, p, left);
        if (w < 0) {
          if (errno == EINTR) continue;
          throw std::runtime_error(std::string("wal write failed: ") + strerror(errno));
        }
        p += w; left -= w;
      }
      if (sync && ::fdatasync(fd_) != 0)
        throw std::runtime_error(std::string("wal fdatasync failed: ") + strerror(errno));
    }
    buf_.clear();
    seg_bytes_ += n;
    return seg_bytes_;
  }

  uint64_t segment_bytes() const { return seg_bytes_; }

 private:
  int fd_ = -1;
  std::string path_;
  uint64_t seg_bytes_ = 0;
  std::vector<uint8_t> buf_;
};

// Read back one WAL segment: returns list of (region_id, seq, payload bytes).
// Stops at the first corrupt/truncated frame (torn tail after crash).
static py::list wal_read_segment(const std::string& path) {
  py::list out;
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0) throw std::runtime_error("wal open failed: " + path);
  off_t sz = ::lseek(fd, 0, SEEK_END);
  ::lseek(fd, 0, SEEK_SET);
  std::vector<uint8_t> data(sz);
  size_t got = 0;
  while (got < static_cast<size_t>(sz)) {
    ssize_t r = ::read(fd, data.data() + got, sz - got);
    if (r <= 0) break;
    got += r;
  }
  ::close(fd);
  size_t off = 0;
  while (off + 8 <= got) {
    uint32_t body_len, crc;
    std::memcpy(&body_len, data.data() + off, 4);
    std::memcpy(&crc, data.data() + off + 4, 4);
    if (body_len < 16 || off + 8 + body_len > got) break;
    const uint8_t* body = data.data() + off + 8;
    if (crc32_update(0, body, body_len) != crc) break;
    uint64_t region, seq;
    std::memcpy(&region, body, 8);
    std::memcpy(&seq, body + 8, 8);
    out.append(py::make_tuple(region, seq,
        py::bytes(reinterpret_cast<const char*>(body + 16), body_len - 16)));
    off += 8 + body_len;
  }
  return out;
}

// ---------------------------------------------------------------- snappy
// Snappy block-format decompressor (format: google/snappy format_description.txt).
// Used for Prometheus remote write request bodies.

static bool snappy_uncompress(const uint8_t* in, size_t n, std::vector<uint8_t>& out) {
  size_t ip = 0;
  // preamble: uncompressed length varint
  uint64_t ulen = 0;
  int shift = 0;
  while (ip < n) {
    uint8_t b = in[ip++];
    ulen |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
    if (shift > 35) return false;
  }
  out.clear();
  out.reserve(ulen);
  while (ip < n) {
    const uint8_t tag = in[ip++];
    const int type = tag & 3;
    if (type == 0) {  // literal
      size_t len = (tag >> 2) + 1;
      if (len > 60) {
        const int extra = (int)len - 60;
        if (ip + extra > n) return false;
        len = 0;
        for (int i = 0; i < extra; i++) len |= (size_t)in[ip + i] << (8 * i);
        len += 1;
        ip += extra;
      }
      if (ip + len > n) return false;
      out.insert(out.end(), in + ip, in + ip + len);
      ip += len;
    } else {
      size_t len, off;
      if (type == 1) {
        if (ip >= n) return false;
        len = ((tag >> 2) & 7) + 4;
        off = ((size_t)(tag >> 5) << 8) | in[ip++];
      } else if (type == 2) {
        if (ip + 2 > n) return false;
        len = (tag >> 2) + 1;
        off = in[ip] | ((size_t)in[ip + 1] << 8);
        ip += 2;
      } else {
        if (ip + 4 > n) return false;
        len = (tag >> 2) + 1;
        off = in[ip] | ((size_t)in[ip + 1] << 8) |
              ((size_t)in[ip + 2] << 16) | ((size_t)in[ip + 3] << 24);
        ip += 4;
      }
      if (off == 0 || off > out.size()) return false;
      size_t src = out.size() - off;
      for (size_t i = 0; i < len; i++) out.push_back(out[src + i]);  // may overlap
    }
  }
  return out.size() == ulen;
}

static py::bytes py_snappy_uncompress(py::bytes data) {
  char* buf; Py_ssize_t len;
  if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
    throw std::runtime_error("expected bytes");
  std::vector<uint8_t> out;
  if (!snappy_uncompress(reinterpret_cast<const uint8_t*>(buf), len, out))
    throw std::runtime_error("snappy: corrupt input");
  return py::bytes(reinterpret_cast<const char*>(out.data()), out.size());
}

// ------------------------------------------------- prometheus remote write
// Minimal protobuf wire parser for prometheus.WriteRequest:
//   WriteRequest{ repeated TimeSeries timeseries = 1 }
//   TimeSeries{ repeated Label labels = 1; repeated Sample samples = 2 }
//   Label{ string name = 1; string value = 2 }
//   Sample{ double value = 1; int64 timestamp = 2 }
// Label sets are interned (prometheus senders emit sorted labels) to dense
// series refs like the influx LineParser.

struct PwSlice { const uint8_t* p; size_t n; };

static inline uint64_t pw_varint(const uint8_t*& p, const uint8_t* end) {
  uint64_t v = 0;
  int shift = 0;
  while (p < end) {
    uint8_t b = *p++;
    v |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) return v;
    shift += 7;
  }
  return v;
}

class PromWriteParser {
 public:
  // Returns (series i32[n], ts i64[n] ms, value f64[n],
  //          new_series [(id, metric, [(name,value)...])])
  py::tuple parse(py::bytes compressed, bool is_snappy) {
    char* buf; Py_ssize_t len;
    if (PyBytes_AsStringAndSize(compressed.ptr(), &buf, &len) != 0)
      throw std::runtime_error("expected bytes");
    std::vector<uint8_t> plain;
    const uint8_t* data;
    size_t n;
    if (is_snappy) {
      if (!snappy_uncompress(reinterpret_cast<const uint8_t*>(buf), len, plain))
        throw std::runtime_error("snappy: corrupt input");
      data = plain.data(); n = plain.size();
    } else {
      data = reinterpret_cast<const uint8_t*>(buf); n = len;
    }
    std::vector<int32_t> series;
    std::vector<int64_t> ts;
    std::vector<double> vals;
    py::list new_series;
    {
      const uint8_t* p = data;
      const uint8_t* end = data + n;
      while (p < end) {
        uint64_t key = pw_varint(p, end);
        if ((key >> 3) == 1 && (key & 7) == 2) {
          uint64_t tlen = pw_varint(p, end);
          parse_timeseries(p, p + tlen, series, ts, vals, new_series);
          p += tlen;
        } else {
          skip_field(key & 7, p, end);
        }
      }
    }
    py::array_t<int32_t> s(series.size());
    py::array_t<int64_t> t(ts.size());
    py::array_t<double> v(vals.size());
    std::memcpy(s.mutable_data(), series.data(), series.size() * 4);
    std::memcpy(t.mutable_data(), ts.data(), ts.size() * 8);
    std::memcpy(v.mutable_data(), vals.data(), vals.size() * 8);
    return py::make_tuple(std::move(s), std::move(t), std::move(v), std::move(new_series));
  }

  size_t num_series() const { return next_id_; }

 private:
  static void skip_field(int wt, const uint8_t*& p, const uint8_t* end) {
    if (wt == 0) { pw_varint(p, end); }
    else if (wt == 1) { p += 8; }
    else if (wt == 2) { uint64_t l = pw_varint(p, end); p += l; }
    else if (wt == 5) { p += 4; }
    else { p = end; }
  }

  void parse_timeseries(const uint8_t* p, const uint8_t* end,
                        std::vector<int32_t>& series, std::vector<int64_t>& ts,
                        std::vector<double>& vals, py::list& new_series) {
    // first pass: find label region bounds to build the intern key
    const uint8_t* q = p;
    std::string key;  // concatenated raw Label messages — canonical per sender
    std::vector<std::pair<std::string, std::string>> labels;
    std::vector<std::pair<double, int64_t>> samples;
    while (q < end) {
      uint64_t k = pw_varint(q, end);
      const int fnum = (int)(k >> 3), wt = (int)(k & 7);
      if (fnum == 1 && wt == 2) {          // Label
        uint64_t l = pw_varint(q, end);
        key.append(reinterpret_cast<const char*>(q), l);
        key.push_back('\xff');
        const uint8_t* lp = q;
        const uint8_t* lend = q + l;
        std::string lname, lval;
        while (lp < lend) {
          uint64_t lk = pw_varint(lp, lend);
          if ((lk >> 3) == 1 && (lk & 7) == 2) {
            uint64_t s = pw_varint(lp, lend);
            lname.assign(reinterpret_cast<const char*>(lp), s);
            lp += s;
          } else if ((lk >> 3) == 2 && (lk & 7) == 2) {
            uint64_t s = pw_varint(lp, lend);
            lval.assign(reinterpret_cast<const char*>(lp), s);
            lp += s;
          } else {
            skip_field(lk & 7, lp, lend);
          }
        }
        labels.emplace_back(std::move(lname), std::move(lval));
        q += l;
      } else if (fnum == 2 && wt == 2) {   // Sample
        uint64_t l = pw_varint(q, end);
        const uint8_t* sp = q;
        const uint8_t* send = q + l;
        double v = std::nan("");
        int64_t t = 0;
        while (sp < send) {
          uint64_t sk = pw_varint(sp, send);
          if ((sk >> 3) == 1 && (sk & 7) == 1) {
            std::memcpy(&v, sp, 8); sp += 8;
          } else if ((sk >> 3) == 2 && (sk & 7) == 0) {
            t = (int64_t)pw_varint(sp, send);
          } else {
            skip_field(sk & 7, sp, send);
          }
        }
        samples.emplace_back(v, t);
        q += l;
      } else {
        skip_field(wt, q, end);
      }
    }
    if (samples.empty()) return;
    const uint64_t h = fnv1a(key.data(), key.size());
    int32_t sid = -1;
    auto range = ids_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it) {
      if (keys_[it->second] == key) { sid = it->second; break; }
    }
    if (sid < 0) {
      sid = next_id_++;
      ids_.emplace(h, sid);
      keys_.resize(std::max<size_t>(keys_.size(), sid + 1));
      keys_[sid] = key;
      std::string metric;
      py::list ls;
      for (auto& [ln, lv] : labels) {
        if (ln == "__name__") metric = lv;
        else ls.append(py::make_tuple(ln, lv));
      }
      new_series.append(py::make_tuple(sid, metric, std::move(ls)));
    }
    for (auto& [v, t] : samples) {
      series.push_back(sid);
      ts.push_back(t);
      vals.push_back(v);
    }
  }

  std::unordered_multimap<uint64_t, int32_t> ids_;
  std::vector<std::string> keys_;
  int32_t next_id_ = 0;
};

// ------------------------------------------------- OTLP traces (protobuf)
// Minimal wire parser for opentelemetry ExportTraceServiceRequest
// (trace.proto): resource_spans → scope_spans → spans. (service.name,
// span name) pairs are interned to dense series refs; ids come back as hex
// strings; attributes flatten to a compact JSON string column.

static const char* HEXD = "0123456789abcdef";

static std::string to_hex(const uint8_t* p, size_t n) {
  std::string s(n * 2, '0');
  for (size_t i = 0; i < n; i++) {
    s[2 * i] = HEXD[p[i] >> 4];
    s[2 * i + 1] = HEXD[p[i] & 0xF];
  }
  return s;
}

static void json_escape_into(std::string& out, const char* p, size_t n) {
  for (size_t i = 0; i < n; i++) {
    char c = p[i];
    if (c == '"' || c == '\\') { out.push_back('\\'); out.push_back(c); }
    else if ((unsigned char)c < 0x20) { out += "\\u0020"; }
    else out.push_back(c);
  }
}

class OtlpTraceParser {
 public:
  py::tuple parse(py::bytes data) {
    char* buf; Py_ssize_t len;
    if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
      throw std::runtime_error("expected bytes");
    const uint8_t* p = reinterpret_cast<const uint8_t*>(buf);
    const uint8_t* end = p + len;
    out_series_.clear(); out_start_.clear(); out_dur_.clear();
    out_status_.clear();
    py::list trace_ids, span_ids, parent_ids, attrs, new_series;
    while (p < end) {
      uint64_t key = pw_varint(p, end);
      if ((key >> 3) == 1 && (key & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        parse_resource_spans(p, p + l, trace_ids, span_ids, parent_ids,
                             attrs, new_series);
        p += l;
      } else {
        skip(key & 7, p, end);
      }
    }
    size_t n = out_series_.size();
    py::array_t<int32_t> s(n);
    py::array_t<int64_t> st(n);
    py::array_t<double> du(n);
    py::array_t<int32_t> stc(n);
    if (n) {
      std::memcpy(s.mutable_data(), out_series_.data(), n * 4);
      std::memcpy(st.mutable_data(), out_start_.data(), n * 8);
      std::memcpy(du.mutable_data(), out_dur_.data(), n * 8);
      std::memcpy(stc.mutable_data(), out_status_.data(), n * 4);
    }
    return py::make_tuple(std::move(s), std::move(st), std::move(du),
                          std::move(stc), std::move(trace_ids),
                          std::move(span_ids), std::move(parent_ids),
                          std::move(attrs), std::move(new_series));
  }

  size_t num_series() const { return next_id_; }

 private:
  static void skip(int wt, const uint8_t*& p, const uint8_t* end) {
    if (wt == 0) pw_varint(p, end);
    else if (wt == 1) p += 8;
    else if (wt == 2) { uint64_t l = pw_varint(p, end); p += l; }
    else if (wt == 5) p += 4;
    else p = end;
  }

  // returns value of attribute "service.name" from a Resource message
  std::string parse_resource_service(const uint8_t* p, const uint8_t* end) {
    std::string service = "unknown";
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      if ((k >> 3) == 1 && (k & 7) == 2) {  // attributes: KeyValue
        uint64_t l = pw_varint(p, end);
        const uint8_t* kp = p; const uint8_t* kend = p + l;
        std::string kname, vstr;
        while (kp < kend) {
          uint64_t kk = pw_varint(kp, kend);
          if ((kk >> 3) == 1 && (kk & 7) == 2) {
            uint64_t s = pw_varint(kp, kend);
            kname.assign(reinterpret_cast<const char*>(kp), s); kp += s;
          } else if ((kk >> 3) == 2 && (kk & 7) == 2) {  // AnyValue
            uint64_t s = pw_varint(kp, kend);
            const uint8_t* vp = kp; const uint8_t* vend = kp + s;
            while (vp < vend) {
              uint64_t vk = pw_varint(vp, vend);
              if ((vk >> 3) == 1 && (vk & 7) == 2) {
                uint64_t sl = pw_varint(vp, vend);
                vstr.assign(reinterpret_cast<const char*>(vp), sl); vp += sl;
              } else skip(vk & 7, vp, vend);
            }
            kp += s;
          } else skip(kk & 7, kp, kend);
        }
        if (kname == "service.name" && !vstr.empty()) service = vstr;
        p += l;
      } else skip(k & 7, p, end);
    }
    return service;
  }

  void parse_resource_spans(const uint8_t* p, const uint8_t* end,
                            py::list& tids, py::list& sids, py::list& pids,
                            py::list& attrs, py::list& new_series) {
    std::string service = "unknown";
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 1 && wt == 2) {  // Resource
        uint64_t l = pw_varint(p, end);
        service = parse_resource_service(p, p + l);
        p += l;
      } else if (f == 2 && wt == 2) {  // ScopeSpans
        uint64_t l = pw_varint(p, end);
        const uint8_t* sp = p; const uint8_t* send = p + l;
        while (sp < send) {
          uint64_t sk = pw_varint(sp, send);
          if ((sk >> 3) == 2 && (sk & 7) == 2) {  // Span
            uint64_t sl = pw_varint(sp, send);
            parse_span(sp, sp + sl, service, tids, sids, pids, attrs, new_series);
            sp += sl;
          } else skip(sk & 7, sp, send);
        }
        p += l;
      } else skip(wt, p, end);
    }
  }

  void parse_span(const uint8_t* p, const uint8_t* end, const std::string& service,
                  py::list& tids, py::list& sids, py::list& pids,
                  py::list& attrs, py::list& new_series) {
    std::string trace_id, span_id, parent_id, name;
    uint64_t start = 0, stop = 0;
    int32_t status = 0;
    std::string attr_json = "{";
    bool first_attr = true;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 1 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        trace_id = to_hex(p, l); p += l;
      } else if (f == 2 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        span_id = to_hex(p, l); p += l;
      } else if (f == 4 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        parent_id = to_hex(p, l); p += l;
      } else if (f == 5 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        name.assign(reinterpret_cast<const char*>(p), l); p += l;
      } else if (f == 7 && wt == 1) {
        std::memcpy(&start, p, 8); p += 8;
      } else if (f == 8 && wt == 1) {
        std::memcpy(&stop, p, 8); p += 8;
      } else if (f == 9 && wt == 2) {  // attributes KeyValue
        uint64_t l = pw_varint(p, end);
        append_attr_json(p, p + l, attr_json, first_attr);
        p += l;
      } else if (f == 15 && wt == 2) {  // Status{message=2?, code=3? -> code=2? }
        uint64_t l = pw_varint(p, end);
        const uint8_t* qp = p; const uint8_t* qend = p + l;
        while (qp < qend) {
          uint64_t qk = pw_varint(qp, qend);
          if ((qk >> 3) == 2 && (qk & 7) == 0) {
            status = (int32_t)pw_varint(qp, qend);
          } else if ((qk >> 3) == 3 && (qk & 7) == 0) {
            status = (int32_t)pw_varint(qp, qend);
          } else skip(qk & 7, qp, qend);
        }
        p += l;
      } else {
        skip(wt, p, end);
      }
    }
    attr_json.push_back('}');
    // intern (service, span name)
    std::string key = service;
    key.push_back('\0');
    key += name;
    const uint64_t h = fnv1a(key.data(), key.size());
    int32_t sid = -1;
    auto range = ids_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it)
      if (keys_[it->second] == key) { sid = it->second; break; }
    if (sid < 0) {
      sid = next_id_++;
      ids_.emplace(h, sid);
      keys_.resize(std::max<size_t>(keys_.size(), sid + 1));
      keys_[sid] = key;
      new_series.append(py::make_tuple(sid, py::str(service), py::str(name)));
    }
    out_series_.push_back(sid);
    out_start_.push_back((int64_t)start);
    out_dur_.push_back(stop >= start ? (double)(stop - start) / 1e6 : 0.0);  // ms
    out_status_.push_back(status);
    tids.append(py::str(trace_id));
    sids.append(py::str(span_id));
    pids.append(py::str(parent_id));
    attrs.append(py::str(attr_json));
  }

  void append_attr_json(const uint8_t* p, const uint8_t* end, std::string& out,
                        bool& first) {
    std::string kname, vjson;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      if ((k >> 3) == 1 && (k & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        kname.assign(reinterpret_cast<const char*>(p), l); p += l;
      } else if ((k >> 3) == 2 && (k & 7) == 2) {  // AnyValue
        uint64_t l = pw_varint(p, end);
        const uint8_t* vp = p; const uint8_t* vend = p + l;
        while (vp < vend) {
          uint64_t vk = pw_varint(vp, vend);
          int vf = (int)(vk >> 3), vwt = (int)(vk & 7);
          if (vf == 1 && vwt == 2) {
            uint64_t sl = pw_varint(vp, vend);
            vjson = "\"";
            json_escape_into(vjson, reinterpret_cast<const char*>(vp), sl);
            vjson += "\"";
            vp += sl;
          } else if (vf == 2 && vwt == 0) {
            vjson = pw_varint(vp, vend) ? "true" : "false";
          } else if (vf == 3 && vwt == 0) {
            vjson = std::to_string((int64_t)pw_varint(vp, vend));
          } else if (vf == 4 && vwt == 1) {
            double d; std::memcpy(&d, vp, 8); vp += 8;
            vjson = std::to_string(d);
          } else skip(vwt, vp, vend);
        }
        p += l;
      } else skip(k & 7, p, end);
    }
    if (!kname.empty() && !vjson.empty()) {
      if (!first) out.push_back(',');
      first = false;
      out.push_back('"');
      json_escape_into(out, kname.data(), kname.size());
      out += "\":";
      out += vjson;
    }
  }

  std::unordered_multimap<uint64_t, int32_t> ids_;
  std::vector<std::string> keys_;
  int32_t next_id_ = 0;
  std::vector<int32_t> out_series_;
  std::vector<int64_t> out_start_;
  std::vector<double> out_dur_;
  std::vector<int32_t> out_status_;
};

// ------------------------------------------------- OTLP metrics + logs
// ExportMetricsServiceRequest: resource_metrics → scope_metrics → Metric
// { name=1; gauge=5 / sum=7 { data_points: NumberDataPoint } }.
// NumberDataPoint{ attributes=7; time_unix_nano=3 (fixed64);
//                  as_double=4 (fixed64) / as_int=6 (sfixed64) }.
// Output: flat [(metric, {attrs}, ts_ms, value)] ready for
// PromStore.write_points.

class OtlpMetricsParser {
 public:
  py::list parse(py::bytes data) {
    char* buf; Py_ssize_t len;
    if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
      throw std::runtime_error("expected bytes");
    py::list out;
    const uint8_t* p = reinterpret_cast<const uint8_t*>(buf);
    const uint8_t* end = p + len;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      if ((k >> 3) == 1 && (k & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        parse_resource_metrics(p, p + l, out);
        p += l;
      } else skip(k & 7, p, end);
    }
    return out;
  }

 private:
  static void skip(int wt, const uint8_t*& p, const uint8_t* end) {
    if (wt == 0) pw_varint(p, end);
    else if (wt == 1) p += 8;
    else if (wt == 2) { uint64_t l = pw_varint(p, end); p += l; }
    else if (wt == 5) p += 4;
    else p = end;
  }

  static void parse_kv(const uint8_t* p, const uint8_t* end, py::dict& attrs) {
    std::string kname, vstr;
    double vnum = 0; bool has_num = false;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      if ((k >> 3) == 1 && (k & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        kname.assign(reinterpret_cast<const char*>(p), l); p += l;
      } else if ((k >> 3) == 2 && (k & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        const uint8_t* vp = p; const uint8_t* vend = p + l;
        while (vp < vend) {
          uint64_t vk = pw_varint(vp, vend);
          int vf = (int)(vk >> 3), vwt = (int)(vk & 7);
          if (vf == 1 && vwt == 2) {
            uint64_t sl = pw_varint(vp, vend);
            vstr.assign(reinterpret_cast<const char*>(vp), sl); vp += sl;
          } else if (vf == 2 && vwt == 0) {
            vstr = pw_varint(vp, vend) ? "true" : "false";
          } else if (vf == 3 && vwt == 0) {
            vstr = std::to_string((int64_t)pw_varint(vp, vend));
          } else if (vf == 4 && vwt == 1) {
            double d; std::memcpy(&d, vp, 8); vp += 8;
            vstr = std::to_string(d);
          } else skip(vwt, vp, vend);
        }
        p += l;
      } else skip(k & 7, p, end);
    }
    if (!kname.empty()) attrs[py::str(kname)] = py::str(vstr);
  }

  void parse_resource_metrics(const uint8_t* p, const uint8_t* end, py::list& out) {
    py::dict res_attrs;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 1 && wt == 2) {  // Resource{attributes=1}
        uint64_t l = pw_varint(p, end);
        const uint8_t* rp = p; const uint8_t* rend = p + l;
        while (rp < rend) {
          uint64_t rk = pw_varint(rp, rend);
          if ((rk >> 3) == 1 && (rk & 7) == 2) {
            uint64_t al = pw_varint(rp, rend);
            parse_kv(rp, rp + al, res_attrs);
            rp += al;
          } else skip(rk & 7, rp, rend);
        }
        p += l;
      } else if (f == 2 && wt == 2) {  // ScopeMetrics
        uint64_t l = pw_varint(p, end);
        const uint8_t* sp = p; const uint8_t* send = p + l;
        while (sp < send) {
          uint64_t sk = pw_varint(sp, send);
          if ((sk >> 3) == 2 && (sk & 7) == 2) {  // Metric
            uint64_t ml = pw_varint(sp, send);
            parse_metric(sp, sp + ml, res_attrs, out);
            sp += ml;
          } else skip(sk & 7, sp, send);
        }
        p += l;
      } else skip(wt, p, end);
    }
  }

  void parse_metric(const uint8_t* p, const uint8_t* end,
                    const py::dict& res_attrs, py::list& out) {
    std::string name;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 1 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        name.assign(reinterpret_cast<const char*>(p), l); p += l;
      } else if ((f == 5 || f == 7) && wt == 2) {  // Gauge / Sum
        uint64_t l = pw_varint(p, end);
        const uint8_t* gp = p; const uint8_t* gend = p + l;
        while (gp < gend) {
          uint64_t gk = pw_varint(gp, gend);
          if ((gk >> 3) == 1 && (gk & 7) == 2) {   // data_points
            uint64_t dl = pw_varint(gp, gend);
            parse_point(gp, gp + dl, name, res_attrs, out);
            gp += dl;
          } else skip(gk & 7, gp, gend);
        }
        p += l;
      } else skip(wt, p, end);
    }
  }

  void parse_point(const uint8_t* p, const uint8_t* end, const std::string& name,
                   const py::dict& res_attrs, py::list& out) {
    py::dict attrs;
    for (auto item : res_attrs) attrs[item.first] = item.second;
    uint64_t ts = 0;
    double val = 0;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 7 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        parse_kv(p, p + l, attrs);
        p += l;
      } else if (f == 3 && wt == 1) {
        std::memcpy(&ts, p, 8); p += 8;
      } else if (f == 4 && wt == 1) {
        std::memcpy(&val, p, 8); p += 8;
      } else if (f == 6 && wt == 1) {
        int64_t iv; std::memcpy(&iv, p, 8); p += 8;
        val = (double)iv;
      } else skip(wt, p, end);
    }
    out.append(py::make_tuple(py::str(name), std::move(attrs),
                              (int64_t)(ts / 1000000ULL), val));
  }
};

// ExportLogsServiceRequest: resource_logs → scope_logs → LogRecord
// { time_unix_nano=1 fixed64; severity_text=3; body=5 AnyValue;
//   attributes=6 }. Output: [(ts_ms, severity, body, {attrs})].
class OtlpLogsParser {
 public:
  py::list parse(py::bytes data) {
    char* buf; Py_ssize_t len;
    if (PyBytes_AsStringAndSize(data.ptr(), &buf, &len) != 0)
      throw std::runtime_error("expected bytes");
    py::list out;
    const uint8_t* p = reinterpret_cast<const uint8_t*>(buf);
    const uint8_t* end = p + len;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      if ((k >> 3) == 1 && (k & 7) == 2) {
        uint64_t l = pw_varint(p, end);
        const uint8_t* rp = p; const uint8_t* rend = p + l;
        std::string service = "unknown";
        while (rp < rend) {
          uint64_t rk = pw_varint(rp, rend);
          int f = (int)(rk >> 3), wt = (int)(rk & 7);
          if (f == 1 && wt == 2) {  // Resource (attrs ignored; senders put
            uint64_t rl = pw_varint(rp, rend);   // routing labels on records)
            rp += rl;
          } else if (f == 2 && wt == 2) {  // ScopeLogs
            uint64_t sl = pw_varint(rp, rend);
            const uint8_t* lp = rp; const uint8_t* lend = rp + sl;
            while (lp < lend) {
              uint64_t lk = pw_varint(lp, lend);
              if ((lk >> 3) == 2 && (lk & 7) == 2) {  // LogRecord
                uint64_t ll = pw_varint(lp, lend);
                parse_record(lp, lp + ll, out);
                lp += ll;
              } else OtlpMetricsSkip(lk & 7, lp, lend);
            }
            rp += sl;
          } else OtlpMetricsSkip(wt, rp, rend);
        }
        p += l;
      } else OtlpMetricsSkip(k & 7, p, end);
    }
    return out;
  }

 private:
  static void OtlpMetricsSkip(int wt, const uint8_t*& p, const uint8_t* end) {
    if (wt == 0) pw_varint(p, end);
    else if (wt == 1) p += 8;
    else if (wt == 2) { uint64_t l = pw_varint(p, end); p += l; }
    else if (wt == 5) p += 4;
    else p = end;
  }

  void parse_record(const uint8_t* p, const uint8_t* end, py::list& out) {
    uint64_t ts = 0;
    std::string severity, body;
    py::dict attrs;
    while (p < end) {
      uint64_t k = pw_varint(p, end);
      int f = (int)(k >> 3), wt = (int)(k & 7);
      if (f == 1 && wt == 1) {
        std::memcpy(&ts, p, 8); p += 8;
      } else if (f == 3 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        severity.assign(reinterpret_cast<const char*>(p), l); p += l;
      } else if (f == 5 && wt == 2) {  // body AnyValue{string_value=1}
        uint64_t l = pw_varint(p, end);
        const uint8_t* bp = p; const uint8_t* bend = p + l;
        while (bp < bend) {
          uint64_t bk = pw_varint(bp, bend);
          if ((bk >> 3) == 1 && (bk & 7) == 2) {
            uint64_t sl = pw_varint(bp, bend);
            body.assign(reinterpret_cast<const char*>(bp), sl); bp += sl;
          } else OtlpMetricsSkip(bk & 7, bp, bend);
        }
        p += l;
      } else if (f == 6 && wt == 2) {
        uint64_t l = pw_varint(p, end);
        // KeyValue (string values only, like metrics attrs)
        const uint8_t* kp = p; const uint8_t* kend = p + l;
        std::string kname, vstr;
        while (kp < kend) {
          uint64_t kk = pw_varint(kp, kend);
          if ((kk >> 3) == 1 && (kk & 7) == 2) {
            uint64_t s = pw_varint(kp, kend);
            kname.assign(reinterpret_cast<const char*>(kp), s); kp += s;
          } else if ((kk >> 3) == 2 && (kk & 7) == 2) {
            uint64_t s = pw_varint(kp, kend);
            const uint8_t* vp = kp; const uint8_t* vend = kp + s;
            while (vp < vend) {
              uint64_t vk = pw_varint(vp, vend);
              if ((vk >> 3) == 1 && (vk & 7) == 2) {
                uint64_t sl = pw_varint(vp, vend);
                vstr.assign(reinterpret_cast<const char*>(vp), sl); vp += sl;
              } else OtlpMetricsSkip(vk & 7, vp, vend);
            }
            kp += s;
          } else OtlpMetricsSkip(kk & 7, kp, kend);
        }
        if (!kname.empty()) attrs[py::str(kname)] = py::str(vstr);
        p += l;
      } else OtlpMetricsSkip(wt, p, end);
    }
    out.append(py::make_tuple((int64_t)(ts / 1000000ULL), py::str(severity),
                              py::str(body), std::move(attrs)));
  }
};

// ---------------------------------------------------------------- tokenizer
// Fulltext tokenizer for log columns (reference: src/index fulltext_index —
// tantivy's default tokenizer ≈ lowercase alphanumeric runs). Terms are
// interned region-wide; output is CSR (doc offsets + term ids) ready to
// build GPU posting lists.

class Tokenizer {
 public:
  // docs: list[str|bytes]. Returns (offsets u64[n+1], term_ids i32[total],
  // new_terms [(id, str)...]).
  py::tuple tokenize(py::list docs) {
    std::vector<uint64_t> offsets;
    std::vector<int32_t> ids;
    py::list new_terms;
    offsets.reserve(docs.size() + 1);
    offsets.push_back(0);
    std::vector<int32_t> doc_terms;
    for (auto& d : docs) {
      char* buf = nullptr;
      Py_ssize_t blen = 0;
      std::string tmp;
      if (PyBytes_Check(d.ptr())) {
        PyBytes_AsStringAndSize(d.ptr(), &buf, &blen);
      } else if (PyUnicode_Check(d.ptr())) {
        tmp = py::cast<std::string>(d);
        buf = tmp.data();
        blen = tmp.size();
      }
      doc_terms.clear();
      size_t i = 0;
      while (i < (size_t)blen) {
        while (i < (size_t)blen && !isalnum((unsigned char)buf[i])) i++;
        size_t s = i;
        while (i < (size_t)blen && isalnum((unsigned char)buf[i])) i++;
        if (i > s) {
          std::string term(buf + s, i - s);
          for (auto& c : term) c = tolower((unsigned char)c);
          const uint64_t h = fnv1a(term.data(), term.size());
          int32_t tid = -1;
          auto range = tmap_.equal_range(h);
          for (auto it = range.first; it != range.second; ++it)
            if (terms_[it->second] == term) { tid = it->second; break; }
          if (tid < 0) {
            tid = (int32_t)terms_.size();
            tmap_.emplace(h, tid);
            terms_.push_back(term);
            new_terms.append(py::make_tuple(tid, py::str(term)));
          }
          // dedupe within doc (posting lists store docs, not positions)
          bool seen = false;
          for (int32_t t : doc_terms) if (t == tid) { seen = true; break; }
          if (!seen) doc_terms.push_back(tid);
        }
      }
      ids.insert(ids.end(), doc_terms.begin(), doc_terms.end());
      offsets.push_back(ids.size());
    }
    py::array_t<uint64_t> off(offsets.size());
    py::array_t<int32_t> tid(ids.size());
    std::memcpy(off.mutable_data(), offsets.data(), offsets.size() * 8);
    if (!ids.empty()) std::memcpy(tid.mutable_data(), ids.data(), ids.size() * 4);
    return py::make_tuple(std::move(off), std::move(tid), std::move(new_terms));
  }

  // term → id (-1 if unknown); query-side probe
  int32_t term_id(const std::string& term_in) const {
    std::string term = term_in;
    for (auto& c : term) c = tolower((unsigned char)c);
    const uint64_t h = fnv1a(term.data(), term.size());
    auto range = tmap_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it)
      if (terms_[it->second] == term) return it->second;
    return -1;
  }

  size_t num_terms() const { return terms_.size(); }

  // get-or-create id for an already-normalized term (persisted fulltext
  // sidecar load: terms come back verbatim from the index file, no doc
  // re-tokenization)
  int32_t intern(const std::string& term) {
    const uint64_t h = fnv1a(term.data(), term.size());
    auto range = tmap_.equal_range(h);
    for (auto it = range.first; it != range.second; ++it)
      if (terms_[it->second] == term) return it->second;
    int32_t tid = (int32_t)terms_.size();
    tmap_.emplace(h, tid);
    terms_.push_back(term);
    return tid;
  }

  // bulk variant: lens i32[k] + concatenated utf8 blob → ids i32[k]
  py::array_t<int32_t> intern_blob(py::array_t<int32_t> lens, py::bytes blob) {
    char* buf = nullptr;
    Py_ssize_t blen = 0;
    PyBytes_AsStringAndSize(blob.ptr(), &buf, &blen);
    const int32_t* ln = lens.data();
    const size_t k = lens.size();
    py::array_t<int32_t> out(k);
    int32_t* o = out.mutable_data();
    size_t off = 0;
    for (size_t i = 0; i < k; i++) {
      o[i] = intern(std::string(buf + off, ln[i]));
      off += ln[i];
    }
    return out;
  }

  py::str term_str(int32_t tid) const {
    if (tid < 0 || (size_t)tid >= terms_.size())
      throw std::out_of_range("tid");
    return py::str(terms_[tid]);
  }

 private:
  std::unordered_multimap<uint64_t, int32_t> tmap_;
  std::vector<std::string> terms_;
};

// Pack a python string column (list of str|None) into (lens i32, utf8 blob)
// — WAL string-field serialization hot path.
static py::tuple pack_str_col(py::list vals) {
  const size_t n = vals.size();
  py::array_t<int32_t> lens(n);
  int32_t* lp = lens.mutable_data();
  std::string blob;
  blob.reserve(n * 16);
  for (size_t i = 0; i < n; i++) {
    PyObject* o = vals[i].ptr();
    if (o == Py_None) {
      lp[i] = -1;
    } else if (PyUnicode_Check(o)) {
      Py_ssize_t sl;
      const char* s = PyUnicode_AsUTF8AndSize(o, &sl);
      lp[i] = (int32_t)sl;
      blob.append(s, sl);
    } else if (PyBytes_Check(o)) {
      char* s; Py_ssize_t sl;
      PyBytes_AsStringAndSize(o, &s, &sl);
      lp[i] = (int32_t)sl;
      blob.append(s, sl);
    } else {
      lp[i] = -1;
    }
  }
  return py::make_tuple(std::move(lens), py::bytes(blob));
}

// One-pass ingest router (K16 host side): per-region WAL payload bodies,
// destination offsets for the scatter_append kernel, and per-region
// counts/min/max — replaces the python argsort + per-region numpy fancy
// gathers on the ingest hot path. Payload layout matches wal.encode_batch:
// [u32 hdr_len][hdr json][i32 codes][i64 ts][f64 fields nf*m].
// `hdr_suffix` is the cached per-table json tail: everything after the
// numeric value of "n" (b', "fields": [...], "strs": [], ...}').
static py::tuple route_ingest(
    py::array_t<int32_t, py::array::c_style | py::array::forcecast> codes,
    py::array_t<int64_t, py::array::c_style | py::array::forcecast> ts,
    py::list field_arrs,          // parser field columns, each f64[n]
    py::array_t<int64_t, py::array::c_style | py::array::forcecast> fmap,
    py::array_t<int32_t, py::array::c_style | py::array::forcecast> region_of,
    int n_regions, py::bytes hdr_suffix, bool durable) {
  const int64_t n = codes.shape(0);
  const int nf = (int)fmap.shape(0);
  if (ts.shape(0) != n || region_of.shape(0) != n)
    throw std::runtime_error("route_ingest: length mismatch");
  const int32_t* cp = codes.data();
  const int64_t* tp = ts.data();
  const int64_t* fm = fmap.data();
  const int32_t* rp = region_of.data();
  std::string suffix = hdr_suffix;

  // assemble the table-ordered field matrix [nf, n] (fmap row -1 → NaN)
  std::vector<const double*> srcs(field_arrs.size());
  for (size_t i = 0; i < field_arrs.size(); i++) {
    auto a = py::cast<py::array_t<double,
        py::array::c_style | py::array::forcecast>>(field_arrs[i]);
    if (a.shape(0) != n) throw std::runtime_error("field length mismatch");
    srcs[i] = a.data();
    field_arrs[i] = a;   // keep the (possibly converted) arrays alive
  }
  py::array_t<double> out_mat({(py::ssize_t)nf, (py::ssize_t)n});
  double* fp = out_mat.mutable_data();
  {
    py::gil_scoped_release nogil;
    const double nan = std::numeric_limits<double>::quiet_NaN();
    for (int f = 0; f < nf; f++) {
      const int64_t s = fm[f];
      if (s < 0 || (size_t)s >= srcs.size()) {
        std::fill(fp + (int64_t)f * n, fp + (int64_t)(f + 1) * n, nan);
      } else {
        std::memcpy(fp + (int64_t)f * n, srcs[s], 8 * n);
      }
    }
  }

  py::array_t<int64_t> dst_off(n);
  py::array_t<int64_t> counts(n_regions), mins(n_regions), maxs(n_regions);
  int64_t* dp = dst_off.mutable_data();
  int64_t* cnt = counts.mutable_data();
  int64_t* mn = mins.mutable_data();
  int64_t* mx = maxs.mutable_data();
  for (int r = 0; r < n_regions; r++) {
    cnt[r] = 0;
    mn[r] = INT64_MAX;
    mx[r] = INT64_MIN;
  }
  {
    py::gil_scoped_release nogil;
    for (int64_t i = 0; i < n; i++) {
      const int32_t r = rp[i];
      dp[i] = cnt[r]++;
      const int64_t t = tp[i];
      if (t < mn[r]) mn[r] = t;
      if (t > mx[r]) mx[r] = t;
    }
  }

  py::list payloads;
  if (!durable) {
    for (int r = 0; r < n_regions; r++) payloads.append(py::none());
    return py::make_tuple(payloads, std::move(out_mat), std::move(dst_off),
                          std::move(counts), std::move(mins), std::move(maxs));
  }
  // allocate per-region payload buffers (header + columns)
  std::vector<char*> bufs(n_regions, nullptr);
  std::vector<int64_t> code_off(n_regions), ts_off(n_regions), f_off(n_regions);
  for (int r = 0; r < n_regions; r++) {
    if (cnt[r] == 0) {
      payloads.append(py::none());
      continue;
    }
    std::string hdr = "{\"n\": " + std::to_string(cnt[r]) + suffix;
    const int64_t m = cnt[r];
    const int64_t total = 4 + (int64_t)hdr.size() + 4 * m + 8 * m + 8LL * nf * m;
    PyObject* b = PyBytes_FromStringAndSize(nullptr, total);
    char* p = PyBytes_AS_STRING(b);
    const uint32_t hl = (uint32_t)hdr.size();
    std::memcpy(p, &hl, 4);
    std::memcpy(p + 4, hdr.data(), hl);
    bufs[r] = p;
    code_off[r] = 4 + hl;
    ts_off[r] = code_off[r] + 4 * m;
    f_off[r] = ts_off[r] + 8 * m;
    payloads.append(py::reinterpret_steal<py::object>(b));
  }
  {
    py::gil_scoped_release nogil;
    for (int64_t i = 0; i < n; i++) {
      const int32_t r = rp[i];
      char* p = bufs[r];
      const int64_t o = dp[i];
      const int64_t m = cnt[r];
      std::memcpy(p + code_off[r] + 4 * o, cp + i, 4);
      std::memcpy(p + ts_off[r] + 8 * o, tp + i, 8);
      char* fdst = p + f_off[r];
      for (int f = 0; f < nf; f++) {
        std::memcpy(fdst + 8 * ((int64_t)f * m + o), fp + (int64_t)f * n + i, 8);
      }
    }
  }
  return py::make_tuple(payloads, std::move(out_mat), std::move(dst_off),
                        std::move(counts), std::move(mins), std::move(maxs));
}

void register_pagedec(py::module_& m);

PYBIND11_MODULE(_native, m) {
  register_pagedec(m);
  m.doc() = "greptimedb_amd host-native ingest path (line parser + WAL)";
  py::class_<LineParser>(m, "LineParser")
      .def(py::init<>())
      .def("parse", &LineParser::parse)
      .def("num_series", &LineParser::num_series)
      .def("tagset_str", &LineParser::tagset_str)
      .def("field_names", &LineParser::field_names)
      .def("register_tagset", &LineParser::register_tagset)
      .def("register_field", &LineParser::register_field);
  py::class_<WalWriter>(m, "WalWriter")
      .def(py::init<>())
      .def("open_segment", &WalWriter::open_segment)
      .def("close_segment", &WalWriter::close_segment)
      .def("append", &WalWriter::append)
      .def("commit", &WalWriter::commit, py::arg("sync") = true)
      .def("segment_bytes", &WalWriter::segment_bytes);
  m.def("wal_read_segment", &wal_read_segment);
  m.def("snappy_uncompress", &py_snappy_uncompress);
  m.def("pack_str_col", &pack_str_col);
  m.def("route_ingest", &route_ingest, "K16 host-side ingest router");
  py::class_<PromWriteParser>(m, "PromWriteParser")
      .def(py::init<>())
      .def("parse", &PromWriteParser::parse, py::arg("data"), py::arg("is_snappy") = true)
      .def("num_series", &PromWriteParser::num_series);
  py::class_<OtlpMetricsParser>(m, "OtlpMetricsParser")
      .def(py::init<>())
      .def("parse", &OtlpMetricsParser::parse);
  py::class_<OtlpLogsParser>(m, "OtlpLogsParser")
      .def(py::init<>())
      .def("parse", &OtlpLogsParser::parse);
  py::class_<OtlpTraceParser>(m, "OtlpTraceParser")
      .def(py::init<>())
      .def("parse", &OtlpTraceParser::parse)
      .def("num_series", &OtlpTraceParser::num_series);
  py::class_<Tokenizer>(m, "Tokenizer")
      .def(py::init<>())
      .def("tokenize", &Tokenizer::tokenize)
      .def("term_id", &Tokenizer::term_id)
      .def("intern", &Tokenizer::intern)
      .def("intern_blob", &Tokenizer::intern_blob)
      .def("term_str", &Tokenizer::term_str)
      .def("num_terms", &Tokenizer::num_terms);
}
