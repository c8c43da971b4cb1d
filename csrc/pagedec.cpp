// K11 host side: parquet column-chunk page parsing + zstd decompress.
//
// Reference hot loop: src/mito2/src/sst/parquet/reader.rs (parquet crate
// page decode). MI355X split (SURVEY.md §7 "hard parts"): the branchy,
// sequential parts run HERE on the host — thrift-compact PageHeader
// parsing and zstd decompress (libzstd) — producing a flat run table +
// payload blob that the DEVICE expands in parallel
// (rle_hybrid_expand_kernel + dict gather in kernels.hip).
//
// Thrift compact protocol subset: just enough to walk PageHeader
// (format/PageHeader in parquet.thrift), skipping unknown fields.

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include <zstd.h>

namespace py = pybind11;

namespace pagedec {

// ------------------------------------------------------------- thrift
struct TReader {
  const uint8_t* p;
  const uint8_t* end;

  uint64_t uvarint() {
    uint64_t v = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      v |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
    }
    throw std::runtime_error("truncated varint");
  }
  int64_t zigzag() {
    uint64_t u = uvarint();
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
  }
  void skip_bytes(size_t n) {
    if ((size_t)(end - p) < n) throw std::runtime_error("truncated thrift");
    p += n;
  }
  // skip a field of compact type `t`
  void skip(uint8_t t) {
    switch (t) {
      case 1: case 2: break;                 // bool true/false (in header)
      case 3: skip_bytes(1); break;          // byte
      case 4: case 5: case 6: uvarint(); break;  // i16/i32/i64 varint
      case 7: skip_bytes(8); break;          // double
      case 8: { uint64_t n = uvarint(); skip_bytes(n); break; }  // binary
      case 9: {                               // list
        uint8_t h = *p++;
        uint64_t n = h >> 4;
        uint8_t et = h & 0x0F;
        if (n == 15) n = uvarint();
        for (uint64_t i = 0; i < n; i++) skip(et);
        break;
      }
      case 12: skip_struct(); break;
      default: throw std::runtime_error("thrift: unsupported type " + std::to_string(t));
    }
  }
  void skip_struct() {
    int16_t fid = 0;
    while (true) {
      uint8_t b = *p++;
      if (b == 0) return;               // STOP
      uint8_t t = b & 0x0F;
      uint8_t delta = b >> 4;
      if (delta == 0) fid = (int16_t)zigzag(); else fid += delta;
      skip(t);
    }
  }
};

struct DataPageInfo {
  int32_t page_type;        // 0=data v1, 2=dict, 3=data v2
  int32_t num_values;
  int32_t encoding;         // 0=PLAIN, 8=RLE_DICTIONARY, 2=PLAIN_DICTIONARY
  int64_t payload_off;      // into the decompressed blob
  int64_t payload_len;
  int32_t def_encoding;     // v1 only
  int32_t dl_byte_length;   // v2: definition levels length (uncompressed)
  int32_t num_nulls;        // v2
};

struct PageHeaderFields {
  int32_t type = -1, uncompressed_size = 0, compressed_size = 0;
  int32_t num_values = 0, encoding = 0, def_enc = 3, rep_enc = 3;
  int32_t v2_num_nulls = 0, v2_dl_len = 0, v2_rl_len = 0;
  bool v2 = false;
};

// parse one PageHeader struct; returns fields, advances reader
static PageHeaderFields parse_page_header(TReader& r) {
  PageHeaderFields out;
  int16_t fid = 0;
  while (true) {
    uint8_t b = *r.p++;
    if (b == 0) break;
    uint8_t t = b & 0x0F;
    uint8_t delta = b >> 4;
    if (delta == 0) fid = (int16_t)r.zigzag(); else fid += delta;
    switch (fid) {
      case 1: out.type = (int32_t)r.zigzag(); break;
      case 2: out.uncompressed_size = (int32_t)r.zigzag(); break;
      case 3: out.compressed_size = (int32_t)r.zigzag(); break;
      case 5: {  // DataPageHeader
        int16_t f2 = 0;
        while (true) {
          uint8_t b2 = *r.p++;
          if (b2 == 0) break;
          uint8_t t2 = b2 & 0x0F;
          uint8_t d2 = b2 >> 4;
          if (d2 == 0) f2 = (int16_t)r.zigzag(); else f2 += d2;
          if (f2 == 1) out.num_values = (int32_t)r.zigzag();
          else if (f2 == 2) out.encoding = (int32_t)r.zigzag();
          else if (f2 == 3) out.def_enc = (int32_t)r.zigzag();
          else if (f2 == 4) out.rep_enc = (int32_t)r.zigzag();
          else r.skip(t2);
        }
        break;
      }
      case 7: {  // DictionaryPageHeader
        int16_t f2 = 0;
        while (true) {
          uint8_t b2 = *r.p++;
          if (b2 == 0) break;
          uint8_t t2 = b2 & 0x0F;
          uint8_t d2 = b2 >> 4;
          if (d2 == 0) f2 = (int16_t)r.zigzag(); else f2 += d2;
          if (f2 == 1) out.num_values = (int32_t)r.zigzag();
          else if (f2 == 2) out.encoding = (int32_t)r.zigzag();
          else r.skip(t2);
        }
        break;
      }
      case 8: {  // DataPageHeaderV2
        out.v2 = true;
        int16_t f2 = 0;
        while (true) {
          uint8_t b2 = *r.p++;
          if (b2 == 0) break;
          uint8_t t2 = b2 & 0x0F;
          uint8_t d2 = b2 >> 4;
          if (d2 == 0) f2 = (int16_t)r.zigzag(); else f2 += d2;
          if (f2 == 1) out.num_values = (int32_t)r.zigzag();
          else if (f2 == 2) out.v2_num_nulls = (int32_t)r.zigzag();
          else if (f2 == 4) out.encoding = (int32_t)r.zigzag();
          else if (f2 == 5) out.v2_dl_len = (int32_t)r.zigzag();
          else if (f2 == 6) out.v2_rl_len = (int32_t)r.zigzag();
          else r.skip(t2);
        }
        break;
      }
      default:
        r.skip(t);
    }
  }
  return out;
}

static std::vector<uint8_t> zstd_decompress(const uint8_t* src, size_t n,
                                            size_t expect) {
  std::vector<uint8_t> out(expect);
  size_t got = ZSTD_decompress(out.data(), expect, src, n);
  if (ZSTD_isError(got))
    throw std::runtime_error(std::string("zstd: ") + ZSTD_getErrorName(got));
  out.resize(got);
  return out;
}

// Parse a whole column chunk (bytes as stored in the file, ZSTD or
// uncompressed codec) into: payload blob (decompressed values sections,
// def/rep levels stripped) + page descriptors. Nullable columns whose
// pages contain nulls raise (caller falls back to the CPU reader).
//
// Returns (blob bytes,
//          pages int64[n,4]  = {kind, payload_off, payload_len, num_values}
//            kind: 0 = PLAIN values, 1 = RLE_DICTIONARY indices
//          dict_off, dict_len)        (-1 when no dictionary page)
static py::tuple parse_column_chunk(py::bytes chunk, int codec,
                                    int max_def_level) {
  std::string_view raw = std::string_view(chunk);
  const uint8_t* p = (const uint8_t*)raw.data();
  const uint8_t* end = p + raw.size();
  std::vector<uint8_t> blob;
  std::vector<int64_t> pages;
  int64_t dict_off = -1, dict_len = -1;

  // pure C++ from here (thrift walk + zstd) — release the GIL so region
  // open can parse many columns in parallel threads
  py::gil_scoped_release nogil;
  while (p < end) {
    TReader r{p, end};
    PageHeaderFields h = parse_page_header(r);
    const uint8_t* body = r.p;
    if (body + h.compressed_size > end)
      throw std::runtime_error("page body past end of chunk");
    std::vector<uint8_t> plain;
    const uint8_t* payload;
    size_t payload_len;
    if (codec == 0) {  // UNCOMPRESSED
      payload = body;
      payload_len = h.compressed_size;
    } else {           // ZSTD (the only codec this engine writes)
      if (h.v2) {
        // v2: levels are NOT compressed; only the values section is
        size_t lvl = (size_t)h.v2_dl_len + h.v2_rl_len;
        plain.resize(lvl);
        memcpy(plain.data(), body, lvl);
        auto vals = zstd_decompress(body + lvl, h.compressed_size - lvl,
                                    h.uncompressed_size - lvl);
        plain.insert(plain.end(), vals.begin(), vals.end());
      } else {
        plain = zstd_decompress(body, h.compressed_size, h.uncompressed_size);
      }
      payload = plain.data();
      payload_len = plain.size();
    }

    if (h.type == 2) {  // dictionary page (PLAIN values)
      dict_off = (int64_t)blob.size();
      dict_len = (int64_t)payload_len;
      blob.insert(blob.end(), payload, payload + payload_len);
    } else if (h.type == 0 || h.type == 3) {
      const uint8_t* v = payload;
      size_t vlen = payload_len;
      if (max_def_level > 0) {
        // strip definition levels; verify "all defined" (no nulls)
        size_t dl_len;
        const uint8_t* dl;
        if (h.v2) {
          dl = v;
          dl_len = (size_t)h.v2_dl_len;
          if (h.v2_num_nulls != 0)
            throw std::runtime_error("page has nulls (fallback)");
          v += dl_len + h.v2_rl_len;
          vlen -= dl_len + h.v2_rl_len;
        } else {
          uint32_t len4;
          memcpy(&len4, v, 4);
          dl = v + 4;
          dl_len = len4;
          v += 4 + dl_len;
          vlen -= 4 + dl_len;
          // v1 def levels RLE: verify single run of 1s (bit width 1)
          // run header varint: (count<<1) | 0 ; value byte 1
          TReader dr{dl, dl + dl_len};
          while (dr.p < dr.end) {
            uint64_t hdr = dr.uvarint();
            if (hdr & 1) {  // bit-packed group of literal bits
              uint64_t groups = hdr >> 1;
              size_t nbytes = groups;  // bit width 1 → 1 byte per 8 values
              for (size_t i = 0; i < nbytes && dr.p < dr.end; i++) {
                if (*dr.p != 0xFF && dr.p + 1 != dr.end)
                  throw std::runtime_error("page has nulls (fallback)");
                dr.p++;
              }
            } else {
              if (dr.p >= dr.end || *dr.p != 1)
                throw std::runtime_error("page has nulls (fallback)");
              dr.p++;  // RLE value (bit width 1 → 1 byte)
            }
          }
        }
      }
      int64_t kind;
      if (h.encoding == 0) kind = 0;                       // PLAIN
      else if (h.encoding == 8 || h.encoding == 2) kind = 1;  // RLE_DICT
      else throw std::runtime_error("unsupported page encoding " +
                                    std::to_string(h.encoding));
      pages.push_back(kind);
      pages.push_back((int64_t)blob.size());
      pages.push_back((int64_t)vlen);
      pages.push_back(h.num_values);
      blob.insert(blob.end(), v, v + vlen);
    }
    p = body + h.compressed_size;
  }

  py::gil_scoped_acquire gil;
  py::array_t<int64_t> parr({(py::ssize_t)(pages.size() / 4), (py::ssize_t)4});
  if (!pages.empty())
    memcpy(parr.mutable_data(), pages.data(), pages.size() * 8);
  return py::make_tuple(py::bytes((const char*)blob.data(), blob.size()),
                        parr, dict_off, dict_len);
}

// Build the flat run table for parquet hybrid RLE/bit-packed data
// (dictionary indices): runs int64[n,5] =
//   {is_packed, blob_byte_off, value (RLE) / 0, out_start, count}
// The first payload byte is the bit width (RLE_DICTIONARY pages).
static py::array_t<int64_t> rle_run_table(py::bytes blob_b, int64_t off,
                                          int64_t len, int64_t num_values,
                                          int* out_bw) {
  std::string_view raw = std::string_view(blob_b);
  const uint8_t* base = (const uint8_t*)raw.data() + off;
  const uint8_t* p = base;
  const uint8_t* end = base + len;
  int bw = *p++;
  std::vector<int64_t> runs;
  int64_t out_pos = 0;
  TReader r{p, end};
  while (r.p < r.end && out_pos < num_values) {
    uint64_t hdr = r.uvarint();
    if (hdr & 1) {  // bit-packed: (hdr>>1) groups of 8 values
      int64_t groups = (int64_t)(hdr >> 1);
      int64_t count = groups * 8;
      if (out_pos + count > num_values) count = num_values - out_pos;
      runs.insert(runs.end(), {1, (int64_t)(r.p - (const uint8_t*)raw.data()),
                               0, out_pos, count});
      r.p += groups * bw;  // bw bits × 8 values = bw bytes per group
      out_pos += count;
    } else {
      int64_t count = (int64_t)(hdr >> 1);
      int64_t nbytes = (bw + 7) / 8;
      uint64_t v = 0;
      for (int64_t i = 0; i < nbytes; i++) v |= (uint64_t)r.p[i] << (8 * i);
      r.p += nbytes;
      if (out_pos + count > num_values) count = num_values - out_pos;
      runs.insert(runs.end(), {0, 0, (int64_t)v, out_pos, count});
      out_pos += count;
    }
  }
  if (out_pos < num_values)
    throw std::runtime_error("rle: short page");
  *out_bw = bw;
  py::array_t<int64_t> arr({(py::ssize_t)(runs.size() / 5), (py::ssize_t)5});
  if (!runs.empty()) memcpy(arr.mutable_data(), runs.data(), runs.size() * 8);
  return arr;
}

static py::tuple py_rle_run_table(py::bytes blob, int64_t off, int64_t len,
                                  int64_t num_values) {
  int bw = 0;
  auto arr = rle_run_table(blob, off, len, num_values, &bw);
  return py::make_tuple(arr, bw);
}

}  // namespace pagedec

void register_pagedec(py::module_& m) {
  m.def("parse_column_chunk", &pagedec::parse_column_chunk,
        "K11 host half: page headers + zstd → payload blob + page table");
  m.def("rle_run_table", &pagedec::py_rle_run_table,
        "hybrid RLE/bit-packed run table for device expansion");
}
