// Parquet page-header walker: a minimal Thrift *compact protocol* parser for
// the PageHeader struct, written from the Thrift/Parquet format specs.
//
// This is the host half of the native Parquet page decode path (the device
// half is snappy.hip + parquet_decode.hip).  It replaces the page iteration
// that the reference delegates to Arrow C++ inside piece.read()
// (reference petastorm/arrow_reader_worker.py:358).
//
// Input: the raw column-chunk byte ranges (as read from disk, already in the
// pinned host buffer).  Output: flat per-page arrays the Python pipeline
// turns into batched kernel launches.
#include <torch/extension.h>
#include <cstdint>
#include <vector>

namespace psa {

namespace {

struct Cursor {
  const uint8_t* p;
  int64_t pos, end;
  bool ok = true;

  uint8_t byte() {
    if (pos >= end) { ok = false; return 0; }
    return p[pos++];
  }
  uint64_t uvarint() {
    uint64_t v = 0;
    int shift = 0;
    while (shift < 64) {
      uint8_t b = byte();
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    return v;
  }
  int64_t zigzag() {
    uint64_t u = uvarint();
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
  }
  void skip_bytes(int64_t n) {
    // A corrupt uvarint length >= 2^63 casts to negative; a backwards move
    // would never trip the pos > end check and could re-parse forever.
    if (n < 0 || n > (int64_t)(end - pos)) { ok = false; return; }
    pos += n;
  }
};

// thrift compact type codes
enum {
  T_STOP = 0, T_TRUE = 1, T_FALSE = 2, T_BYTE = 3, T_I16 = 4, T_I32 = 5,
  T_I64 = 6, T_DOUBLE = 7, T_BINARY = 8, T_LIST = 9, T_SET = 10, T_MAP = 11,
  T_STRUCT = 12
};

void skip_value(Cursor& c, int type);

void skip_struct(Cursor& c) {
  int16_t field_id = 0;
  while (c.ok) {
    uint8_t b = c.byte();
    if (b == T_STOP) return;
    int type = b & 0x0f;
    int delta = (b >> 4) & 0x0f;
    if (delta == 0)
      field_id = (int16_t)c.zigzag();
    else
      field_id = (int16_t)(field_id + delta);
    skip_value(c, type);
  }
}

void skip_value(Cursor& c, int type) {
  switch (type) {
    case T_TRUE: case T_FALSE: break;
    case T_BYTE: c.byte(); break;
    case T_I16: case T_I32: case T_I64: c.zigzag(); break;
    case T_DOUBLE: c.skip_bytes(8); break;
    case T_BINARY: c.skip_bytes((int64_t)c.uvarint()); break;
    case T_LIST: case T_SET: {
      uint8_t h = c.byte();
      int64_t n = (h >> 4) & 0x0f;
      int et = h & 0x0f;
      if (n == 15) n = (int64_t)c.uvarint();
      for (int64_t i = 0; i < n && c.ok; ++i) skip_value(c, et);
      break;
    }
    case T_MAP: {
      int64_t n = (int64_t)c.uvarint();
      if (n > 0) {
        uint8_t kv = c.byte();
        for (int64_t i = 0; i < n && c.ok; ++i) {
          skip_value(c, (kv >> 4) & 0x0f);
          skip_value(c, kv & 0x0f);
        }
      }
      break;
    }
    case T_STRUCT: skip_struct(c); break;
    default: c.ok = false;
  }
}

struct PageHeader {
  int32_t type = -1;           // 0 data_v1, 2 dict, 3 data_v2
  int32_t uncompressed = 0, compressed = 0;
  int32_t num_values = 0;
  int32_t encoding = -1, def_encoding = -1;
  int32_t dl_bytes = 0, rl_bytes = 0;   // v2 only
  bool v2_is_compressed = true;
};

// parse one PageHeader struct; returns false on malformed input
bool parse_page_header(Cursor& c, PageHeader& h) {
  int16_t fid = 0;
  while (c.ok) {
    uint8_t b = c.byte();
    if (b == T_STOP) return c.ok;
    int type = b & 0x0f;
    int delta = (b >> 4) & 0x0f;
    fid = delta ? (int16_t)(fid + delta) : (int16_t)c.zigzag();
    switch (fid) {
      case 1: h.type = (int32_t)c.zigzag(); break;
      case 2: h.uncompressed = (int32_t)c.zigzag(); break;
      case 3: h.compressed = (int32_t)c.zigzag(); break;
      case 4: skip_value(c, type); break;  // crc
      case 5: {  // DataPageHeader
        int16_t f2 = 0;
        while (c.ok) {
          uint8_t b2 = c.byte();
          if (b2 == T_STOP) break;
          int t2 = b2 & 0x0f;
          int d2 = (b2 >> 4) & 0x0f;
          f2 = d2 ? (int16_t)(f2 + d2) : (int16_t)c.zigzag();
          switch (f2) {
            case 1: h.num_values = (int32_t)c.zigzag(); break;
            case 2: h.encoding = (int32_t)c.zigzag(); break;
            case 3: h.def_encoding = (int32_t)c.zigzag(); break;
            default: skip_value(c, t2);
          }
        }
        break;
      }
      case 7: {  // DictionaryPageHeader
        int16_t f2 = 0;
        while (c.ok) {
          uint8_t b2 = c.byte();
          if (b2 == T_STOP) break;
          int t2 = b2 & 0x0f;
          int d2 = (b2 >> 4) & 0x0f;
          f2 = d2 ? (int16_t)(f2 + d2) : (int16_t)c.zigzag();
          switch (f2) {
            case 1: h.num_values = (int32_t)c.zigzag(); break;
            case 2: h.encoding = (int32_t)c.zigzag(); break;
            default: skip_value(c, t2);
          }
        }
        break;
      }
      case 8: {  // DataPageHeaderV2
        int16_t f2 = 0;
        while (c.ok) {
          uint8_t b2 = c.byte();
          if (b2 == T_STOP) break;
          int t2 = b2 & 0x0f;
          int d2 = (b2 >> 4) & 0x0f;
          f2 = d2 ? (int16_t)(f2 + d2) : (int16_t)c.zigzag();
          switch (f2) {
            case 1: h.num_values = (int32_t)c.zigzag(); break;
            case 4: h.encoding = (int32_t)c.zigzag(); break;
            case 5: h.dl_bytes = (int32_t)c.zigzag(); break;
            case 6: h.rl_bytes = (int32_t)c.zigzag(); break;
            case 7: h.v2_is_compressed = (t2 == T_TRUE); break;
            default: skip_value(c, t2);
          }
        }
        break;
      }
      default: skip_value(c, type);
    }
  }
  return false;
}

}  // namespace

// Walk all pages of each column chunk.  chunk_off/chunk_len address `buf`.
// Returns a dict of flat int64 tensors:
//   page_chunk, page_type (0 data_v1 / 2 dict / 3 data_v2), data_off (abs,
//   into buf), comp_size, uncomp_size, num_values, encoding, def_encoding,
//   dl_bytes, rl_bytes
py::dict parquet_walk_pages(torch::Tensor buf, torch::Tensor chunk_off,
                            torch::Tensor chunk_len) {
  TORCH_CHECK(!buf.is_cuda() && buf.scalar_type() == torch::kUInt8);
  const uint8_t* base = buf.data_ptr<uint8_t>();
  const int64_t* coff = chunk_off.data_ptr<int64_t>();
  const int64_t* clen = chunk_len.data_ptr<int64_t>();
  const int64_t nchunks = chunk_off.numel();

  std::vector<int64_t> p_chunk, p_type, p_off, p_comp, p_uncomp, p_nval,
      p_enc, p_denc, p_dl, p_rl, p_v2c;
  for (int64_t ci = 0; ci < nchunks; ++ci) {
    Cursor c{base, coff[ci], coff[ci] + clen[ci]};
    while (c.ok && c.pos < c.end) {
      PageHeader h;
      bool good = parse_page_header(c, h);
      TORCH_CHECK(good, "malformed page header in chunk ", ci, " at ", c.pos);
      TORCH_CHECK(h.type >= 0,
                  "page header without a page type in chunk ", ci, " at ",
                  c.pos);
      TORCH_CHECK(h.compressed >= 0 && c.pos + h.compressed <= c.end,
                  "page data overruns chunk ", ci);
      p_chunk.push_back(ci);
      p_type.push_back(h.type);
      p_off.push_back(c.pos);
      p_comp.push_back(h.compressed);
      p_uncomp.push_back(h.uncompressed);
      p_nval.push_back(h.num_values);
      p_enc.push_back(h.encoding);
      p_denc.push_back(h.def_encoding);
      p_dl.push_back(h.dl_bytes);
      p_rl.push_back(h.rl_bytes);
      p_v2c.push_back(h.v2_is_compressed ? 1 : 0);
      c.skip_bytes(h.compressed);
    }
  }
  auto mk = [&](std::vector<int64_t>& v) {
    torch::Tensor t = torch::empty({(int64_t)v.size()},
                                   torch::TensorOptions().dtype(torch::kInt64));
    std::memcpy(t.data_ptr<int64_t>(), v.data(), v.size() * 8);
    return t;
  };
  py::dict out;
  out["page_chunk"] = mk(p_chunk);
  out["page_type"] = mk(p_type);
  out["data_off"] = mk(p_off);
  out["comp_size"] = mk(p_comp);
  out["uncomp_size"] = mk(p_uncomp);
  out["num_values"] = mk(p_nval);
  out["encoding"] = mk(p_enc);
  out["def_encoding"] = mk(p_denc);
  out["dl_bytes"] = mk(p_dl);
  out["rl_bytes"] = mk(p_rl);
  out["v2_is_compressed"] = mk(p_v2c);
  return out;
}

}  // namespace psa
