// Host-side PNG container parsing for the GPU decode path
// (reference cv2.imdecode png branch, petastorm/codecs.py:106).
//
// Produces the tables the inflate + unfilter kernels consume: per-image
// geometry and the IDAT segment list (PNG splits the zlib stream across
// IDAT chunks; the GPU bit-reader chains segments).
//
// Supported: 8/16-bit greyscale (color type 0), 8-bit RGB (2), 8-bit RGBA
// (6), 8-bit grey+alpha (4), non-interlaced.  Palette (3) and Adam7
// interlacing are rejected loudly -> CPU fallback.
#include <torch/extension.h>
#include <cstdint>
#include <cstring>
#include <vector>

namespace psa {

namespace {
inline uint32_t be32(const uint8_t* p) {
  return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
         ((uint32_t)p[2] << 8) | (uint32_t)p[3];
}
}  // namespace

py::dict png_parse_batch(torch::Tensor buf, torch::Tensor val_off,
                         torch::Tensor val_len) {
  TORCH_CHECK(!buf.is_cuda() && buf.scalar_type() == torch::kUInt8);
  const uint8_t* base = buf.data_ptr<uint8_t>();
  const int64_t* off = val_off.data_ptr<int64_t>();
  const int64_t n = val_off.numel();
  const int64_t* vlen64 = val_len.scalar_type() == torch::kInt64
                              ? val_len.data_ptr<int64_t>() : nullptr;
  const int32_t* vlen32 = vlen64 ? nullptr : val_len.data_ptr<int32_t>();
  auto vlen = [&](int64_t i) {
    return vlen64 ? vlen64[i] : (int64_t)vlen32[i];
  };

  auto i32o = torch::TensorOptions().dtype(torch::kInt32);
  auto i64o = torch::TensorOptions().dtype(torch::kInt64);
  torch::Tensor width = torch::empty({n}, i32o);
  torch::Tensor height = torch::empty({n}, i32o);
  torch::Tensor channels = torch::empty({n}, i32o);
  torch::Tensor bit_depth = torch::empty({n}, i32o);
  torch::Tensor row_bytes = torch::empty({n}, i32o);
  torch::Tensor bpp = torch::empty({n}, i32o);       // filter unit
  torch::Tensor raw_size = torch::empty({n}, i64o);  // (rb+1)*h
  torch::Tensor seg_first = torch::empty({n}, i32o);
  torch::Tensor seg_count = torch::empty({n}, i32o);
  std::vector<int64_t> seg_off, seg_len;

  static const uint8_t SIG[8] = {0x89, 'P', 'N', 'G', 0x0D, 0x0A, 0x1A, 0x0A};
  for (int64_t i = 0; i < n; ++i) {
    const uint8_t* p = base + off[i];
    const int64_t len = vlen(i);
    TORCH_CHECK(len > 8 && memcmp(p, SIG, 8) == 0,
                "image ", i, ": not a PNG");
    int64_t pos = 8;
    int w = 0, h = 0, depth = 0, ctype = -1;
    seg_first[i] = (int32_t)seg_off.size();
    int nsegs = 0;
    while (pos + 8 <= len) {
      uint32_t clen = be32(p + pos);
      const uint8_t* ctag = p + pos + 4;
      const uint8_t* cdata = p + pos + 8;
      // chunk payloads we CONSUME must lie inside the buffer; a corrupt
      // length would otherwise send the GPU inflate past the upload
      if (!memcmp(ctag, "IHDR", 4)) {
        TORCH_CHECK(clen >= 13 && pos + 8 + 13 <= len,
                    "image ", i, ": truncated IHDR");
        w = (int)be32(cdata);
        h = (int)be32(cdata + 4);
        depth = cdata[8];
        ctype = cdata[9];
        TORCH_CHECK(cdata[12] == 0, "image ", i,
                    ": Adam7 interlacing unsupported — CPU fallback");
        TORCH_CHECK(ctype != 3, "image ", i,
                    ": palette PNG unsupported — CPU fallback");
        TORCH_CHECK(depth == 8 || depth == 16, "image ", i,
                    ": bit depth ", depth, " unsupported");
      } else if (!memcmp(ctag, "IDAT", 4)) {
        TORCH_CHECK(pos + 8 + (int64_t)clen <= len,
                    "image ", i, ": IDAT overruns buffer");
        seg_off.push_back(off[i] + pos + 8);
        seg_len.push_back((int64_t)clen);
        ++nsegs;
      } else if (!memcmp(ctag, "IEND", 4)) {
        break;
      }
      pos += 12 + (int64_t)clen;  // len + tag + data + crc
    }
    TORCH_CHECK(w > 0 && h > 0 && nsegs > 0,
                "image ", i, ": missing IHDR/IDAT");
    TORCH_CHECK(w <= (1 << 24) && h <= (1 << 24),
                "image ", i, ": implausible dimensions ", w, "x", h);
    int ch = (ctype == 2) ? 3 : (ctype == 6) ? 4 : (ctype == 4) ? 2 : 1;
    int fu = ch * (depth / 8);
    width[i] = w;
    height[i] = h;
    channels[i] = ch;
    bit_depth[i] = depth;
    row_bytes[i] = w * fu;
    bpp[i] = fu;
    raw_size[i] = ((int64_t)w * fu + 1) * h;
    seg_count[i] = nsegs;
  }

  int64_t ns = (int64_t)seg_off.size();
  torch::Tensor seg_off_t = torch::empty({ns}, i64o);
  torch::Tensor seg_len_t = torch::empty({ns}, i64o);
  std::memcpy(seg_off_t.data_ptr<int64_t>(), seg_off.data(), ns * 8);
  std::memcpy(seg_len_t.data_ptr<int64_t>(), seg_len.data(), ns * 8);

  py::dict out;
  out["width"] = width;
  out["height"] = height;
  out["channels"] = channels;
  out["bit_depth"] = bit_depth;
  out["row_bytes"] = row_bytes;
  out["bpp"] = bpp;
  out["raw_size"] = raw_size;
  out["seg_first"] = seg_first;
  out["seg_count"] = seg_count;
  out["seg_off"] = seg_off_t;
  out["seg_len"] = seg_len_t;
  return out;
}

// Host-side mirror of byte_array_offsets_batch for uncompressed chunks:
// per-value absolute offsets so codecs needing host header parsing (jpeg,
// png) can see the bytes without a python loop.
py::dict byte_array_host_offsets(torch::Tensor buf, torch::Tensor val_start,
                                 torch::Tensor counts) {
  TORCH_CHECK(!buf.is_cuda());
  const uint8_t* base = buf.data_ptr<uint8_t>();
  const int64_t* starts = val_start.data_ptr<int64_t>();
  const int64_t* cnt = counts.data_ptr<int64_t>();
  int64_t n_pages = val_start.numel();
  int64_t total = 0;
  for (int64_t j = 0; j < n_pages; ++j) total += cnt[j];
  auto i64o = torch::TensorOptions().dtype(torch::kInt64);
  torch::Tensor off = torch::empty({total}, i64o);
  torch::Tensor len = torch::empty({total}, i64o);
  int64_t* o = off.data_ptr<int64_t>();
  int64_t* l = len.data_ptr<int64_t>();
  int64_t k = 0;
  const int64_t bufsz = buf.numel();
  {
    // pure C++ scan: release the GIL so IO-thread parsing overlaps python
    py::gil_scoped_release nogil;
    for (int64_t j = 0; j < n_pages; ++j) {
      int64_t pos = starts[j];
      for (int64_t v = 0; v < cnt[j]; ++v) {
        TORCH_CHECK(pos >= 0 && pos + 4 <= bufsz,
                    "byte-array page ", j, ": length prefix overruns "
                    "buffer at value ", v);
        uint32_t ln;
        std::memcpy(&ln, base + pos, 4);
        TORCH_CHECK((int64_t)ln <= bufsz - pos - 4,
                    "byte-array page ", j, ": value ", v,
                    " overruns buffer");
        o[k] = pos + 4;
        l[k] = ln;
        ++k;
        pos += 4 + (int64_t)ln;
      }
    }
  }
  py::dict out;
  out["off"] = off;
  out["len"] = len;
  return out;
}

}  // namespace psa
