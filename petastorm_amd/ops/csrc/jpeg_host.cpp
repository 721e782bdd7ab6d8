// Host-side baseline-JPEG header parsing for the MI355X batch decoder.
//
// Replaces the header/metadata half of cv2.imdecode in the reference
// (petastorm/codecs.py:106).  The bitstream (Huffman/IDCT/color) half runs
// on-GPU (jpeg.hip); this file produces the batch tables those kernels
// consume:
//   * per-image geometry (dims, sampling factors, MCU grid)
//   * dequant tables (uint16 -> float)
//   * derived Huffman decode tables (8-bit fast LUT + max/min-code arrays)
//   * restart-segment table: the encoder (petastorm_amd.codecs
//     CompressedImageCodec) writes an RSTn marker every MCU row, so the scan
//     splits into independently decodable segments -> one GPU thread each.
//
// Supports baseline sequential (SOF0), 8-bit, 1 or 3 components, sampling
// (1,1)/(2,1)/(1,2)/(2,2) on component 0.  Progressive/arithmetic input is
// rejected loudly (the Python layer falls back to PIL on CPU for those).
#include <torch/extension.h>
#include <cstdint>
#include <cstring>
#include <map>
#include <thread>
#include <vector>

namespace psa {

namespace {

struct HuffTable {
  // derived decode tables (standard canonical-code derivation)
  int32_t lut[256];        // (len << 16) | symbol for codes <= 8 bits; -1 slow
  int32_t maxcode[18];     // per code length 1..16; maxcode[l] = -1 if none
  int32_t mincode[18];
  int32_t valptr[18];
  uint8_t huffval[256];
  bool used = false;
};

struct ImgInfo {
  int w = 0, h = 0, ncomp = 0;
  int comp_h[3] = {1, 1, 1}, comp_v[3] = {1, 1, 1};
  int comp_q[3] = {0, 0, 0};
  int comp_dc[3] = {0, 0, 0}, comp_ac[3] = {0, 0, 0};
  int restart_interval = 0;  // in MCUs; 0 = none
  int64_t scan_start = 0, scan_end = 0;
};

// Returns false for a non-canonical table (corrupt bits[] would otherwise
// drive the LUT fill out of bounds: 255 claimed 1-bit codes -> base up to
// 255<<7 in an int32[256]).
bool build_huff(const uint8_t* bits, const uint8_t* vals, int nvals,
                HuffTable& t) {
  // canonical code assignment (ITU T.81 Annex C)
  uint32_t code = 0;
  int k = 0;
  int codes_of_len[17];
  for (int l = 1; l <= 16; ++l) codes_of_len[l] = bits[l - 1];
  std::memcpy(t.huffval, vals, nvals);
  for (int i = 0; i < 256; ++i) t.lut[i] = -1;
  for (int l = 1; l <= 16; ++l) {
    if (codes_of_len[l]) {
      if (code + (uint32_t)codes_of_len[l] > (1u << l)) return false;
      t.valptr[l] = k;
      t.mincode[l] = code;
      for (int i = 0; i < codes_of_len[l]; ++i) {
        if (l <= 8) {
          // fill every LUT slot whose 8-bit prefix starts with this code
          int shift = 8 - l;
          int base = code << shift;
          for (int j = 0; j < (1 << shift); ++j)
            t.lut[base + j] = (l << 16) | vals[k];
        }
        ++k;
        ++code;
      }
      t.maxcode[l] = code - 1;
    } else {
      t.maxcode[l] = -1;
      t.mincode[l] = 0;
      t.valptr[l] = 0;
    }
    code <<= 1;
  }
  t.used = true;
  return k == nvals;
}

inline uint16_t be16(const uint8_t* p) {
  return (uint16_t)((p[0] << 8) | p[1]);
}

}  // namespace

// Returns a dict of CPU tensors describing the batch; see jpeg.hip for the
// kernel-side consumption.  Offsets in the segment table are absolute into
// the SAME buffer layout on device (the page bytes are uploaded verbatim).
py::dict jpeg_parse_batch(torch::Tensor buf, torch::Tensor val_off,
                          torch::Tensor val_len) {
  TORCH_CHECK(!buf.is_cuda(), "jpeg_parse_batch reads the host copy");
  TORCH_CHECK(buf.scalar_type() == torch::kUInt8);
  const uint8_t* base = buf.data_ptr<uint8_t>();
  const int64_t* off = val_off.data_ptr<int64_t>();
  const int64_t n = val_off.numel();
  const int64_t* vlen64 = nullptr;
  const int32_t* vlen32 = nullptr;
  if (val_len.scalar_type() == torch::kInt64)
    vlen64 = val_len.data_ptr<int64_t>();
  else
    vlen32 = val_len.data_ptr<int32_t>();
  auto vlen = [&](int64_t i) -> int64_t {
    return vlen64 ? vlen64[i] : (int64_t)vlen32[i];
  };

  std::vector<ImgInfo> imgs(n);
  // batch-level table pools; per image we remember which pool slot each of
  // its table ids (0..3) resolved to at SOS time.  Tables are de-duplicated
  // by content: a batch of images from one encoder collapses to a handful
  // of pool entries instead of thousands (this parse runs per row-group on
  // the hot path).
  std::vector<std::array<uint16_t, 64>> qpool;
  std::vector<HuffTable> hpool;
  std::map<std::array<uint16_t, 64>, int> qdedup;
  std::map<std::vector<uint8_t>, int> hdedup;

  std::vector<int32_t> seg_img;
  std::vector<int64_t> seg_pos, seg_end;
  std::vector<int32_t> seg_mcu0, seg_nmcu;
  seg_img.reserve(n * 64);
  seg_pos.reserve(n * 64);
  seg_end.reserve(n * 64);
  seg_mcu0.reserve(n * 64);
  seg_nmcu.reserve(n * 64);

  // the scan below is pure C++ over the pinned buffer: release the GIL so
  // the IO prefetch thread's parsing overlaps python work on other threads
  py::gil_scoped_release nogil;
  for (int64_t i = 0; i < n; ++i) {
    const uint8_t* p = base + off[i];
    const int64_t len = vlen(i);
    ImgInfo& im = imgs[i];
    TORCH_CHECK(len >= 4 && p[0] == 0xFF && p[1] == 0xD8,
                "image ", i, ": not a JPEG (missing SOI)");
    // current table slots for this image (jpeg allows redefinition)
    int cur_q[4] = {-1, -1, -1, -1};
    int cur_dc[4] = {-1, -1, -1, -1};
    int cur_ac[4] = {-1, -1, -1, -1};
    int64_t pos = 2;
    bool got_sof = false, got_sos = false;
    while (pos + 4 <= len && !got_sos) {
      TORCH_CHECK(p[pos] == 0xFF, "image ", i, ": bad marker sync at ", pos);
      uint8_t m = p[pos + 1];
      if (m == 0xD8 || (m >= 0xD0 && m <= 0xD7) || m == 0x01) {
        pos += 2;
        continue;
      }
      uint16_t seglen = be16(p + pos + 2);
      // every marker reaching here carries a length field; the payload
      // must fit in the buffer or the handlers below would read past it
      TORCH_CHECK(seglen >= 2 && pos + 2 + (int64_t)seglen <= len,
                  "image ", i, ": segment overruns buffer at ", pos);
      const uint8_t* s = p + pos + 4;
      switch (m) {
        case 0xDB: {  // DQT
          int64_t q = 0;
          while (q + 65 <= seglen - 2) {
            int prec = s[q] >> 4, id = s[q] & 15;
            TORCH_CHECK(prec == 0, "image ", i, ": 16-bit quant tables "
                        "unsupported");
            TORCH_CHECK(id < 4, "image ", i, ": quant table id ", id);
            std::array<uint16_t, 64> tab;
            for (int k2 = 0; k2 < 64; ++k2) tab[k2] = s[q + 1 + k2];
            auto it = qdedup.find(tab);
            if (it == qdedup.end()) {
              qpool.push_back(tab);
              it = qdedup.emplace(tab, (int)qpool.size() - 1).first;
            }
            cur_q[id] = it->second;
            q += 65;
          }
          break;
        }
        case 0xC4: {  // DHT
          int64_t q = 0;
          while (q + 17 <= seglen - 2) {
            int cls = s[q] >> 4, id = s[q] & 15;
            TORCH_CHECK(cls < 2 && id < 4,
                        "image ", i, ": huffman table class/id ", cls,
                        "/", id);
            const uint8_t* bits = s + q + 1;
            int nvals = 0;
            for (int l = 0; l < 16; ++l) nvals += bits[l];
            TORCH_CHECK(nvals <= 256, "image ", i, ": bad DHT");
            TORCH_CHECK(q + 17 + nvals <= (int64_t)seglen - 2,
                        "image ", i, ": DHT values overrun segment");
            std::vector<uint8_t> key(s + q + 1, s + q + 17 + nvals);
            auto it = hdedup.find(key);
            if (it == hdedup.end()) {
              HuffTable t{};
              TORCH_CHECK(build_huff(bits, s + q + 17, nvals, t),
                          "image ", i, ": non-canonical huffman table");
              hpool.push_back(t);
              it = hdedup.emplace(std::move(key),
                                  (int)hpool.size() - 1).first;
            }
            if (cls == 0)
              cur_dc[id] = it->second;
            else
              cur_ac[id] = it->second;
            q += 17 + nvals;
          }
          break;
        }
        case 0xC0: {  // SOF0 baseline
          TORCH_CHECK(seglen - 2 >= 6, "image ", i, ": truncated SOF0");
          TORCH_CHECK(s[0] == 8, "image ", i, ": only 8-bit precision");
          im.h = be16(s + 1);
          im.w = be16(s + 3);
          im.ncomp = s[5];
          TORCH_CHECK(im.ncomp == 1 || im.ncomp == 3,
                      "image ", i, ": ", im.ncomp, " components unsupported");
          TORCH_CHECK(seglen - 2 >= 6 + 3 * im.ncomp,
                      "image ", i, ": truncated SOF0 components");
          TORCH_CHECK(im.w > 0 && im.h > 0,
                      "image ", i, ": zero-sized frame");
          for (int c = 0; c < im.ncomp; ++c) {
            im.comp_h[c] = s[7 + 3 * c] >> 4;
            im.comp_v[c] = s[7 + 3 * c] & 15;
            im.comp_q[c] = s[8 + 3 * c];  // table id, resolved at SOS
            TORCH_CHECK(im.comp_h[c] >= 1 && im.comp_h[c] <= 2 &&
                        im.comp_v[c] >= 1 && im.comp_v[c] <= 2,
                        "image ", i, ": sampling factor > 2 unsupported");
            if (c > 0)
              TORCH_CHECK(im.comp_h[c] == 1 && im.comp_v[c] == 1,
                          "image ", i, ": chroma subsampling of chroma "
                          "components unsupported");
          }
          got_sof = true;
          break;
        }
        case 0xC2:
        case 0xC1:
        case 0xC3:
        case 0xC5: case 0xC6: case 0xC7:
        case 0xC9: case 0xCA: case 0xCB:
        case 0xCD: case 0xCE: case 0xCF:
          TORCH_CHECK(false, "image ", i, ": non-baseline JPEG (SOF",
                      (int)(m - 0xC0), ") — CPU fallback required");
          break;
        case 0xDD:  // DRI
          TORCH_CHECK(seglen - 2 >= 2, "image ", i, ": truncated DRI");
          im.restart_interval = be16(s);
          break;
        case 0xDA: {  // SOS
          TORCH_CHECK(got_sof, "image ", i, ": SOS before SOF");
          TORCH_CHECK(seglen - 2 >= 1, "image ", i, ": truncated SOS");
          int ns = s[0];
          TORCH_CHECK(ns == im.ncomp, "image ", i,
                      ": multi-scan JPEG unsupported");
          TORCH_CHECK(seglen - 2 >= 1 + 2 * ns,
                      "image ", i, ": truncated SOS components");
          for (int c = 0; c < ns; ++c) {
            int tid = s[2 + 2 * c];
            int dc_id = tid >> 4, ac_id = tid & 15;
            TORCH_CHECK(dc_id < 4 && ac_id < 4,
                        "image ", i, ": scan table id out of range");
            TORCH_CHECK(cur_dc[dc_id] >= 0 && cur_ac[ac_id] >= 0,
                        "image ", i, ": missing huffman table");
            im.comp_dc[c] = cur_dc[dc_id];
            im.comp_ac[c] = cur_ac[ac_id];
            TORCH_CHECK(im.comp_q[c] < 4 && cur_q[im.comp_q[c]] >= 0,
                        "image ", i, ": missing quant table");
          }
          for (int c = 0; c < im.ncomp; ++c)
            im.comp_q[c] = cur_q[im.comp_q[c]];
          im.scan_start = pos + 2 + seglen;
          got_sos = true;
          break;
        }
        default:
          break;  // APPn/COM etc: skip
      }
      pos += 2 + seglen;
    }
    TORCH_CHECK(got_sos, "image ", i, ": no scan found");

  }

  // ---- restart-segment scan (parallel over images) ----
  // memchr-driven: jump 0xFF to 0xFF instead of walking every byte; scans
  // are independent per image, so they fan out over host threads (this is
  // the heaviest host stage at fine restart intervals).
  struct SegList {
    std::vector<int64_t> pos, end;
    std::vector<int32_t> mcu0, nmcu;
  };
  std::vector<SegList> per_img(n);
  auto scan_image = [&](int64_t i) {
    ImgInfo& im = imgs[i];
    const uint8_t* p = base + off[i];
    const int64_t len = vlen(i);
    int hmax = 1, vmax = 1;
    for (int c = 0; c < im.ncomp; ++c) {
      hmax = std::max(hmax, im.comp_h[c]);
      vmax = std::max(vmax, im.comp_v[c]);
    }
    int mcus_x = (im.w + 8 * hmax - 1) / (8 * hmax);
    int mcus_y = (im.h + 8 * vmax - 1) / (8 * vmax);
    int total_mcus = mcus_x * mcus_y;
    int ri = im.restart_interval > 0 ? im.restart_interval : total_mcus;
    SegList& sl = per_img[i];
    sl.pos.reserve(total_mcus / std::max(ri, 1) + 2);
    sl.end.reserve(sl.pos.capacity());
    sl.mcu0.reserve(sl.pos.capacity());
    sl.nmcu.reserve(sl.pos.capacity());

    int64_t sp = im.scan_start;
    int64_t seg_begin = sp;
    int mcu_done = 0;
    const int64_t abs0 = off[i];
    while (sp + 1 < len) {
      const void* hit = memchr(p + sp, 0xFF, (size_t)(len - sp - 1));
      if (hit == nullptr) { sp = len; break; }
      sp = (const uint8_t*)hit - p;
      uint8_t m = p[sp + 1];
      if (m == 0x00) {  // stuffed data byte
        sp += 2;
        continue;
      }
      if (m >= 0xD0 && m <= 0xD7) {  // RSTn
        sl.pos.push_back(abs0 + seg_begin);
        sl.end.push_back(abs0 + sp);
        sl.mcu0.push_back(mcu_done);
        sl.nmcu.push_back(std::min(ri, total_mcus - mcu_done));
        mcu_done += ri;
        sp += 2;
        seg_begin = sp;
        continue;
      }
      if (m == 0xD9) break;  // EOI
      sp += 2;  // other markers: shouldn't appear in a baseline scan
    }
    im.scan_end = sp;
    if (mcu_done < total_mcus) {
      sl.pos.push_back(abs0 + seg_begin);
      sl.end.push_back(abs0 + sp);
      sl.mcu0.push_back(mcu_done);
      sl.nmcu.push_back(total_mcus - mcu_done);
    }
  };
  {
    unsigned hw = std::thread::hardware_concurrency();
    int n_threads = (int)std::min<int64_t>(std::max(1u, hw / 2), n);
    // thread spawn costs ~50us each: parallel scan only pays off for large
    // batches (measured: threads REGRESSED 256-image row-groups ~3x)
    if (n_threads <= 1 || n < 2048) {
      for (int64_t i = 0; i < n; ++i) scan_image(i);
    } else {
      std::vector<std::thread> pool;
      pool.reserve(n_threads);
      for (int t = 0; t < n_threads; ++t)
        pool.emplace_back([&, t]() {
          for (int64_t i = t; i < n; i += n_threads) scan_image(i);
        });
      for (auto& th : pool) th.join();
    }
  }
  for (int64_t i = 0; i < n; ++i) {
    SegList& sl = per_img[i];
    for (size_t k2 = 0; k2 < sl.pos.size(); ++k2) {
      seg_img.push_back((int32_t)i);
      seg_pos.push_back(sl.pos[k2]);
      seg_end.push_back(sl.end[k2]);
      seg_mcu0.push_back(sl.mcu0[k2]);
      seg_nmcu.push_back(sl.nmcu[k2]);
    }
  }

  // ---- flatten to tensors (GIL re-acquired for python object creation) ----
  py::gil_scoped_acquire gil;
  auto i32 = torch::TensorOptions().dtype(torch::kInt32);
  auto i64 = torch::TensorOptions().dtype(torch::kInt64);
  auto f32 = torch::TensorOptions().dtype(torch::kFloat32);

  int64_t nimg = n;
  torch::Tensor width = torch::empty({nimg}, i32);
  torch::Tensor height = torch::empty({nimg}, i32);
  torch::Tensor ncomp = torch::empty({nimg}, i32);
  torch::Tensor mcus_x_t = torch::empty({nimg}, i32);
  torch::Tensor mcus_y_t = torch::empty({nimg}, i32);
  torch::Tensor comp_h = torch::zeros({nimg, 3}, i32);
  torch::Tensor comp_v = torch::zeros({nimg, 3}, i32);
  torch::Tensor comp_q = torch::zeros({nimg, 3}, i32);
  torch::Tensor comp_dc = torch::zeros({nimg, 3}, i32);
  torch::Tensor comp_ac = torch::zeros({nimg, 3}, i32);
  torch::Tensor samp_off = torch::zeros({nimg, 3}, i64);
  torch::Tensor samp_stride = torch::zeros({nimg, 3}, i32);
  torch::Tensor img_block0 = torch::empty({nimg + 1}, i64);
  torch::Tensor bpm_t = torch::empty({nimg}, i32);
  torch::Tensor kmap = torch::zeros({nimg, 8}, i32);

  int64_t samp_total = 0;
  int64_t block_total = 0;
  int32_t* W = width.data_ptr<int32_t>();
  int32_t* H = height.data_ptr<int32_t>();
  int32_t* NC = ncomp.data_ptr<int32_t>();
  int32_t* MX = mcus_x_t.data_ptr<int32_t>();
  int32_t* MY = mcus_y_t.data_ptr<int32_t>();
  int32_t* CH = comp_h.data_ptr<int32_t>();
  int32_t* CV = comp_v.data_ptr<int32_t>();
  int32_t* CQ = comp_q.data_ptr<int32_t>();
  int32_t* CDC = comp_dc.data_ptr<int32_t>();
  int32_t* CAC = comp_ac.data_ptr<int32_t>();
  int64_t* SOFF = samp_off.data_ptr<int64_t>();
  int32_t* SSTR = samp_stride.data_ptr<int32_t>();
  int64_t* IB0 = img_block0.data_ptr<int64_t>();
  int32_t* BPM = bpm_t.data_ptr<int32_t>();
  int32_t* KMAP = kmap.data_ptr<int32_t>();
  for (int64_t i = 0; i < n; ++i) {
    ImgInfo& im = imgs[i];
    int hmax = 1, vmax = 1;
    for (int c = 0; c < im.ncomp; ++c) {
      hmax = std::max(hmax, im.comp_h[c]);
      vmax = std::max(vmax, im.comp_v[c]);
    }
    int mx = (im.w + 8 * hmax - 1) / (8 * hmax);
    int my = (im.h + 8 * vmax - 1) / (8 * vmax);
    W[i] = im.w;
    H[i] = im.h;
    NC[i] = im.ncomp;
    MX[i] = mx;
    MY[i] = my;
    IB0[i] = block_total;
    int bpm = 0;
    for (int c = 0; c < im.ncomp; ++c) {
      CH[i * 3 + c] = im.comp_h[c];
      CV[i * 3 + c] = im.comp_v[c];
      CQ[i * 3 + c] = im.comp_q[c];
      CDC[i * 3 + c] = im.comp_dc[c];
      CAC[i * 3 + c] = im.comp_ac[c];
      int pw = mx * im.comp_h[c] * 8;   // padded plane width in samples
      int ph = my * im.comp_v[c] * 8;
      SOFF[i * 3 + c] = samp_total;
      SSTR[i * 3 + c] = pw;
      samp_total += (int64_t)pw * ph;
      for (int v = 0; v < im.comp_v[c]; ++v)
        for (int hh = 0; hh < im.comp_h[c]; ++hh)
          KMAP[i * 8 + bpm++] = (c << 8) | (v << 4) | hh;
    }
    BPM[i] = bpm;
    block_total += (int64_t)mx * my * bpm;
  }
  IB0[nimg] = block_total;

  int64_t nq = (int64_t)qpool.size();
  torch::Tensor qtabs = torch::empty({std::max<int64_t>(nq, 1), 64}, f32);
  float* QT = qtabs.data_ptr<float>();
  for (int64_t t = 0; t < nq; ++t)
    for (int k2 = 0; k2 < 64; ++k2)
      QT[t * 64 + k2] = (float)qpool[t][k2];

  int64_t nh = (int64_t)hpool.size();
  torch::Tensor lut = torch::empty({std::max<int64_t>(nh, 1), 256}, i32);
  torch::Tensor maxcode = torch::empty({std::max<int64_t>(nh, 1), 18}, i32);
  torch::Tensor mincode = torch::empty({std::max<int64_t>(nh, 1), 18}, i32);
  torch::Tensor valptr = torch::empty({std::max<int64_t>(nh, 1), 18}, i32);
  torch::Tensor huffval = torch::zeros(
      {std::max<int64_t>(nh, 1), 256},
      torch::TensorOptions().dtype(torch::kUInt8));
  for (int64_t t = 0; t < nh; ++t) {
    std::memcpy(lut[t].data_ptr<int32_t>(), hpool[t].lut, 256 * 4);
    std::memcpy(maxcode[t].data_ptr<int32_t>(), hpool[t].maxcode, 18 * 4);
    std::memcpy(mincode[t].data_ptr<int32_t>(), hpool[t].mincode, 18 * 4);
    std::memcpy(valptr[t].data_ptr<int32_t>(), hpool[t].valptr, 18 * 4);
    std::memcpy(huffval[t].data_ptr<uint8_t>(), hpool[t].huffval, 256);
  }

  int64_t ns = (int64_t)seg_img.size();
  torch::Tensor seg_img_t = torch::empty({ns}, i32);
  torch::Tensor seg_pos_t = torch::empty({ns}, i64);
  torch::Tensor seg_end_t = torch::empty({ns}, i64);
  torch::Tensor seg_mcu0_t = torch::empty({ns}, i32);
  torch::Tensor seg_nmcu_t = torch::empty({ns}, i32);
  std::memcpy(seg_img_t.data_ptr<int32_t>(), seg_img.data(), ns * 4);
  std::memcpy(seg_pos_t.data_ptr<int64_t>(), seg_pos.data(), ns * 8);
  std::memcpy(seg_end_t.data_ptr<int64_t>(), seg_end.data(), ns * 8);
  std::memcpy(seg_mcu0_t.data_ptr<int32_t>(), seg_mcu0.data(), ns * 4);
  std::memcpy(seg_nmcu_t.data_ptr<int32_t>(), seg_nmcu.data(), ns * 4);

  py::dict out;
  out["width"] = width;
  out["height"] = height;
  out["ncomp"] = ncomp;
  out["mcus_x"] = mcus_x_t;
  out["mcus_y"] = mcus_y_t;
  out["comp_h"] = comp_h;
  out["comp_v"] = comp_v;
  out["comp_q"] = comp_q;
  out["comp_dc"] = comp_dc;
  out["comp_ac"] = comp_ac;
  out["samp_off"] = samp_off;
  out["samp_stride"] = samp_stride;
  out["samp_total"] = py::int_(samp_total);
  out["img_block0"] = img_block0;
  out["bpm"] = bpm_t;
  out["kmap"] = kmap;
  out["qtabs"] = qtabs;
  out["lut"] = lut;
  out["maxcode"] = maxcode;
  out["mincode"] = mincode;
  out["valptr"] = valptr;
  out["huffval"] = huffval;
  out["seg_img"] = seg_img_t;
  out["seg_pos"] = seg_pos_t;
  out["seg_end"] = seg_end_t;
  out["seg_mcu0"] = seg_mcu0_t;
  out["seg_nmcu"] = seg_nmcu_t;
  out["block_total"] = py::int_(block_total);
  return out;
}

}  // namespace psa
