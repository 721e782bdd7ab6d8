// GPU DEFLATE (zlib/raw) inflate + PNG unfilter.
//
// Replaces the zlib half of CompressedNdarrayCodec (reference np.savez /
// zlib, petastorm/codecs.py:174-212) and the PNG half of
// CompressedImageCodec (reference cv2.imdecode, petastorm/codecs.py:106).
//
// Parallel structure: DEFLATE is bit-serial with LZ77 back-references into
// the *output*, so unlike snappy there is no cheap wave cooperation; the
// MI355X design exploits *stream* parallelism instead — a row-group batch
// carries hundreds-to-thousands of independent streams (one per image /
// ndarray value), one thread each (SURVEY.md §7 "per-image parallelism").
//
// PNG unfilter: Sub/Avg/Paeth have a serial dependence on the left pixel at
// stride bpp, so a wave assigns one lane per byte-channel (bpp lanes active)
// and walks rows; rows are pipelined wave-wide via the up-row already being
// complete.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace psa {

// ---------------------------------------------------------------------------
// DEFLATE
// ---------------------------------------------------------------------------

struct InflateBits {
  const uint8_t* p;
  int64_t pos, end;       // byte cursor over (possibly segmented) stream
  const int64_t* seg_off;  // segment table (absolute offsets/lengths)
  const int64_t* seg_len;
  int seg_idx, n_segs;
  uint32_t bitbuf;
  int bitcnt;

  __device__ void init(const uint8_t* data, const int64_t* so,
                       const int64_t* sl, int nseg) {
    p = data; seg_off = so; seg_len = sl; n_segs = nseg; seg_idx = 0;
    pos = so[0]; end = so[0] + sl[0];
    bitbuf = 0; bitcnt = 0;
  }
  __device__ __forceinline__ int next_byte() {
    while (pos >= end) {
      if (++seg_idx >= n_segs) return -1;
      pos = seg_off[seg_idx];
      end = seg_off[seg_idx] + seg_len[seg_idx];
    }
    return p[pos++];
  }
  __device__ __forceinline__ uint32_t bits(int n) {
    while (bitcnt < n) {
      int b = next_byte();
      if (b < 0) b = 0;
      bitbuf |= (uint32_t)b << bitcnt;
      bitcnt += 8;
    }
    uint32_t v = bitbuf & ((1u << n) - 1u);
    bitbuf >>= n;
    bitcnt -= n;
    return v;
  }
  __device__ void align_byte() {
    bitbuf = 0;
    bitcnt = 0;
  }
};

// canonical huffman decode from code-length counts (RFC 1951 §3.2.2)
struct HuffDec {
  uint16_t count[16];   // codes of each length
  uint16_t symbol[288 + 32];

  __device__ int build(const uint8_t* lengths, int n) {
    for (int i = 0; i < 16; ++i) count[i] = 0;
    for (int i = 0; i < n; ++i) count[lengths[i]]++;
    count[0] = 0;
    uint16_t offs[16];
    offs[1] = 0;
    for (int l = 1; l < 15; ++l) offs[l + 1] = offs[l] + count[l];
    for (int i = 0; i < n; ++i)
      if (lengths[i]) symbol[offs[lengths[i]]++] = (uint16_t)i;
    return 0;
  }
  __device__ __forceinline__ int decode(InflateBits& br) {
    int code = 0, first = 0, index = 0;
    for (int len = 1; len <= 15; ++len) {
      code |= (int)br.bits(1);
      int cnt = count[len];
      if (code - first < cnt) return symbol[index + (code - first)];
      index += cnt;
      first = (first + cnt) << 1;
      code <<= 1;
    }
    return -1;
  }
};

__constant__ uint16_t LEN_BASE[29] = {3, 4, 5, 6, 7, 8, 9, 10, 11, 13, 15,
                                      17, 19, 23, 27, 31, 35, 43, 51, 59, 67,
                                      83, 99, 115, 131, 163, 195, 227, 258};
__constant__ uint8_t LEN_EXTRA[29] = {0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2,
                                      2, 2, 2, 3, 3, 3, 3, 4, 4, 4, 4, 5, 5,
                                      5, 5, 0};
__constant__ uint16_t DIST_BASE[30] = {1, 2, 3, 4, 5, 7, 9, 13, 17, 25, 33,
                                       49, 65, 97, 129, 193, 257, 385, 513,
                                       769, 1025, 1537, 2049, 3073, 4097,
                                       6145, 8193, 12289, 16385, 24577};
__constant__ uint8_t DIST_EXTRA[30] = {0, 0, 0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 5,
                                       5, 6, 6, 7, 7, 8, 8, 9, 9, 10, 10, 11,
                                       11, 12, 12, 13, 13};
__constant__ uint8_t CLC_ORDER[19] = {16, 17, 18, 0, 8, 7, 9, 6, 10, 5, 11,
                                      4, 12, 3, 13, 2, 14, 1, 15};

// One thread per stream.  mode 0 = zlib (skip 2-byte header), 1 = raw.
__global__ void inflate_kernel(const uint8_t* __restrict__ src,
                               const int64_t* __restrict__ seg_off,
                               const int64_t* __restrict__ seg_len,
                               const int32_t* __restrict__ seg_first,
                               const int32_t* __restrict__ seg_count,
                               uint8_t* __restrict__ dst,
                               const int64_t* __restrict__ dst_off,
                               const int64_t* __restrict__ dst_cap,
                               int64_t* __restrict__ produced,
                               int32_t mode, int32_t* __restrict__ status,
                               int n_streams) {
  int s = blockIdx.x * blockDim.x + threadIdx.x;
  if (s >= n_streams) return;

  InflateBits br;
  br.init(src, seg_off + seg_first[s], seg_len + seg_first[s], seg_count[s]);
  if (mode == 0) {  // zlib: CMF+FLG (assume no preset dict)
    br.next_byte();
    int flg = br.next_byte();
    if (flg & 0x20) { status[s] = 20; return; }  // FDICT unsupported
  } else if (mode == 2) {  // gzip member header (RFC 1952)
    int id1 = br.next_byte(), id2 = br.next_byte();
    if (id1 != 0x1f || id2 != 0x8b) { status[s] = 26; return; }
    int cm = br.next_byte();
    if (cm != 8) { status[s] = 26; return; }
    int flg = br.next_byte();
    for (int i = 0; i < 6; ++i) br.next_byte();  // mtime, xfl, os
    if (flg & 0x04) {  // FEXTRA
      int xl = br.next_byte();
      xl |= br.next_byte() << 8;
      for (int i = 0; i < xl; ++i) br.next_byte();
    }
    if (flg & 0x08)  // FNAME
      while (br.next_byte() > 0) {}
    if (flg & 0x10)  // FCOMMENT
      while (br.next_byte() > 0) {}
    if (flg & 0x02) { br.next_byte(); br.next_byte(); }  // FHCRC
  }
  uint8_t* out = dst + dst_off[s];
  const int64_t cap = dst_cap[s];
  int64_t o = 0;

  HuffDec lit, dist;
  uint8_t lengths[288 + 32];

  while (true) {
    uint32_t bfinal = br.bits(1);
    uint32_t btype = br.bits(2);
    if (btype == 0) {  // stored
      br.align_byte();
      int l0 = br.next_byte(), l1 = br.next_byte();
      br.next_byte(); br.next_byte();  // NLEN
      if (l0 < 0 || l1 < 0) { status[s] = 21; return; }
      int len = l0 | (l1 << 8);
      for (int i = 0; i < len; ++i) {
        int b = br.next_byte();
        if (b < 0 || o >= cap) { status[s] = 21; return; }
        out[o++] = (uint8_t)b;
      }
    } else if (btype == 1 || btype == 2) {
      if (btype == 1) {  // fixed codes
        for (int i = 0; i < 144; ++i) lengths[i] = 8;
        for (int i = 144; i < 256; ++i) lengths[i] = 9;
        for (int i = 256; i < 280; ++i) lengths[i] = 7;
        for (int i = 280; i < 288; ++i) lengths[i] = 8;
        lit.build(lengths, 288);
        for (int i = 0; i < 30; ++i) lengths[i] = 5;
        dist.build(lengths, 30);
      } else {  // dynamic codes
        int hlit = (int)br.bits(5) + 257;
        int hdist = (int)br.bits(5) + 1;
        int hclen = (int)br.bits(4) + 4;
        uint8_t clc_len[19];
        for (int i = 0; i < 19; ++i) clc_len[i] = 0;
        for (int i = 0; i < hclen; ++i)
          clc_len[CLC_ORDER[i]] = (uint8_t)br.bits(3);
        HuffDec clc;
        clc.build(clc_len, 19);
        int n = 0;
        while (n < hlit + hdist) {
          int sym = clc.decode(br);
          if (sym < 0) { status[s] = 22; return; }
          if (sym < 16) {
            lengths[n++] = (uint8_t)sym;
          } else if (sym == 16) {
            if (n == 0) { status[s] = 22; return; }
            int rep = 3 + (int)br.bits(2);
            uint8_t prev = lengths[n - 1];
            for (int i = 0; i < rep && n < hlit + hdist; ++i)
              lengths[n++] = prev;
          } else if (sym == 17) {
            int rep = 3 + (int)br.bits(3);
            for (int i = 0; i < rep && n < hlit + hdist; ++i)
              lengths[n++] = 0;
          } else {
            int rep = 11 + (int)br.bits(7);
            for (int i = 0; i < rep && n < hlit + hdist; ++i)
              lengths[n++] = 0;
          }
        }
        lit.build(lengths, hlit);
        dist.build(lengths + hlit, hdist);
      }
      // decode symbols
      while (true) {
        int sym = lit.decode(br);
        if (sym < 0) { status[s] = 23; return; }
        if (sym < 256) {
          if (o >= cap) { status[s] = 24; return; }
          out[o++] = (uint8_t)sym;
        } else if (sym == 256) {
          break;
        } else {
          sym -= 257;
          if (sym >= 29) { status[s] = 23; return; }
          int len = LEN_BASE[sym] + (int)br.bits(LEN_EXTRA[sym]);
          int dsym = dist.decode(br);
          if (dsym < 0 || dsym >= 30) { status[s] = 23; return; }
          int64_t d = DIST_BASE[dsym] + (int64_t)br.bits(DIST_EXTRA[dsym]);
          if (d > o || o + len > cap) { status[s] = 24; return; }
          const uint8_t* from = out + o - d;
          for (int i = 0; i < len; ++i) out[o + i] = from[i];
          o += len;
        }
      }
    } else {
      status[s] = 25;
      return;
    }
    if (bfinal) break;
  }
  produced[s] = o;
}

void inflate_batch(torch::Tensor src, torch::Tensor seg_off,
                   torch::Tensor seg_len, torch::Tensor seg_first,
                   torch::Tensor seg_count, torch::Tensor dst,
                   torch::Tensor dst_off, torch::Tensor dst_cap,
                   torch::Tensor produced, int64_t mode,
                   torch::Tensor status) {
  int n = (int)seg_first.numel();
  if (!n) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(inflate_kernel, dim3((n + 63) / 64), dim3(64), 0,
                     stream, src.data_ptr<uint8_t>(),
                     seg_off.data_ptr<int64_t>(),
                     seg_len.data_ptr<int64_t>(),
                     seg_first.data_ptr<int32_t>(),
                     seg_count.data_ptr<int32_t>(), dst.data_ptr<uint8_t>(),
                     dst_off.data_ptr<int64_t>(),
                     dst_cap.data_ptr<int64_t>(),
                     produced.data_ptr<int64_t>(), (int32_t)mode,
                     status.data_ptr<int32_t>(), n);
}

// ---------------------------------------------------------------------------
// PNG unfilter: one wave per image; bpp lanes cooperate on the serial
// left-dependency; remaining lanes idle (image parallelism dominates).
// Input: raw inflated scanlines [filter_byte + row_bytes] x height.
// Output: packed rows (uint8) — 16-bit samples stay big-endian here and are
// byte-swapped by the python layer's .view().byteswap-free path (we swap in
// the copy kernel below).
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint8_t paeth(int a, int b, int c) {
  int p = a + b - c;
  int pa = abs(p - a), pb = abs(p - b), pc = abs(p - c);
  if (pa <= pb && pa <= pc) return (uint8_t)a;
  if (pb <= pc) return (uint8_t)b;
  return (uint8_t)c;
}

__global__ void png_unfilter_kernel(
    uint8_t* __restrict__ raw,            // inflated scanline data (in-place)
    const int64_t* __restrict__ raw_off,  // per image
    uint8_t* __restrict__ out, const int64_t* __restrict__ out_off,
    const int32_t* __restrict__ height, const int32_t* __restrict__ row_bytes,
    const int32_t* __restrict__ bpp,      // filter unit (bytes per pixel)
    int32_t* __restrict__ status, int n_imgs) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int img = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (img >= n_imgs) return;
  const int lane = lane_id();
  const int h = height[img];
  const int rb = row_bytes[img];
  const int fu = bpp[img];
  uint8_t* r = raw + raw_off[img];
  uint8_t* o = out + out_off[img];

  for (int y = 0; y < h; ++y) {
    const uint8_t* cur_in = r + (int64_t)y * (rb + 1);
    uint8_t ft = cur_in[0];
    const uint8_t* cur = cur_in + 1;
    uint8_t* dst_row = o + (int64_t)y * rb;
    const uint8_t* up = y ? o + (int64_t)(y - 1) * rb : nullptr;
    switch (ft) {
      case 0:  // none
        for (int i = lane; i < rb; i += PSA_WAVE) dst_row[i] = cur[i];
        break;
      case 2:  // up
        if (up)
          for (int i = lane; i < rb; i += PSA_WAVE)
            dst_row[i] = (uint8_t)(cur[i] + up[i]);
        else
          for (int i = lane; i < rb; i += PSA_WAVE) dst_row[i] = cur[i];
        break;
      case 1:  // sub: serial chain at stride fu -> fu lanes
        if (lane < fu) {
          int prev = 0;
          for (int i = lane; i < rb; i += fu) {
            prev = (uint8_t)(cur[i] + prev);
            dst_row[i] = (uint8_t)prev;
          }
        }
        break;
      case 3:  // average
        if (lane < fu) {
          int prev = 0;
          for (int i = lane; i < rb; i += fu) {
            int u = up ? up[i] : 0;
            prev = (uint8_t)(cur[i] + ((prev + u) >> 1));
            dst_row[i] = (uint8_t)prev;
          }
        }
        break;
      case 4:  // paeth
        if (lane < fu) {
          int a = 0, c = 0;
          for (int i = lane; i < rb; i += fu) {
            int b = up ? up[i] : 0;
            int cc = (up && i >= fu) ? c : 0;
            int v = (uint8_t)(cur[i] + paeth(a, b, (i >= fu) ? cc : 0));
            c = b;
            a = v;
            dst_row[i] = (uint8_t)v;
          }
        }
        break;
      default:
        if (lane == 0) status[img] = 30;
        return;
    }
    __builtin_amdgcn_wave_barrier();
  }
}

void png_unfilter_batch(torch::Tensor raw, torch::Tensor raw_off,
                        torch::Tensor out, torch::Tensor out_off,
                        torch::Tensor height, torch::Tensor row_bytes,
                        torch::Tensor bpp, torch::Tensor status) {
  int n = (int)height.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(png_unfilter_kernel, dim3((n + WPB - 1) / WPB),
                     dim3(WPB * PSA_WAVE), 0, stream,
                     raw.data_ptr<uint8_t>(), raw_off.data_ptr<int64_t>(),
                     out.data_ptr<uint8_t>(), out_off.data_ptr<int64_t>(),
                     height.data_ptr<int32_t>(),
                     row_bytes.data_ptr<int32_t>(), bpp.data_ptr<int32_t>(),
                     status.data_ptr<int32_t>(), n);
}

// byte-swap 16-bit big-endian PNG samples to little-endian in place
__global__ void bswap16_kernel(uint8_t* __restrict__ data, int64_t n_pairs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n_pairs; i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t a = data[2 * i];
    data[2 * i] = data[2 * i + 1];
    data[2 * i + 1] = a;
  }
}

void bswap16(torch::Tensor data) {
  int64_t n_pairs = data.numel() / 2;
  if (!n_pairs) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(bswap16_kernel, dim3(psa::grid_for(n_pairs, 256)),
                     dim3(256), 0, stream, data.data_ptr<uint8_t>(),
                     n_pairs);
}

}  // namespace psa
