// GPU baseline-JPEG decode for MI355X: the bitstream half of the
// CompressedImageCodec (reference cv2.imdecode, petastorm/codecs.py:106),
// re-designed for CDNA4 instead of delegating to a CPU library.
//
// Three kernels, sized by their natural parallelism:
//  1. huffman_decode: one thread per *restart segment*.  The encoder
//     (petastorm_amd CompressedImageCodec) emits an RSTn marker every MCU
//     row, so a batch of B images yields B x mcu_rows independent bitstreams
//     — thousands of threads.  Output: dequantized coefficients.
//  2. idct8x8: one thread per 8x8 block (hundreds of thousands per batch) —
//     direct separable basis-matrix IDCT in fp32 (exact, matches libjpeg
//     within +-1), fully unrolled so the 128 block values live in VGPRs
//     (guide rule: runtime-indexed arrays spill to scratch).
//  3. upsample_color: one thread per output pixel; libjpeg-compatible
//     "fancy" (triangle) chroma upsampling + BT.601 YCbCr->RGB, coalesced
//     NHWC uint8 stores.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace psa {

// zigzag index -> natural (row-major) index
__constant__ int ZIGZAG_NAT[64] = {
    0,  1,  8,  16, 9,  2,  3,  10, 17, 24, 32, 25, 18, 11, 4,  5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6,  7,  14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63};

// 8x8 IDCT basis: B[u][x] = c(u)/2 * cos((2x+1) u pi / 16), c(0)=1/sqrt(2)
__constant__ float IDCT_B[8][8] = {
    {0.35355339f, 0.35355339f, 0.35355339f, 0.35355339f,
     0.35355339f, 0.35355339f, 0.35355339f, 0.35355339f},
    {0.49039264f, 0.41573481f, 0.27778512f, 0.09754516f,
     -0.09754516f, -0.27778512f, -0.41573481f, -0.49039264f},
    {0.46193977f, 0.19134172f, -0.19134172f, -0.46193977f,
     -0.46193977f, -0.19134172f, 0.19134172f, 0.46193977f},
    {0.41573481f, -0.09754516f, -0.49039264f, -0.27778512f,
     0.27778512f, 0.49039264f, 0.09754516f, -0.41573481f},
    {0.35355339f, -0.35355339f, -0.35355339f, 0.35355339f,
     0.35355339f, -0.35355339f, -0.35355339f, 0.35355339f},
    {0.27778512f, -0.49039264f, 0.09754516f, 0.41573481f,
     -0.41573481f, -0.09754516f, 0.49039264f, -0.27778512f},
    {0.19134172f, -0.46193977f, 0.46193977f, -0.19134172f,
     -0.19134172f, 0.46193977f, -0.46193977f, 0.19134172f},
    {0.09754516f, -0.27778512f, 0.41573481f, -0.49039264f,
     0.49039264f, -0.41573481f, 0.27778512f, -0.09754516f}};

struct JpegTables {
  const int32_t* lut;        // [nt][256] (len<<16)|sym, -1 = slow path
  const int32_t* maxcode;    // [nt][18]
  const int32_t* mincode;    // [nt][18]
  const int32_t* valptr;     // [nt][18]
  const uint8_t* huffval;    // [nt][256]
  const float* qtabs;        // [nq][64] (zigzag order, as stored in DQT)
};

struct JpegGeom {
  const int32_t* comp_h;     // [n][3]
  const int32_t* comp_v;
  const int32_t* comp_q;
  const int32_t* comp_dc;
  const int32_t* comp_ac;
  const int64_t* samp_off;   // [n][3]
  const int32_t* samp_stride;
  const int32_t* mcus_x;
  const int32_t* mcus_y;
  const int32_t* ncomp;
  const int32_t* width;
  const int32_t* height;
  const int64_t* img_block0;  // [n+1]
  const int32_t* bpm;         // blocks per MCU
  const int32_t* kmap;        // [n][8] (comp<<8)|(v<<4)|h
};

// ---------------------------------------------------------------------------
// kernel 1: restart-segment Huffman decode -> dequantized coefficients
// ---------------------------------------------------------------------------

struct BitReader {
  const uint8_t* p;
  int64_t pos, end;
  uint64_t buf;   // left-aligned: next bit is bit 63
  int cnt;
  uint32_t sw0, sw1;    // prefetched words (little-endian composed)
  int64_t spos;         // byte position of sw0
  int scount;           // prefetched words available (0..2)

  __device__ void init(const uint8_t* data, int64_t lo, int64_t hi) {
    p = data; pos = lo; end = hi; buf = 0; cnt = 0; scount = 0; spos = 0;
  }
  // JPEG entropy data marks every 0xFF with a stuffed 0x00.  The common
  // case (no 0xFF in the next 4 bytes, ~98%) refills 32 bits from a word
  // load; the kernel is L2-LATENCY bound (PMC: 61% wave cycles waiting at
  // 92% L2 hit), so each refill trip also PREFETCHES the following word —
  // the two loads issue together and every other refill costs no memory
  // trip at all.
  __device__ __forceinline__ void fill() {
    while (cnt <= 32) {
      if (pos + 4 <= end) {
        uint32_t w;
        if (scount > 0 && spos == pos) {
          w = sw0;
          sw0 = sw1;
          spos += 4;
          --scount;
        } else {
          scount = 0;
          w = (uint32_t)p[pos] | ((uint32_t)p[pos + 1] << 8) |
              ((uint32_t)p[pos + 2] << 16) | ((uint32_t)p[pos + 3] << 24);
          if (pos + 12 <= end) {  // all 12 byte loads issue together (ILP)
            sw0 = (uint32_t)p[pos + 4] | ((uint32_t)p[pos + 5] << 8) |
                  ((uint32_t)p[pos + 6] << 16) |
                  ((uint32_t)p[pos + 7] << 24);
            sw1 = (uint32_t)p[pos + 8] | ((uint32_t)p[pos + 9] << 8) |
                  ((uint32_t)p[pos + 10] << 16) |
                  ((uint32_t)p[pos + 11] << 24);
            spos = pos + 4;
            scount = 2;
          } else if (pos + 8 <= end) {
            sw0 = (uint32_t)p[pos + 4] | ((uint32_t)p[pos + 5] << 8) |
                  ((uint32_t)p[pos + 6] << 16) |
                  ((uint32_t)p[pos + 7] << 24);
            spos = pos + 4;
            scount = 1;
          }
        }
        // detect any 0xFF byte: a byte of ~w is zero iff the byte is 0xFF
        uint32_t inv = ~w;
        if (!((inv - 0x01010101u) & ~inv & 0x80808080u)) {
          uint32_t be = __builtin_bswap32(w);
          buf |= (uint64_t)be << (32 - cnt);
          cnt += 32;
          pos += 4;
          continue;
        }
      }
      // slow path: one byte with stuffing/marker handling
      uint8_t b = 0;
      if (pos < end) {
        b = p[pos];
        if (b == 0xFF) {
          if (pos + 1 < end && p[pos + 1] == 0x00) {
            pos += 2;  // stuffed 0xFF data byte
          } else {
            pos = end;  // marker: stream over, pad with zeros
            b = 0;
          }
        } else {
          pos += 1;
        }
      } else {
        // padding beyond the segment: zeros
        buf |= 0;
        cnt += 8;
        continue;
      }
      buf |= (uint64_t)b << (56 - cnt);
      cnt += 8;
    }
  }
  __device__ __forceinline__ uint32_t peek(int n) {
    return (uint32_t)(buf >> (64 - n));
  }
  __device__ __forceinline__ void consume(int n) {
    buf <<= n;
    cnt -= n;
  }
};

__device__ __forceinline__ int huff_decode(BitReader& br,
                                           const JpegTables& t, int tid) {
  br.fill();
  uint32_t look = br.peek(8);
  int32_t hit = t.lut[tid * 256 + look];
  if (hit >= 0) {
    br.consume(hit >> 16);
    return hit & 0xFF;
  }
  // slow path: codes of length 9..16
  for (int l = 9; l <= 16; ++l) {
    int32_t code = (int32_t)br.peek(l);
    int32_t mc = t.maxcode[tid * 18 + l];
    if (mc >= 0 && code <= mc) {
      br.consume(l);
      int idx = t.valptr[tid * 18 + l] + code - t.mincode[tid * 18 + l];
      return t.huffval[tid * 256 + idx];
    }
  }
  return -1;  // corrupt
}

__device__ __forceinline__ int receive_extend(BitReader& br, int s) {
  br.fill();
  int v = (int)br.peek(s);
  br.consume(s);
  if (v < (1 << (s - 1))) v += (-1 << s) + 1;
  return v;
}

// Decode one 8x8 block's Huffman-coded coefficients.  `pred` is a named
// register passed by reference (a pred[c] array would be runtime-indexed and
// spill to scratch — guide rule: dynamic-indexed locals live in local
// memory).  Returns 0 on success, an error code otherwise.
template <bool NT>
__device__ __forceinline__ int decode_block(BitReader& br,
                                            const JpegTables& tabs,
                                            const float* __restrict__ q,
                                            int dc_t, int ac_t,
                                            float* __restrict__ out,
                                            int& pred) {
  int t = huff_decode(br, tabs, dc_t);
  if (t < 0 || t > 15) return 10;
  int diff = t ? receive_extend(br, t) : 0;
  pred += diff;
  // Non-temporal stores: coefficients are written once here and read
  // once (much later) by the IDCT kernel — keeping them OUT of L2
  // preserves it for this kernel's latency-critical bitstream refills
  // (85.7% L2 hit at the tuned config, profiles/r2_huffman_pmc.txt).
  if (NT) __builtin_nontemporal_store((float)pred * q[0], out);
  else out[0] = (float)pred * q[0];
  int kk = 1;
  while (kk < 64) {
    int rs = huff_decode(br, tabs, ac_t);
    if (rs < 0) return 11;
    int r = rs >> 4, sz = rs & 15;
    if (sz == 0) {
      if (r != 15) break;  // EOB
      kk += 16;
    } else {
      kk += r;
      if (kk > 63) return 12;
      int v = receive_extend(br, sz);
      if (NT) __builtin_nontemporal_store((float)v * q[kk],
                                          out + ZIGZAG_NAT[kk]);
      else out[ZIGZAG_NAT[kk]] = (float)v * q[kk];
      ++kk;
    }
  }
  return 0;
}

template <bool NT>
__global__ void jpeg_huffman_kernel(
    const uint8_t* __restrict__ data, JpegTables tabs, JpegGeom g,
    const int32_t* __restrict__ seg_img, const int64_t* __restrict__ seg_pos,
    const int64_t* __restrict__ seg_end, const int32_t* __restrict__ seg_mcu0,
    const int32_t* __restrict__ seg_nmcu, float* __restrict__ coef,
    int32_t* __restrict__ status, int n_segs) {
  int s = blockIdx.x * blockDim.x + threadIdx.x;
  if (s >= n_segs) return;
  const int img = seg_img[s];
  const int bpm = g.bpm[img];
  const int ncomp = g.ncomp[img];

  BitReader br;
  br.init(data, seg_pos[s], seg_end[s]);

  // hoist per-component tables into named registers (per-image constants)
  const float* q0 = tabs.qtabs + (int64_t)g.comp_q[img * 3 + 0] * 64;
  const int dc0 = g.comp_dc[img * 3 + 0], ac0 = g.comp_ac[img * 3 + 0];
  const int rep0 = g.comp_h[img * 3 + 0] * g.comp_v[img * 3 + 0];
  const float* q1 = q0;
  const float* q2 = q0;
  int dc1 = 0, ac1 = 0, dc2 = 0, ac2 = 0;
  if (ncomp == 3) {
    q1 = tabs.qtabs + (int64_t)g.comp_q[img * 3 + 1] * 64;
    dc1 = g.comp_dc[img * 3 + 1];
    ac1 = g.comp_ac[img * 3 + 1];
    q2 = tabs.qtabs + (int64_t)g.comp_q[img * 3 + 2] * 64;
    dc2 = g.comp_dc[img * 3 + 2];
    ac2 = g.comp_ac[img * 3 + 2];
  }

  int pred0 = 0, pred1 = 0, pred2 = 0;
  const int mcu0 = seg_mcu0[s];
  const int nmcu = seg_nmcu[s];
  const int64_t blk0 = g.img_block0[img];

  for (int m = 0; m < nmcu; ++m) {
    float* mcu_out = coef + (blk0 + (int64_t)(mcu0 + m) * bpm) * 64;
    int rc = 0;
    for (int r = 0; r < rep0 && !rc; ++r, mcu_out += 64)
      rc = decode_block<NT>(br, tabs, q0, dc0, ac0, mcu_out, pred0);
    if (!rc && ncomp == 3) {
      rc = decode_block<NT>(br, tabs, q1, dc1, ac1, mcu_out, pred1);
      mcu_out += 64;
      if (!rc)
        rc = decode_block<NT>(br, tabs, q2, dc2, ac2, mcu_out, pred2);
    }
    if (rc) { status[s] = rc; return; }
  }
}

// ---------------------------------------------------------------------------
// kernel 2: 8x8 IDCT per block -> uint8 component planes
// ---------------------------------------------------------------------------

__global__ void jpeg_idct_kernel(const float* __restrict__ coef,
                                 JpegGeom g, uint8_t* __restrict__ samples,
                                 int n_imgs, int64_t block_total) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= block_total) return;
  // binary-search the image owning block b
  int lo = 0, hi = n_imgs - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (g.img_block0[mid] <= b) lo = mid; else hi = mid - 1;
  }
  const int img = lo;
  const int bpm = g.bpm[img];
  const int64_t r = b - g.img_block0[img];
  const int m = (int)(r / bpm);
  const int k = (int)(r % bpm);
  const int km = g.kmap[img * 8 + k];
  const int c = km >> 8, v = (km >> 4) & 15, h = km & 15;
  const int mx = m % g.mcus_x[img];
  const int my = m / g.mcus_x[img];
  const int stride = g.samp_stride[img * 3 + c];
  uint8_t* dst = samples + g.samp_off[img * 3 + c] +
                 ((int64_t)(my * g.comp_v[img * 3 + c] + v) * 8) * stride +
                 (int64_t)(mx * g.comp_h[img * 3 + c] + h) * 8;

  const float* C = coef + b * 64;
  float tmp[64];
  // rows pass: tmp[v][x] = sum_u C[v][u] * B[u][x]
#pragma unroll
  for (int vv = 0; vv < 8; ++vv) {
#pragma unroll
    for (int x = 0; x < 8; ++x) {
      float acc = 0.f;
#pragma unroll
      for (int u = 0; u < 8; ++u) acc += C[vv * 8 + u] * IDCT_B[u][x];
      tmp[vv * 8 + x] = acc;
    }
  }
  // cols pass + level shift + clamp + store
#pragma unroll
  for (int y = 0; y < 8; ++y) {
    uint8_t row[8];
#pragma unroll
    for (int x = 0; x < 8; ++x) {
      float acc = 0.f;
#pragma unroll
      for (int vv = 0; vv < 8; ++vv) acc += tmp[vv * 8 + x] * IDCT_B[vv][y];
      int pix = __float2int_rn(acc) + 128;
      row[x] = (uint8_t)min(255, max(0, pix));
    }
    // 8-byte store
    *(uint64_t*)(dst + (int64_t)y * stride) = *(const uint64_t*)row;
  }
}

// ---------------------------------------------------------------------------
// kernel 3: fancy (triangle) chroma upsample + YCbCr->RGB, NHWC uint8 out
// ---------------------------------------------------------------------------

__device__ __forceinline__ int fancy_sample(const uint8_t* plane, int stride,
                                            int cw, int ch, int x, int y,
                                            int sx, int sy) {
  // sx/sy in {1,2}: upsampling factor of this component in each axis.
  if (sx == 1 && sy == 1) return plane[(int64_t)y * stride + x] << 4;  // x16
  int i = x, j = y, dx = 0, dy = 0;
  if (sx == 2) { i = x >> 1; dx = x & 1; }
  if (sy == 2) { j = y >> 1; dy = y & 1; }
  int i2 = min(max(dx ? i + 1 : i - 1, 0), cw - 1);
  int j2 = min(max(dy ? j + 1 : j - 1, 0), ch - 1);
  if (sx == 2 && sy == 2) {
    int t0 = 3 * plane[(int64_t)j * stride + i] +
             plane[(int64_t)j2 * stride + i];
    int t1 = 3 * plane[(int64_t)j * stride + i2] +
             plane[(int64_t)j2 * stride + i2];
    return (3 * t0 + t1 + (dx ? 7 : 8));  // x16 scale
  }
  if (sx == 2) {  // h2v1
    int a = plane[(int64_t)j * stride + i], bq = plane[(int64_t)j * stride + i2];
    return (3 * a + bq + (dx ? 2 : 1)) << 2;  // x16
  }
  // h1v2
  int a = plane[(int64_t)j * stride + i], bq = plane[(int64_t)j2 * stride + i];
  return (3 * a + bq + (dy ? 2 : 1)) << 2;
}

__global__ void jpeg_color_kernel(const uint8_t* __restrict__ samples,
                                  JpegGeom g,
                                  uint8_t* __restrict__ out,
                                  const int64_t* __restrict__ out_off,
                                  int n_imgs) {
  const int img = blockIdx.y;
  const int W = g.width[img], H = g.height[img];
  const int64_t npix = (int64_t)W * H;
  const int nc = g.ncomp[img];
  const uint8_t* yplane = samples + g.samp_off[img * 3 + 0];
  const int ystride = g.samp_stride[img * 3 + 0];
  uint8_t* dst = out + out_off[img];

  const int hmax = g.comp_h[img * 3 + 0];
  const int vmax = g.comp_v[img * 3 + 0];

  // Quad-per-thread: 4 consecutive pixels of ONE row, so the 12 RGB bytes
  // leave as three u32 stores and the 4 luma samples arrive as one u32 load
  // (CDNA4 rule: >=4B per access on the hot path; byte stores were 9% of
  // the imagenet GPU trace).  Quads never straddle rows; the row tail and
  // unaligned destinations fall back to byte stores.
  const int Wq = (W + 3) >> 2;           // quads per row
  const int64_t nquads = (int64_t)Wq * H;
  const uint8_t* cbp = samples + g.samp_off[img * 3 + 1];
  const uint8_t* crp = samples + g.samp_off[img * 3 + 2];
  const int cstride = g.samp_stride[img * 3 + 1];
  const int cw = (W + hmax - 1) / hmax;   // chroma valid width
  const int chh = (H + vmax - 1) / vmax;

  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       q < nquads; q += (int64_t)gridDim.x * blockDim.x) {
    const int y = (int)(q / Wq);
    const int x0 = (int)(q % Wq) << 2;
    const int cnt = min(4, W - x0);
    const int64_t pix0 = (int64_t)y * W + x0;
    const uint8_t* yrow = yplane + (int64_t)y * ystride + x0;

    if (nc == 1) {
      if (cnt == 4 && (((uintptr_t)(dst + pix0) & 3) == 0) &&
          (((uintptr_t)yrow & 3) == 0))
        *(uint32_t*)(dst + pix0) = *(const uint32_t*)yrow;
      else
        for (int k = 0; k < cnt; ++k) dst[pix0 + k] = yrow[k];
      continue;
    }

    uint8_t rgb[12];
    uint32_t y4 = (cnt == 4) ? load_u32_unaligned(yrow) : 0;
    for (int k = 0; k < cnt; ++k) {
      const int x = x0 + k;
      const float fy = (cnt == 4) ? (float)((y4 >> (8 * k)) & 0xff)
                                  : (float)yrow[k];
      // x16 fixed-point chroma after fancy upsample
      float cb = fancy_sample(cbp, cstride, cw, chh, x, y, hmax, vmax)
                     * (1.f / 16.f) - 128.f;
      float cr = fancy_sample(crp, cstride, cw, chh, x, y, hmax, vmax)
                     * (1.f / 16.f) - 128.f;
      int rv = __float2int_rn(fy + 1.40200f * cr);
      int gv = __float2int_rn(fy - 0.34414f * cb - 0.71414f * cr);
      int bv = __float2int_rn(fy + 1.77200f * cb);
      rgb[k * 3 + 0] = (uint8_t)min(255, max(0, rv));
      rgb[k * 3 + 1] = (uint8_t)min(255, max(0, gv));
      rgb[k * 3 + 2] = (uint8_t)min(255, max(0, bv));
    }
    uint8_t* d = dst + pix0 * 3;
    if (cnt == 4 && (((uintptr_t)d & 3) == 0)) {
      ((uint32_t*)d)[0] = *(const uint32_t*)(rgb + 0);
      ((uint32_t*)d)[1] = *(const uint32_t*)(rgb + 4);
      ((uint32_t*)d)[2] = *(const uint32_t*)(rgb + 8);
    } else {
      for (int k = 0; k < cnt * 3; ++k) d[k] = rgb[k];
    }
  }
}

// Fused variant: YCbCr->RGB + normalize + NCHW fp32 in one pass.  Replaces
// color kernel + separate nhwc_to_nchw_normalize for the dominant
// training-input transform: skips writing/re-reading the NHWC uint8
// intermediate (154 MB per 1024-image row-group — profiled at ~15% of
// imagenet GPU time as a separate kernel).  Bit-exact with the two-step
// path: the u8 clamp/round happens first, then the normalize.
__global__ void jpeg_color_norm_kernel(
    const uint8_t* __restrict__ samples, JpegGeom g,
    float* __restrict__ out, const int64_t* __restrict__ out_off,
    const float* __restrict__ mean, const float* __restrict__ inv_std,
    float scale, int n_imgs) {
  const int img = blockIdx.y;
  const int W = g.width[img], H = g.height[img];
  const int64_t plane = (int64_t)W * H;
  const int nc = g.ncomp[img];
  const uint8_t* yplane = samples + g.samp_off[img * 3 + 0];
  const int ystride = g.samp_stride[img * 3 + 0];
  float* dst = out + out_off[img];          // [3][H][W] floats

  const int hmax = g.comp_h[img * 3 + 0];
  const int vmax = g.comp_v[img * 3 + 0];
  const int Wq = (W + 3) >> 2;
  const int64_t nquads = (int64_t)Wq * H;
  const uint8_t* cbp = samples + g.samp_off[img * 3 + 1];
  const uint8_t* crp = samples + g.samp_off[img * 3 + 2];
  const int cstride = g.samp_stride[img * 3 + 1];
  const int cw = (W + hmax - 1) / hmax;
  const int chh = (H + vmax - 1) / vmax;
  const float m0 = mean[0], m1 = mean[1], m2 = mean[2];
  const float s0 = inv_std[0], s1 = inv_std[1], s2 = inv_std[2];

  for (int64_t q = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       q < nquads; q += (int64_t)gridDim.x * blockDim.x) {
    const int y = (int)(q / Wq);
    const int x0 = (int)(q % Wq) << 2;
    const int cnt = min(4, W - x0);
    const int64_t pix0 = (int64_t)y * W + x0;
    const uint8_t* yrow = yplane + (int64_t)y * ystride + x0;

    float rpix[4], gpix[4], bpix[4];
    uint32_t y4 = (cnt == 4) ? load_u32_unaligned(yrow) : 0;
    for (int k = 0; k < cnt; ++k) {
      const int x = x0 + k;
      const float fy = (cnt == 4) ? (float)((y4 >> (8 * k)) & 0xff)
                                  : (float)yrow[k];
      int rv, gv, bv;
      if (nc == 1) {
        rv = gv = bv = (int)fy;
      } else {
        float cb = fancy_sample(cbp, cstride, cw, chh, x, y, hmax, vmax)
                       * (1.f / 16.f) - 128.f;
        float cr = fancy_sample(crp, cstride, cw, chh, x, y, hmax, vmax)
                       * (1.f / 16.f) - 128.f;
        rv = __float2int_rn(fy + 1.40200f * cr);
        gv = __float2int_rn(fy - 0.34414f * cb - 0.71414f * cr);
        bv = __float2int_rn(fy + 1.77200f * cb);
      }
      rpix[k] = ((float)min(255, max(0, rv)) * scale - m0) * s0;
      gpix[k] = ((float)min(255, max(0, gv)) * scale - m1) * s1;
      bpix[k] = ((float)min(255, max(0, bv)) * scale - m2) * s2;
    }
    float* dr = dst + pix0;
    float* dg = dst + plane + pix0;
    float* db = dst + 2 * plane + pix0;
    if (cnt == 4 && ((pix0 & 3) == 0)) {   // 16B-aligned float4 stores
      *(float4*)dr = make_float4(rpix[0], rpix[1], rpix[2], rpix[3]);
      *(float4*)dg = make_float4(gpix[0], gpix[1], gpix[2], gpix[3]);
      *(float4*)db = make_float4(bpix[0], bpix[1], bpix[2], bpix[3]);
    } else {
      for (int k = 0; k < cnt; ++k) {
        dr[k] = rpix[k];
        dg[k] = gpix[k];
        db[k] = bpix[k];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static JpegTables make_tables(const py::dict& d) {
  JpegTables t;
  t.lut = d["lut"].cast<torch::Tensor>().data_ptr<int32_t>();
  t.maxcode = d["maxcode"].cast<torch::Tensor>().data_ptr<int32_t>();
  t.mincode = d["mincode"].cast<torch::Tensor>().data_ptr<int32_t>();
  t.valptr = d["valptr"].cast<torch::Tensor>().data_ptr<int32_t>();
  t.huffval = d["huffval"].cast<torch::Tensor>().data_ptr<uint8_t>();
  t.qtabs = d["qtabs"].cast<torch::Tensor>().data_ptr<float>();
  return t;
}

static JpegGeom make_geom(const py::dict& d) {
  JpegGeom g;
  g.comp_h = d["comp_h"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.comp_v = d["comp_v"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.comp_q = d["comp_q"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.comp_dc = d["comp_dc"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.comp_ac = d["comp_ac"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.samp_off = d["samp_off"].cast<torch::Tensor>().data_ptr<int64_t>();
  g.samp_stride = d["samp_stride"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.mcus_x = d["mcus_x"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.mcus_y = d["mcus_y"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.ncomp = d["ncomp"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.width = d["width"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.height = d["height"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.img_block0 = d["img_block0"].cast<torch::Tensor>().data_ptr<int64_t>();
  g.bpm = d["bpm"].cast<torch::Tensor>().data_ptr<int32_t>();
  g.kmap = d["kmap"].cast<torch::Tensor>().data_ptr<int32_t>();
  return g;
}

// meta: the jpeg_parse_batch dict with every tensor moved to the device.
void jpeg_decode_batch(torch::Tensor data, py::dict meta, torch::Tensor coef,
                       torch::Tensor samples, torch::Tensor out,
                       torch::Tensor out_off, torch::Tensor status) {
  TORCH_CHECK(data.is_cuda() && coef.is_cuda() && samples.is_cuda() &&
              out.is_cuda());
  JpegTables tabs = make_tables(meta);
  JpegGeom g = make_geom(meta);
  auto seg_img = meta["seg_img"].cast<torch::Tensor>();
  auto seg_pos = meta["seg_pos"].cast<torch::Tensor>();
  auto seg_end = meta["seg_end"].cast<torch::Tensor>();
  auto seg_mcu0 = meta["seg_mcu0"].cast<torch::Tensor>();
  auto seg_nmcu = meta["seg_nmcu"].cast<torch::Tensor>();
  int n_segs = (int)seg_img.numel();
  int n_imgs = (int)meta["width"].cast<torch::Tensor>().numel();
  int64_t block_total = coef.numel() / 64;
  hipStream_t stream = c10::hip::getCurrentHIPStream();

  // PSA_JPEG_NT=1 enables non-temporal coefficient stores.  MEASURED
  // WORSE (same-box A/B, r2: 485k vs 615k samples/s): the scattered
  // 4-byte stores lose L2 write-combining and hit HBM as partial lines.
  // Kept as a knob because the result is instructive; default OFF.
  static const bool use_nt = [] {
    const char* e = getenv("PSA_JPEG_NT");
    return e && e[0] == '1';
  }();
  if (use_nt)
    hipLaunchKernelGGL(jpeg_huffman_kernel<true>,
                       dim3((n_segs + 63) / 64), dim3(64), 0, stream,
                       data.data_ptr<uint8_t>(), tabs, g,
                       seg_img.data_ptr<int32_t>(),
                       seg_pos.data_ptr<int64_t>(),
                       seg_end.data_ptr<int64_t>(),
                       seg_mcu0.data_ptr<int32_t>(),
                       seg_nmcu.data_ptr<int32_t>(), coef.data_ptr<float>(),
                       status.data_ptr<int32_t>(), n_segs);
  else
    hipLaunchKernelGGL(jpeg_huffman_kernel<false>,
                       dim3((n_segs + 63) / 64), dim3(64), 0, stream,
                       data.data_ptr<uint8_t>(), tabs, g,
                       seg_img.data_ptr<int32_t>(),
                       seg_pos.data_ptr<int64_t>(),
                       seg_end.data_ptr<int64_t>(),
                       seg_mcu0.data_ptr<int32_t>(),
                       seg_nmcu.data_ptr<int32_t>(), coef.data_ptr<float>(),
                       status.data_ptr<int32_t>(), n_segs);

  hipLaunchKernelGGL(jpeg_idct_kernel,
                     dim3((unsigned)((block_total + 255) / 256)), dim3(256),
                     0, stream, coef.data_ptr<float>(), g,
                     samples.data_ptr<uint8_t>(), n_imgs, block_total);

  // color: grid.y = image, grid.x covers the largest image (the kernel is
  // grid-stride over pixels, so an estimate from the average size is fine)
  int64_t avg_bytes = n_imgs > 0 ? out.numel() / n_imgs : 0;
  int64_t est_blocks = (avg_bytes / 12 + 255) / 256;  // quad-per-thread
  int grid_x = (int)std::min<int64_t>(std::max<int64_t>(est_blocks, 1), 2048);
  hipLaunchKernelGGL(jpeg_color_kernel, dim3(grid_x, n_imgs), dim3(256), 0,
                     stream, samples.data_ptr<uint8_t>(), g,
                     out.data_ptr<uint8_t>(), out_off.data_ptr<int64_t>(),
                     n_imgs);
}

// Same pipeline with the fused color+normalize epilogue: `out` is fp32
// [n, 3, H, W]; out_off is in FLOAT elements.
void jpeg_decode_fused_batch(torch::Tensor data, py::dict meta,
                             torch::Tensor coef, torch::Tensor samples,
                             torch::Tensor out, torch::Tensor out_off,
                             torch::Tensor mean, torch::Tensor inv_std,
                             double scale, torch::Tensor status) {
  TORCH_CHECK(data.is_cuda() && coef.is_cuda() && samples.is_cuda() &&
              out.is_cuda() && out.scalar_type() == torch::kFloat32);
  TORCH_CHECK(mean.is_cuda() && inv_std.is_cuda() &&
              mean.numel() == 3 && inv_std.numel() == 3);
  JpegTables tabs = make_tables(meta);
  JpegGeom g = make_geom(meta);
  auto seg_img = meta["seg_img"].cast<torch::Tensor>();
  auto seg_pos = meta["seg_pos"].cast<torch::Tensor>();
  auto seg_end = meta["seg_end"].cast<torch::Tensor>();
  auto seg_mcu0 = meta["seg_mcu0"].cast<torch::Tensor>();
  auto seg_nmcu = meta["seg_nmcu"].cast<torch::Tensor>();
  int n_segs = (int)seg_img.numel();
  int n_imgs = (int)meta["width"].cast<torch::Tensor>().numel();
  int64_t block_total = coef.numel() / 64;
  hipStream_t stream = c10::hip::getCurrentHIPStream();

  // PSA_JPEG_NT=1 enables non-temporal coefficient stores.  MEASURED
  // WORSE (same-box A/B, r2: 485k vs 615k samples/s): the scattered
  // 4-byte stores lose L2 write-combining and hit HBM as partial lines.
  // Kept as a knob because the result is instructive; default OFF.
  static const bool use_nt = [] {
    const char* e = getenv("PSA_JPEG_NT");
    return e && e[0] == '1';
  }();
  if (use_nt)
    hipLaunchKernelGGL(jpeg_huffman_kernel<true>,
                       dim3((n_segs + 63) / 64), dim3(64), 0, stream,
                       data.data_ptr<uint8_t>(), tabs, g,
                       seg_img.data_ptr<int32_t>(),
                       seg_pos.data_ptr<int64_t>(),
                       seg_end.data_ptr<int64_t>(),
                       seg_mcu0.data_ptr<int32_t>(),
                       seg_nmcu.data_ptr<int32_t>(), coef.data_ptr<float>(),
                       status.data_ptr<int32_t>(), n_segs);
  else
    hipLaunchKernelGGL(jpeg_huffman_kernel<false>,
                       dim3((n_segs + 63) / 64), dim3(64), 0, stream,
                       data.data_ptr<uint8_t>(), tabs, g,
                       seg_img.data_ptr<int32_t>(),
                       seg_pos.data_ptr<int64_t>(),
                       seg_end.data_ptr<int64_t>(),
                       seg_mcu0.data_ptr<int32_t>(),
                       seg_nmcu.data_ptr<int32_t>(), coef.data_ptr<float>(),
                       status.data_ptr<int32_t>(), n_segs);

  hipLaunchKernelGGL(jpeg_idct_kernel,
                     dim3((unsigned)((block_total + 255) / 256)), dim3(256),
                     0, stream, coef.data_ptr<float>(), g,
                     samples.data_ptr<uint8_t>(), n_imgs, block_total);

  int64_t avg_px = n_imgs > 0 ? out.numel() / (3 * n_imgs) : 0;
  int64_t est_blocks = (avg_px / 4 + 255) / 256;
  int grid_x = (int)std::min<int64_t>(std::max<int64_t>(est_blocks, 1), 2048);
  hipLaunchKernelGGL(jpeg_color_norm_kernel, dim3(grid_x, n_imgs), dim3(256),
                     0, stream, samples.data_ptr<uint8_t>(), g,
                     out.data_ptr<float>(), out_off.data_ptr<int64_t>(),
                     mean.data_ptr<float>(), inv_std.data_ptr<float>(),
                     (float)scale, n_imgs);
}

}  // namespace psa
