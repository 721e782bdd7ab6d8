// GPU Parquet page decode: RLE/bit-packed hybrid, PLAIN byte-array offset
// extraction, unaligned variable-length gather, .npy payload location.
//
// Replaces the Arrow C++ Parquet value decoders the reference reaches through
// piece.read() (reference petastorm/arrow_reader_worker.py:358,
// py_dict_reader_worker.py:267) and the NdarrayCodec np.load
// (reference petastorm/codecs.py:155-157).
//
// Parallel structure mirrors snappy.hip: bitstream run headers are parsed by
// lane 0 of a wave and the 64 lanes expand values in parallel; page-level
// parallelism fills the 256 CUs.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace psa {

// ---------------------------------------------------------------------------
// RLE / bit-packed hybrid (Parquet spec "RLE" encoding): definition levels
// and dictionary indices.
//
//   run-header varint h: (h & 1) == 0 -> RLE run of (h >> 1) copies of a
//   ceil(bw/8)-byte LE value; (h & 1) == 1 -> (h >> 1) groups of 8
//   bit-packed values, LSB-first.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint32_t read_varint_u32(const uint8_t* p,
                                                    int64_t& pos,
                                                    int64_t end) {
  uint32_t result = 0;
  int shift = 0;
  while (pos < end && shift < 35) {
    uint8_t b = p[pos++];
    result |= (uint32_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  return result;
}

__device__ __forceinline__ uint32_t extract_bits(const uint8_t* base,
                                                 int64_t bit_off, int bw) {
  // read up to 24+8 bits spanning <=5 bytes
  int64_t byte = bit_off >> 3;
  int shift = (int)(bit_off & 7);
  uint64_t w = 0;
  for (int i = 0; i < 5; ++i) w |= (uint64_t)base[byte + i] << (8 * i);
  return (uint32_t)((w >> shift) & ((1u << bw) - 1u));
}

// One wave per stream. streams are (data, start, end, bit_width, n_values,
// out_offset) tuples.
__global__ void rle_hybrid_decode_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int64_t* __restrict__ end, const int32_t* __restrict__ bit_width,
    const int32_t* __restrict__ n_values, const int64_t* __restrict__ out_off,
    int32_t* __restrict__ out, int32_t* __restrict__ status, int n_streams) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int s = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (s >= n_streams) return;
  const int lane = lane_id();

  const int bw = bit_width[s];
  const int64_t lo = start[s], hi = end[s];
  const int32_t want = n_values[s];
  int32_t* dst = out + out_off[s];

  if (bw == 0) {
    // all values are zero (e.g. required column def levels)
    for (int32_t i = lane; i < want; i += PSA_WAVE) dst[i] = 0;
    return;
  }
  if (bw > 24) {
    if (lane == 0) status[s] = 4;  // unsupported width
    return;
  }

  int64_t pos = lo;
  int32_t produced = 0;
  while (produced < want) {
    uint32_t header = 0;
    int64_t payload = 0;
    int32_t run_len = 0, is_packed = 0;
    uint32_t rle_value = 0;
    int done = 0;
    if (lane == 0) {
      if (pos >= hi) {
        done = 1;
      } else {
        header = read_varint_u32(data, pos, hi);
        is_packed = header & 1;
        if (is_packed) {
          run_len = (int32_t)(header >> 1) * 8;
          payload = pos;               // bit-packed payload starts here
          pos += (int64_t)(header >> 1) * bw;
        } else {
          run_len = (int32_t)(header >> 1);
          int nbytes = (bw + 7) / 8;
          rle_value = 0;
          for (int i = 0; i < nbytes; ++i)
            rle_value |= (uint32_t)data[pos + i] << (8 * i);
          pos += nbytes;
        }
      }
    }
    done = wave_bcast(done);
    if (done) {
      if (lane == 0 && produced < want) status[s] = 5;  // truncated
      break;
    }
    is_packed = wave_bcast(is_packed);
    run_len = wave_bcast(run_len);
    if (is_packed) {
      payload = wave_bcast(payload);
      int32_t emit = min(run_len, want - produced);
      for (int32_t i = lane; i < emit; i += PSA_WAVE)
        dst[produced + i] =
            (int32_t)extract_bits(data, payload * 8 + (int64_t)i * bw, bw);
      produced += emit;
    } else {
      rle_value = wave_bcast(rle_value);
      int32_t emit = min(run_len, want - produced);
      for (int32_t i = lane; i < emit; i += PSA_WAVE)
        dst[produced + i] = (int32_t)rle_value;
      produced += emit;
    }
  }
}

void rle_hybrid_decode_batch(torch::Tensor data, torch::Tensor start,
                             torch::Tensor end, torch::Tensor bit_width,
                             torch::Tensor n_values, torch::Tensor out_off,
                             torch::Tensor out, torch::Tensor status) {
  int n = (int)start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rle_hybrid_decode_kernel,
                     dim3((n + WPB - 1) / WPB), dim3(WPB * PSA_WAVE), 0,
                     stream, data.data_ptr<uint8_t>(),
                     start.data_ptr<int64_t>(), end.data_ptr<int64_t>(),
                     bit_width.data_ptr<int32_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_off.data_ptr<int64_t>(), out.data_ptr<int32_t>(),
                     status.data_ptr<int32_t>(), n);
}

// ---------------------------------------------------------------------------
// PLAIN byte-array pages: [u32 len][bytes]... -> absolute (offset, length)
// per value.  Downstream kernels (jpeg/npy/inflate) read values in place —
// no copy of the blob bytes.
// ---------------------------------------------------------------------------

__global__ void byte_array_offsets_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int64_t* __restrict__ end, const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_off, int64_t* __restrict__ val_off,
    int32_t* __restrict__ val_len, int32_t* __restrict__ status,
    int n_pages) {
  // lane 0 of a wave scans its page (the scan per value is one u32 read);
  // other lanes idle — page parallelism dominates.
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages || lane_id() != 0) return;
  int64_t pos = start[page];
  const int64_t hi = end[page];
  const int32_t want = n_values[page];
  int64_t* o = val_off + out_off[page];
  int32_t* l = val_len + out_off[page];
  for (int32_t i = 0; i < want; ++i) {
    if (pos + 4 > hi) { status[page] = 6; return; }
    uint32_t len = load_u32_unaligned(data + pos);
    pos += 4;
    if (pos + len > hi) { status[page] = 6; return; }
    o[i] = pos;
    l[i] = (int32_t)len;
    pos += len;
  }
}

void byte_array_offsets_batch(torch::Tensor data, torch::Tensor start,
                              torch::Tensor end, torch::Tensor n_values,
                              torch::Tensor out_off, torch::Tensor val_off,
                              torch::Tensor val_len, torch::Tensor status) {
  int n = (int)start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(byte_array_offsets_kernel,
                     dim3((n + WPB - 1) / WPB), dim3(WPB * PSA_WAVE), 0,
                     stream, data.data_ptr<uint8_t>(),
                     start.data_ptr<int64_t>(), end.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_off.data_ptr<int64_t>(),
                     val_off.data_ptr<int64_t>(),
                     val_len.data_ptr<int32_t>(),
                     status.data_ptr<int32_t>(), n);
}

// ---------------------------------------------------------------------------
// Unaligned variable-length gather: copy item i's bytes
// src[src_off[i] .. +len[i]) -> dst[dst_off[i] ..).
//
// The src side is arbitrarily aligned (values sit mid-page); the copy uses
// aligned u32 loads + funnel shift so each lane still moves 4B per
// instruction (guide §6 Guideline 13: never scalar-byte a memory-bound loop).
// One block per item, threads stride the length.
// ---------------------------------------------------------------------------

__global__ void varlen_gather_kernel(
    const uint8_t* __restrict__ src, const int64_t* __restrict__ src_off,
    const int64_t* __restrict__ len64, uint8_t* __restrict__ dst,
    const int64_t* __restrict__ dst_off, int n_items) {
  for (int item = blockIdx.x; item < n_items; item += gridDim.x) {
    const int64_t n = len64[item];
    const uint8_t* s = src + src_off[item];
    uint8_t* d = dst + dst_off[item];
    // head: bytes until dst is 4-aligned
    int64_t head = min(n, (int64_t)((4 - ((uintptr_t)d & 3)) & 3));
    for (int64_t i = threadIdx.x; i < head; i += blockDim.x) d[i] = s[i];
    s += head; d += head;
    int64_t body = (n - head) & ~(int64_t)3;
    // aligned-u32 + funnel shift over the body
    const int shift = (int)((uintptr_t)s & 3) * 8;
    const uint32_t* s4 = (const uint32_t*)((uintptr_t)s & ~(uintptr_t)3);
    uint32_t* d4 = (uint32_t*)d;
    const int64_t words = body >> 2;
    if (shift == 0) {
      for (int64_t w = threadIdx.x; w < words; w += blockDim.x)
        d4[w] = s4[w];
    } else {
      for (int64_t w = threadIdx.x; w < words; w += blockDim.x) {
        uint32_t lo = s4[w], hi = s4[w + 1];
        d4[w] = (lo >> shift) | (hi << (32 - shift));
      }
    }
    // tail
    for (int64_t i = body + threadIdx.x; i < n - head; i += blockDim.x)
      d[i] = s[i];
  }
}

void varlen_gather(torch::Tensor src, torch::Tensor src_off,
                   torch::Tensor lengths, torch::Tensor dst,
                   torch::Tensor dst_off) {
  int n = (int)src_off.numel();
  if (!n) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  int blocks = n < 2048 ? n : 2048;
  hipLaunchKernelGGL(varlen_gather_kernel, dim3(blocks), dim3(256), 0,
                     stream, src.data_ptr<uint8_t>(),
                     src_off.data_ptr<int64_t>(),
                     lengths.data_ptr<int64_t>(), dst.data_ptr<uint8_t>(),
                     dst_off.data_ptr<int64_t>(), n);
}

// ---------------------------------------------------------------------------
// .npy payload location: value bytes are a full .npy container
// (reference NdarrayCodec, petastorm/codecs.py:133-171).  Header:
//   \x93NUMPY <ver_major> <ver_minor> <u16 hlen> <hlen dict bytes> payload
// This kernel turns (value offset, value length) into (payload offset,
// payload length) so varlen_gather can assemble the dense batch tensor.
// ---------------------------------------------------------------------------

__global__ void npy_payload_offsets_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ val_off,
    const int32_t* __restrict__ val_len, int64_t* __restrict__ pay_off,
    int64_t* __restrict__ pay_len, int32_t* __restrict__ status, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const uint8_t* p = data + val_off[i];
  if (val_len[i] < 10 || p[0] != 0x93 || p[1] != 'N' || p[2] != 'U') {
    status[0] = 7;  // not an npy container
    pay_off[i] = val_off[i];
    pay_len[i] = 0;
    return;
  }
  int64_t hlen;
  int64_t hdr;
  if (p[6] == 1) {           // version 1.0: u16 header length
    hlen = (int64_t)p[8] | ((int64_t)p[9] << 8);
    hdr = 10 + hlen;
  } else {                   // version 2.0+: u32 header length
    if (val_len[i] < 12) {
      status[0] = 7;
      pay_off[i] = val_off[i];
      pay_len[i] = 0;
      return;
    }
    hlen = (int64_t)p[8] | ((int64_t)p[9] << 8) | ((int64_t)p[10] << 16) |
           ((int64_t)p[11] << 24);
    hdr = 12 + hlen;
  }
  if (hdr > (int64_t)val_len[i]) {  // corrupt header claims > payload
    status[0] = 7;
    pay_off[i] = val_off[i];
    pay_len[i] = 0;
    return;
  }
  pay_off[i] = val_off[i] + hdr;
  pay_len[i] = (int64_t)val_len[i] - hdr;
}

void npy_payload_offsets(torch::Tensor data, torch::Tensor val_off,
                         torch::Tensor val_len, torch::Tensor pay_off,
                         torch::Tensor pay_len, torch::Tensor status) {
  int n = (int)val_off.numel();
  if (!n) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(npy_payload_offsets_kernel,
                     dim3((n + 255) / 256), dim3(256), 0, stream,
                     data.data_ptr<uint8_t>(), val_off.data_ptr<int64_t>(),
                     val_len.data_ptr<int32_t>(), pay_off.data_ptr<int64_t>(),
                     pay_len.data_ptr<int64_t>(),
                     status.data_ptr<int32_t>(), n);
}

}  // namespace psa

namespace psa {

// ---------------------------------------------------------------------------
// Fused PLAIN fixed-width data-page decode: definition levels + non-null
// prefix scan + value scatter in ONE kernel, one wave per page.
//
// Replaces the multi-launch path (rle levels -> cumsum -> masked scatter ->
// varlen gather) which cost one host sync per page on nullable columns
// (pyarrow writes every top-level column as OPTIONAL, so this is the common
// case even for null-free data).
//
// Page payload layout (v1 data page, PLAIN values):
//   [u32 dl_len][def-level RLE hybrid, bw=1][values...]   when max_def == 1
//   [values...]                                            when max_def == 0
//
// Wave algorithm for bit-packed runs: each chunk of 64 levels ballots its
// valid bits; lane's value index = running val_cursor + popc(mask & lanes
// below me).  RLE runs copy (valid) or fill (null) in parallel.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void copy_elem(uint8_t* dst, const uint8_t* src,
                                          int esize) {
  if (esize == 8) {
    uint32_t lo = load_u32_unaligned(src);
    uint32_t hi = load_u32_unaligned(src + 4);
    // dst is naturally aligned (column base + row*esize)
    *(uint32_t*)dst = lo;
    *(uint32_t*)(dst + 4) = hi;
  } else if (esize == 4) {
    *(uint32_t*)dst = load_u32_unaligned(src);
  } else if (esize == 12) {  // INT96 (legacy Spark timestamps)
    *(uint32_t*)dst = load_u32_unaligned(src);
    *(uint32_t*)(dst + 4) = load_u32_unaligned(src + 4);
    *(uint32_t*)(dst + 8) = load_u32_unaligned(src + 8);
  } else if (esize == 2) {
    dst[0] = src[0];
    dst[1] = src[1];
  } else {
    dst[0] = src[0];
  }
}

__device__ __forceinline__ void fill_elem(uint8_t* dst, uint64_t pattern,
                                          int esize) {
  for (int b = 0; b < esize; ++b) dst[b] = (uint8_t)(pattern >> (8 * (b & 7)));
}

// def_mode: 0 = required (no levels); 1 = v1 page (u32 length prefix then
// RLE levels, then values, all at payload_start); 2 = v2 page (levels live
// in def_buf[def_start..+def_len] — uncompressed — and payload_start points
// directly at the values).
__global__ void plain_fixed_decode_kernel(
    const uint8_t* __restrict__ page_buf,
    const int64_t* __restrict__ payload_start,  // per page
    const int64_t* __restrict__ payload_end,
    const int32_t* __restrict__ n_values,       // rows in page (incl nulls)
    const int64_t* __restrict__ row0,           // first output row of page
    int32_t def_mode, int32_t esize, uint64_t fill_pattern,
    const uint8_t* __restrict__ def_buf,        // v2 levels (may == page_buf)
    const int64_t* __restrict__ def_start_arr,  // v2: per-page level offset
    const int64_t* __restrict__ def_len_arr,    // v2: per-page level bytes
    uint8_t* __restrict__ out,                  // column base (row-major)
    uint8_t* __restrict__ valid_out,            // [total rows] or nullptr
    int32_t* __restrict__ status, int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  const int lane = lane_id();
  const int32_t want = n_values[page];
  uint8_t* out_base = out + row0[page] * esize;
  uint8_t* vbase = valid_out ? valid_out + row0[page] : nullptr;

  int64_t pos = payload_start[page];
  const int64_t hi = payload_end[page];

  if (!def_mode) {
    // required column: straight parallel copy
    const uint8_t* src = page_buf + pos;
    for (int32_t i = lane; i < want; i += PSA_WAVE)
      copy_elem(out_base + (int64_t)i * esize, src + (int64_t)i * esize,
                esize);
    if (vbase)
      for (int32_t i = lane; i < want; i += PSA_WAVE) vbase[i] = 1;
    return;
  }

  // definition levels: RLE hybrid at bit width 1.
  // v1: u32 length prefix at payload start, values follow the levels;
  // v2: levels live in def_buf at an explicit per-page range.
  const uint8_t* lev = page_buf;
  int64_t def_pos, def_end;
  const uint8_t* values;
  if (def_mode == 1) {
    uint32_t dl_len = 0;
    if (lane == 0) dl_len = load_u32_unaligned(page_buf + pos);
    dl_len = wave_bcast(dl_len);
    def_pos = pos + 4;
    def_end = def_pos + dl_len;
    values = page_buf + def_end;
  } else {
    lev = def_buf;
    def_pos = def_start_arr[page];
    def_end = def_pos + def_len_arr[page];
    values = page_buf + pos;
  }

  int32_t row_cursor = 0;   // rows emitted
  int32_t val_cursor = 0;   // non-null values consumed
  while (row_cursor < want) {
    uint32_t header = 0;
    int is_packed = 0, run_len = 0;
    uint32_t rle_value = 0;
    int64_t payload = 0;
    int done = 0;
    if (lane == 0) {
      if (def_pos >= def_end) {
        done = 1;
      } else {
        header = read_varint_u32(lev, def_pos, def_end);
        is_packed = header & 1;
        if (is_packed) {
          run_len = (int32_t)(header >> 1) * 8;
          payload = def_pos;
          def_pos += (int64_t)(header >> 1);  // bw=1: one byte per group
        } else {
          run_len = (int32_t)(header >> 1);
          rle_value = lev[def_pos];
          def_pos += 1;
        }
      }
    }
    done = wave_bcast(done);
    if (done) { if (lane == 0) status[page] = 8; return; }
    is_packed = wave_bcast(is_packed);
    run_len = wave_bcast(run_len);
    int32_t emit = min(run_len, want - row_cursor);
    if (!is_packed) {
      rle_value = wave_bcast(rle_value);
      if (rle_value) {
        const uint8_t* src = values + (int64_t)val_cursor * esize;
        for (int32_t i = lane; i < emit; i += PSA_WAVE)
          copy_elem(out_base + (int64_t)(row_cursor + i) * esize,
                    src + (int64_t)i * esize, esize);
        val_cursor += emit;
      } else {
        for (int32_t i = lane; i < emit; i += PSA_WAVE)
          fill_elem(out_base + (int64_t)(row_cursor + i) * esize,
                    fill_pattern, esize);
      }
      if (vbase)
        for (int32_t i = lane; i < emit; i += PSA_WAVE)
          vbase[row_cursor + i] = (uint8_t)(rle_value ? 1 : 0);
      row_cursor += emit;
    } else {
      payload = wave_bcast(payload);
      // process 64 levels per iteration: ballot + prefix popcount
      for (int32_t base = 0; base < emit; base += PSA_WAVE) {
        int32_t i = base + lane;
        int my_bit = 0;
        if (i < emit) {
          int64_t bit = (int64_t)i;  // bit index within this run
          my_bit = (lev[payload + (bit >> 3)] >> (bit & 7)) & 1;
        }
        unsigned long long mask = __ballot(my_bit != 0);
        // lanes-below-me mask; lane 63 special-cased ((1ull<<64) is UB)
        unsigned long long below = mask & ((lane < 63)
                                   ? ((1ull << lane) - 1ull)
                                   : 0x7FFFFFFFFFFFFFFFull);
        int my_val_idx = val_cursor + (int)__popcll(below);
        if (i < emit) {
          uint8_t* dst = out_base + (int64_t)(row_cursor + i) * esize;
          if (my_bit) {
            copy_elem(dst, values + (int64_t)my_val_idx * esize, esize);
          } else {
            fill_elem(dst, fill_pattern, esize);
          }
          if (vbase) vbase[row_cursor + i] = (uint8_t)my_bit;
        }
        // lanes past `emit` contributed my_bit=0, so mask is already clean
        val_cursor += (int)__popcll(mask);
      }
      row_cursor += emit;
    }
    (void)hi;
  }
}

void plain_fixed_decode_batch(torch::Tensor page_buf,
                              torch::Tensor payload_start,
                              torch::Tensor payload_end,
                              torch::Tensor n_values, torch::Tensor row0,
                              int64_t def_mode, int64_t esize,
                              int64_t fill_pattern, torch::Tensor def_buf,
                              torch::Tensor def_start, torch::Tensor def_len,
                              torch::Tensor out,
                              torch::Tensor valid_out, torch::Tensor status) {
  int n = (int)payload_start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  uint8_t* vptr = valid_out.numel() ? valid_out.data_ptr<uint8_t>() : nullptr;
  const uint8_t* dbp = def_buf.numel() ? def_buf.data_ptr<uint8_t>()
                                       : page_buf.data_ptr<uint8_t>();
  const int64_t* dsp = def_start.numel() ? def_start.data_ptr<int64_t>()
                                         : payload_start.data_ptr<int64_t>();
  const int64_t* dlp = def_len.numel() ? def_len.data_ptr<int64_t>()
                                       : payload_start.data_ptr<int64_t>();
  hipLaunchKernelGGL(plain_fixed_decode_kernel, dim3((n + WPB - 1) / WPB),
                     dim3(WPB * PSA_WAVE), 0, stream,
                     page_buf.data_ptr<uint8_t>(),
                     payload_start.data_ptr<int64_t>(),
                     payload_end.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     row0.data_ptr<int64_t>(), (int32_t)def_mode,
                     (int32_t)esize, (uint64_t)fill_pattern, dbp, dsp, dlp,
                     out.data_ptr<uint8_t>(), vptr,
                     status.data_ptr<int32_t>(), n);
}


// ---------------------------------------------------------------------------
// DELTA_BINARY_PACKED (Parquet spec encodings.md): header = <block_size>
// <miniblocks_per_block> <total_count> <first_value zigzag>, then blocks of
// <min_delta zigzag> <1 bitwidth byte per miniblock> <bit-packed deltas>.
// value[i+1] = value[i] + min_delta + delta[i].
//
// One wave per page: lane 0 walks the varints; the wave cooperatively
// extracts 64 deltas at a time and resolves the prefix dependency with a
// shfl inclusive scan + running carry — the CDNA4-idiomatic form of a
// sequential dependency.  Shared by the int-column kernel, the
// delta-length and the delta-byte-array (prefix) string kernels.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t read_varint_u64(const uint8_t* p,
                                                    int64_t& pos,
                                                    int64_t end) {
  uint64_t result = 0;
  int shift = 0;
  while (pos < end && shift < 70) {
    uint8_t b = p[pos++];
    result |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  return result;
}

__device__ __forceinline__ int64_t zigzag64(uint64_t u) {
  return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
}

__device__ __forceinline__ uint64_t extract_bits64(const uint8_t* base,
                                                   int64_t bit_off, int bw) {
  // up to 64 bits spanning <= 9 bytes
  int64_t byte = bit_off >> 3;
  int shift = (int)(bit_off & 7);
  uint64_t lo = 0;
  for (int i = 0; i < 8; ++i) lo |= (uint64_t)base[byte + i] << (8 * i);
  uint64_t v = lo >> shift;
  if (shift) v |= (uint64_t)base[byte + 8] << (64 - shift);
  if (bw < 64) v &= (~0ull) >> (64 - bw);
  return v;
}

// wave-wide inclusive prefix sum (int64), Hillis-Steele over 64 lanes
__device__ __forceinline__ int64_t wave_incl_scan_i64(int64_t v) {
  const int lane = lane_id();
  for (int d = 1; d < PSA_WAVE; d <<= 1) {
    int64_t up = __shfl_up((long long)v, d, PSA_WAVE);
    if (lane >= d) v += up;
  }
  return v;
}

// Wave-cooperative decode of one DELTA_BINARY_PACKED stream into `out`
// (first `want` values).  Returns the stream end position (where the
// following section, e.g. delta-length byte data, begins), or -1 on
// malformed input (err code written via *err).
template <typename OutT>
__device__ int64_t delta_ints_decode_wave(const uint8_t* __restrict__ base,
                                          int64_t pos, int64_t pend,
                                          int32_t want,
                                          OutT* __restrict__ out,
                                          int32_t* err) {
  const int lane = lane_id();
  uint64_t block_size = 0, mbs_per_block = 0, total = 0;
  int64_t first = 0;
  if (lane == 0) {
    block_size = read_varint_u64(base, pos, pend);
    mbs_per_block = read_varint_u64(base, pos, pend);
    total = read_varint_u64(base, pos, pend);
    first = zigzag64(read_varint_u64(base, pos, pend));
  }
  block_size = wave_bcast((unsigned long long)block_size);
  mbs_per_block = wave_bcast((unsigned long long)mbs_per_block);
  total = wave_bcast((unsigned long long)total);
  first = wave_bcast((long long)first);
  pos = wave_bcast((long long)pos);
  if (block_size == 0 || mbs_per_block == 0 ||
      block_size % (mbs_per_block * 8) != 0) {
    if (lane == 0) *err = 40;
    return -1;
  }
  const int vpm = (int)(block_size / mbs_per_block);

  if (lane == 0 && want > 0) out[0] = (OutT)first;
  int64_t produced = 1;
  int64_t prev = first;

  while (produced < want && produced < (int64_t)total) {
    int64_t min_delta = 0;
    if (lane == 0) min_delta = zigzag64(read_varint_u64(base, pos, pend));
    min_delta = wave_bcast((long long)min_delta);
    pos = wave_bcast((long long)pos);
    int64_t bw_pos = pos;               // one bitwidth byte per miniblock
    pos += mbs_per_block;
    if (pos > pend) { if (lane == 0) *err = 41; return -1; }
    for (uint64_t mb = 0; mb < mbs_per_block; ++mb) {
      int bw = base[bw_pos + mb];
      if (bw > 64) { if (lane == 0) *err = 42; return -1; }
      const int64_t mb_bits = (int64_t)vpm * bw;
      if (produced >= want || produced >= (int64_t)total) {
        pos += (mb_bits + 7) >> 3;      // skip remaining miniblocks
        continue;
      }
      for (int off = 0; off < vpm; off += PSA_WAVE) {
        int64_t my_delta = 0;
        int j = off + lane;
        if (j < vpm && bw > 0)
          my_delta = (int64_t)extract_bits64(base + pos, (int64_t)j * bw,
                                             bw);
        // lanes beyond the miniblock contribute NOTHING to the scan
        int64_t step = (j < vpm) ? (my_delta + min_delta) : 0;
        int64_t incl = wave_incl_scan_i64(step);
        int64_t v = prev + incl;
        if (j < vpm && produced + j < want) out[produced + j] = (OutT)v;
        // carry: value at the LAST VALID lane of this pass
        int last = min(PSA_WAVE, vpm - off) - 1;
        prev = wave_bcast_from(v, last);
      }
      produced += vpm;
      pos += (mb_bits + 7) >> 3;
      if (pos > pend + 8) { if (lane == 0) *err = 43; return -1; }
    }
  }
  return pos;
}

__global__ void delta_binary_packed_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int64_t* __restrict__ end, const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_off, uint8_t* __restrict__ out,
    int32_t esize, int32_t* __restrict__ status, int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  uint8_t* dst = out + out_off[page] * esize;
  if (esize == 8)
    delta_ints_decode_wave<int64_t>(data, start[page], end[page],
                                    n_values[page], (int64_t*)dst,
                                    status + page);
  else
    delta_ints_decode_wave<int32_t>(data, start[page], end[page],
                                    n_values[page], (int32_t*)dst,
                                    status + page);
}

void delta_binary_packed_batch(torch::Tensor page_buf, torch::Tensor start,
                               torch::Tensor end, torch::Tensor n_values,
                               torch::Tensor out_off, torch::Tensor out,
                               int64_t esize, torch::Tensor status) {
  int n = (int)start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(delta_binary_packed_kernel, dim3((n + WPB - 1) / WPB),
                     dim3(WPB * PSA_WAVE), 0, stream,
                     page_buf.data_ptr<uint8_t>(), start.data_ptr<int64_t>(),
                     end.data_ptr<int64_t>(), n_values.data_ptr<int32_t>(),
                     out_off.data_ptr<int64_t>(), out.data_ptr<uint8_t>(),
                     (int32_t)esize, status.data_ptr<int32_t>(), n);
}

// DELTA_LENGTH_BYTE_ARRAY: a DELTA_BINARY_PACKED int32 length block, then
// the concatenated value bytes.  Emits per-value (offset, length) tables
// pointing into the page (same contract as byte_array_offsets_batch).
__global__ void delta_length_byte_array_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int64_t* __restrict__ end, const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_idx,
    int64_t* __restrict__ val_off, int32_t* __restrict__ val_len,
    int32_t* __restrict__ status, int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  const int lane = lane_id();
  const int32_t want = n_values[page];
  int64_t* po = val_off + out_idx[page];
  int32_t* pl = val_len + out_idx[page];
  int64_t bytes0 = delta_ints_decode_wave<int32_t>(
      data, start[page], end[page], want, pl, status + page);
  if (bytes0 < 0) return;
  __threadfence();  // lane 0 re-reads lengths other lanes stored
  if (lane == 0) {
    int64_t byte_pos = bytes0;
    for (int32_t i = 0; i < want; ++i) {
      if (pl[i] < 0) { status[page] = 49; return; }  // corrupt length:
      po[i] = byte_pos;        // negatives would send gathers out of bounds
      byte_pos += pl[i];
    }
    if (byte_pos > end[page]) status[page] = 49;
  }
}

void delta_length_byte_array_batch(torch::Tensor page_buf,
                                   torch::Tensor start, torch::Tensor end,
                                   torch::Tensor n_values,
                                   torch::Tensor out_idx,
                                   torch::Tensor val_off,
                                   torch::Tensor val_len,
                                   torch::Tensor status) {
  int n = (int)start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(delta_length_byte_array_kernel,
                     dim3((n + WPB - 1) / WPB), dim3(WPB * PSA_WAVE), 0,
                     stream, page_buf.data_ptr<uint8_t>(),
                     start.data_ptr<int64_t>(), end.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_idx.data_ptr<int64_t>(),
                     val_off.data_ptr<int64_t>(),
                     val_len.data_ptr<int32_t>(),
                     status.data_ptr<int32_t>(), n);
}

// ---------------------------------------------------------------------------
// DELTA_BYTE_ARRAY (incremental/front-coded strings — Spark v2 writers):
// <prefix lengths: delta block> <suffix lengths+bytes: DELTA_LENGTH>.
// value[i] = value[i-1][:prefix_len[i]] + suffix[i].  Two kernels: the
// LENGTHS pass decodes prefix/suffix lengths (host then sizes the exact
// output and cumsums offsets); the RECONSTRUCT pass materializes values
// (sequential prefix dependency per page, wave-parallel byte copies).
// ---------------------------------------------------------------------------
__global__ void delta_byte_array_lengths_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int64_t* __restrict__ end, const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_idx,
    int32_t* __restrict__ pre, int32_t* __restrict__ sfx,
    int64_t* __restrict__ suf_data_pos, int32_t* __restrict__ status,
    int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  const int32_t want = n_values[page];
  int64_t suf_pos = delta_ints_decode_wave<int32_t>(
      data, start[page], end[page], want, pre + out_idx[page],
      status + page);
  if (suf_pos < 0) return;
  int64_t bytes0 = delta_ints_decode_wave<int32_t>(
      data, suf_pos, end[page], want, sfx + out_idx[page], status + page);
  if (bytes0 < 0) return;
  if (lane_id() == 0) suf_data_pos[page] = bytes0;
}

__global__ void delta_byte_array_reconstruct_kernel(
    const uint8_t* __restrict__ data, const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_idx,
    const int32_t* __restrict__ pre, const int32_t* __restrict__ sfx,
    const int64_t* __restrict__ suf_data_pos,
    const int64_t* __restrict__ val_off,   // per value, into `out`
    uint8_t* __restrict__ out, int64_t out_cap,
    int32_t* __restrict__ status, int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  const int lane = lane_id();
  const int32_t want = n_values[page];
  const int64_t base_idx = out_idx[page];
  int64_t src = suf_data_pos[page];
  for (int32_t i = 0; i < want; ++i) {
    int32_t p = pre[base_idx + i], sl = sfx[base_idx + i];
    int64_t dst = val_off[base_idx + i];
    if (p < 0 || sl < 0 || (i == 0 && p != 0) || dst + p + sl > out_cap) {
      if (lane == 0) status[page] = 50;
      return;
    }
    if (p > 0) {
      int64_t prev = val_off[base_idx + i - 1];
      for (int b = lane; b < p; b += PSA_WAVE) out[dst + b] = out[prev + b];
    }
    for (int b = lane; b < sl; b += PSA_WAVE)
      out[dst + p + b] = data[src + b];
    src += sl;
    // iteration i+1's prefix copy reads THIS value's bytes: wait for the
    // wave's own stores (no cross-wave traffic; vmcnt wait only)
    __threadfence();
  }
}

void delta_byte_array_lengths_batch(torch::Tensor page_buf,
                                    torch::Tensor start, torch::Tensor end,
                                    torch::Tensor n_values,
                                    torch::Tensor out_idx,
                                    torch::Tensor pre, torch::Tensor sfx,
                                    torch::Tensor suf_data_pos,
                                    torch::Tensor status) {
  int n = (int)start.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(delta_byte_array_lengths_kernel,
                     dim3((n + WPB - 1) / WPB), dim3(WPB * PSA_WAVE), 0,
                     stream, page_buf.data_ptr<uint8_t>(),
                     start.data_ptr<int64_t>(), end.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_idx.data_ptr<int64_t>(), pre.data_ptr<int32_t>(),
                     sfx.data_ptr<int32_t>(),
                     suf_data_pos.data_ptr<int64_t>(),
                     status.data_ptr<int32_t>(), n);
}

void delta_byte_array_reconstruct_batch(
    torch::Tensor page_buf, torch::Tensor n_values, torch::Tensor out_idx,
    torch::Tensor pre, torch::Tensor sfx, torch::Tensor suf_data_pos,
    torch::Tensor val_off, torch::Tensor out, torch::Tensor status) {
  int n = (int)n_values.numel();
  if (!n) return;
  const int WPB = 4;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(delta_byte_array_reconstruct_kernel,
                     dim3((n + WPB - 1) / WPB), dim3(WPB * PSA_WAVE), 0,
                     stream, page_buf.data_ptr<uint8_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_idx.data_ptr<int64_t>(), pre.data_ptr<int32_t>(),
                     sfx.data_ptr<int32_t>(),
                     suf_data_pos.data_ptr<int64_t>(),
                     val_off.data_ptr<int64_t>(), out.data_ptr<uint8_t>(),
                     out.numel(), status.data_ptr<int32_t>(), n);
}

// ---------------------------------------------------------------------------
// BYTE_STREAM_SPLIT: K byte planes (plane j = byte j of every value) —
// de-interleave.  Trivially parallel grid-stride over values.
// ---------------------------------------------------------------------------
__global__ void byte_stream_split_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_off, uint8_t* __restrict__ out,
    int32_t esize, int n_pages) {
  const int page = blockIdx.y;
  if (page >= n_pages) return;
  const int32_t n = n_values[page];
  const uint8_t* src = data + start[page];
  uint8_t* dst = out + out_off[page] * esize;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    for (int j = 0; j < esize; ++j)
      dst[i * esize + j] = src[(int64_t)j * n + i];
}

void byte_stream_split_batch(torch::Tensor page_buf, torch::Tensor start,
                             torch::Tensor n_values, torch::Tensor out_off,
                             torch::Tensor out, int64_t esize) {
  int n = (int)start.numel();
  if (!n) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(byte_stream_split_kernel, dim3(64, n), dim3(256), 0,
                     stream, page_buf.data_ptr<uint8_t>(),
                     start.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_off.data_ptr<int64_t>(), out.data_ptr<uint8_t>(),
                     (int32_t)esize, n);
}

// ---------------------------------------------------------------------------
// PLAIN BOOLEAN: bit-packed LSB-first -> one uint8 per value.
// ---------------------------------------------------------------------------
__global__ void bool_unpack_kernel(
    const uint8_t* __restrict__ data, const int64_t* __restrict__ start,
    const int32_t* __restrict__ n_values,
    const int64_t* __restrict__ out_off, uint8_t* __restrict__ out,
    int n_pages) {
  const int page = blockIdx.y;
  if (page >= n_pages) return;
  const int32_t n = n_values[page];
  const uint8_t* src = data + start[page];
  uint8_t* dst = out + out_off[page];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = (src[i >> 3] >> (i & 7)) & 1;
}

void bool_unpack_batch(torch::Tensor page_buf, torch::Tensor start,
                       torch::Tensor n_values, torch::Tensor out_off,
                       torch::Tensor out) {
  int n = (int)start.numel();
  if (!n) return;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(bool_unpack_kernel, dim3(16, n), dim3(256), 0, stream,
                     page_buf.data_ptr<uint8_t>(), start.data_ptr<int64_t>(),
                     n_values.data_ptr<int32_t>(),
                     out_off.data_ptr<int64_t>(), out.data_ptr<uint8_t>(),
                     n);
}

}  // namespace psa
