// GPU Snappy decompression of Parquet pages.
//
// Replaces the Arrow C++ Snappy decoder the reference calls through
// piece.read() (reference petastorm/arrow_reader_worker.py:358).
//
// Parallelization: Snappy's tag stream is inherently sequential, but the
// *payload* movement is not.  One 64-lane wave per compressed page: lane 0
// walks the tag stream and broadcasts (op, src, dst, len, offset) descriptors
// with __shfl; all 64 lanes execute the byte movement in parallel.
//
//  * literal runs: parallel copy (lanes stride the run)
//  * back-references: out[d+i] = out[d - off + (i % off)] — every read lands
//    in the already-complete region [d-off, d), so the copy is order-free
//    even when off < len (run replication)
//
// Page-level parallelism fills the chip: grid = number of pages in flight
// (a row-group batch typically carries hundreds of pages x WAVES_PER_BLOCK).
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace psa {

// one descriptor broadcast from lane 0
struct SnappyOp {
  int64_t src;     // literal: offset into comp stream; copy: back-offset
  int64_t dst;
  int32_t len;
  int32_t is_copy;
};

__device__ __forceinline__ int64_t read_varint(const uint8_t* p, int64_t& pos,
                                               int64_t end) {
  int64_t result = 0;
  int shift = 0;
  while (pos < end && shift < 35) {
    uint8_t b = p[pos++];
    result |= (int64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) return result;
    shift += 7;
  }
  return -1;
}

__global__ void snappy_decompress_kernel(
    const uint8_t* __restrict__ comp,
    const int64_t* __restrict__ comp_start,
    const int64_t* __restrict__ comp_end,
    uint8_t* __restrict__ out, const int64_t* __restrict__ out_off,
    const int64_t* __restrict__ out_len,
    int32_t* __restrict__ status, int n_pages) {
  const int waves_per_block = blockDim.x / PSA_WAVE;
  const int page = blockIdx.x * waves_per_block + (threadIdx.x / PSA_WAVE);
  if (page >= n_pages) return;
  const int lane = lane_id();

  const uint8_t* in = comp + comp_start[page];
  const int64_t in_len = comp_end[page] - comp_start[page];
  uint8_t* dst_base = out + out_off[page];
  const int64_t expected = out_len[page];

  // Lane 0 parses the tag stream through a register FIFO: one 8-byte
  // unaligned global load refills up to 8 tag bytes, so dependent
  // ~100-cycle global byte loads happen once per 8 consumed bytes instead
  // of per byte (profiled 10ms/page without this — the parse is the
  // bottleneck on poorly-compressing data with many small ops).
  int64_t in_pos = 0, out_pos = 0;
  int64_t total = 0;
  uint64_t fifo = 0;
  int fifo_n = 0;        // valid bytes in fifo (low bytes first)
  int64_t fifo_pos = 0;  // stream position of fifo byte 0

#define FIFO_REFILL(need)                                              \
  if (fifo_n < (need)) {                                               \
    fifo_pos = in_pos;                                                 \
    uint64_t w = 0;                                                    \
    int64_t avail = in_len - in_pos;                                   \
    if (avail >= 8) {                                                  \
      w = (uint64_t)load_u32_unaligned(in + in_pos) |                  \
          ((uint64_t)load_u32_unaligned(in + in_pos + 4) << 32);       \
      fifo_n = 8;                                                      \
    } else {                                                           \
      fifo_n = (int)(avail > 0 ? avail : 0);                           \
      for (int z = 0; z < fifo_n; ++z)                                 \
        w |= (uint64_t)in[in_pos + z] << (8 * z);                      \
    }                                                                  \
    fifo = w;                                                          \
  }

#define FIFO_TAKE(nbytes, out_v)                                       \
  do {                                                                 \
    out_v = fifo & ((nbytes) >= 8 ? ~0ull                              \
                                  : ((1ull << (8 * (nbytes))) - 1ull));\
    fifo >>= 8 * (nbytes);                                             \
    fifo_n -= (nbytes);                                                \
    in_pos += (nbytes);                                                \
  } while (0)

  if (lane == 0) {
    total = read_varint(in, in_pos, in_len);
  }
  total = wave_bcast(total);
  in_pos = wave_bcast(in_pos);
  if (total != expected) {
    if (lane == 0) status[page] = 1;  // length mismatch
    return;
  }

  while (true) {
    SnappyOp op;
    int done = 0;
    if (lane == 0) {
      if (out_pos >= total || in_pos >= in_len) {
        done = 1;
      } else {
        FIFO_REFILL(5);
        uint64_t tagw;
        uint64_t tag = fifo & 0xFF;
        switch (tag & 3) {
          case 0: {  // literal
            int64_t len = (int64_t)(tag >> 2) + 1;
            if (len > 60) {
              int n_extra = (int)(len - 60);  // 1..4 extra length bytes
              FIFO_TAKE(1 + n_extra, tagw);
              len = (int64_t)(tagw >> 8) + 1;
            } else {
              FIFO_TAKE(1, tagw);
            }
            op.is_copy = 0;
            op.src = in_pos;
            op.dst = out_pos;
            op.len = (int32_t)len;
            in_pos += len;
            fifo_n = 0;  // literal bytes skipped: invalidate fifo
            out_pos += len;
            break;
          }
          case 1: {  // copy, 1-byte offset
            FIFO_TAKE(2, tagw);
            int32_t len = (int32_t)((tag >> 2) & 0x7) + 4;
            int64_t off = (int64_t)((tag >> 5) << 8) | ((tagw >> 8) & 0xFF);
            op.is_copy = 1;
            op.src = off;
            op.dst = out_pos;
            op.len = len;
            out_pos += len;
            break;
          }
          case 2: {  // copy, 2-byte offset
            FIFO_TAKE(3, tagw);
            int32_t len = (int32_t)(tag >> 2) + 1;
            int64_t off = (int64_t)((tagw >> 8) & 0xFFFF);
            op.is_copy = 1;
            op.src = off;
            op.dst = out_pos;
            op.len = len;
            out_pos += len;
            break;
          }
          default: {  // copy, 4-byte offset
            FIFO_TAKE(5, tagw);
            int32_t len = (int32_t)(tag >> 2) + 1;
            int64_t off = (int64_t)((tagw >> 8) & 0xFFFFFFFFull);
            op.is_copy = 1;
            op.src = off;
            op.dst = out_pos;
            op.len = len;
            out_pos += len;
            break;
          }
        }
        if (op.is_copy && (op.src <= 0 || op.src > op.dst)) {
          status[page] = 2;  // corrupt back-reference
          done = 1;
        }
      }
    }
    done = wave_bcast(done);
    if (done) break;
    op.src = wave_bcast(op.src);
    op.dst = wave_bcast(op.dst);
    op.len = wave_bcast(op.len);
    op.is_copy = wave_bcast(op.is_copy);

    if (op.is_copy) {
      const int64_t off = op.src;
      uint8_t* d = dst_base + op.dst;
      const uint8_t* s = d - off;
      for (int32_t i = lane; i < op.len; i += PSA_WAVE)
        d[i] = s[i % off];
    } else {
      const uint8_t* s = in + op.src;
      uint8_t* d = dst_base + op.dst;
      // vectorize the common large-literal case
      int32_t len = op.len;
      int32_t vec = len & ~3;
      for (int32_t i = lane * 4; i < vec; i += PSA_WAVE * 4) {
        uint32_t w = load_u32_unaligned(s + i);
        d[i + 0] = (uint8_t)(w);
        d[i + 1] = (uint8_t)(w >> 8);
        d[i + 2] = (uint8_t)(w >> 16);
        d[i + 3] = (uint8_t)(w >> 24);
      }
      for (int32_t i = vec + lane; i < len; i += PSA_WAVE) d[i] = s[i];
    }
  }
  if (lane == 0 && out_pos != total) status[page] = 3;  // truncated stream
}

void snappy_decompress_batch(torch::Tensor comp, torch::Tensor comp_start,
                             torch::Tensor comp_end, torch::Tensor out,
                             torch::Tensor out_offsets,
                             torch::Tensor out_len, torch::Tensor status) {
  TORCH_CHECK(comp.is_cuda() && out.is_cuda(), "tensors must be on device");
  TORCH_CHECK(comp.scalar_type() == torch::kUInt8);
  TORCH_CHECK(comp_start.scalar_type() == torch::kInt64);
  int n_pages = (int)comp_start.numel();
  if (n_pages <= 0) return;
  const int WPB = 4;  // waves per block
  int blocks = (n_pages + WPB - 1) / WPB;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(snappy_decompress_kernel, dim3(blocks),
                     dim3(WPB * PSA_WAVE), 0, stream,
                     comp.data_ptr<uint8_t>(),
                     comp_start.data_ptr<int64_t>(),
                     comp_end.data_ptr<int64_t>(),
                     out.data_ptr<uint8_t>(),
                     out_offsets.data_ptr<int64_t>(),
                     out_len.data_ptr<int64_t>(),
                     status.data_ptr<int32_t>(), n_pages);
}

}  // namespace psa
