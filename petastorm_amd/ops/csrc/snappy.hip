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
  // Two-phase wave algorithm, LDS-staged:
  //   refill: all 64 lanes stage a window of the compressed stream in LDS
  //   parse:  lane 0 walks tags reading ~30-cycle LDS instead of ~400-cycle
  //           global memory, writing op descriptors (also LDS)
  //   exec:   all 64 lanes replay the op batch (coalesced global copies)
  // Snappy back-references always point at completed output, so every copy
  // parallelizes as out[d+i] = out[d-off + i%off].
  constexpr int WIN = 2048;   // staged window bytes
  constexpr int OPB = 128;    // op descriptors per batch
  constexpr int WPB = 4;      // waves per block (must match launcher)
  __shared__ uint8_t stage[WPB][WIN];
  __shared__ int64_t op_src[WPB][OPB];
  __shared__ int64_t op_dst[WPB][OPB];
  __shared__ int32_t op_len[WPB][OPB];   // negative length = back-copy

  const int wave = threadIdx.x / PSA_WAVE;
  const int page = blockIdx.x * (blockDim.x / PSA_WAVE) + wave;
  if (page >= n_pages) return;
  const int lane = lane_id();

  const uint8_t* in = comp + comp_start[page];
  const int64_t in_len = comp_end[page] - comp_start[page];
  uint8_t* dst_base = out + out_off[page];
  const int64_t expected = out_len[page];

  int64_t in_pos = 0, out_pos = 0;
  int64_t total = 0;
  if (lane == 0) total = read_varint(in, in_pos, in_len);
  total = wave_bcast(total);
  in_pos = wave_bcast(in_pos);
  if (total != expected) {
    if (lane == 0) status[page] = 1;  // length mismatch
    return;
  }

  while (true) {
    // ---- refill window at the current stream position ----
    int64_t win_base = wave_bcast(in_pos);
    if (win_base >= in_len || wave_bcast(out_pos) >= total) break;
    for (int i = lane; i < WIN; i += PSA_WAVE)
      stage[wave][i] = (win_base + i < in_len) ? in[win_base + i] : 0;
    // lane 0 reads other lanes' LDS writes: needs a wave-wide LDS fence
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- parse phase (lane 0 only) ----
    int nops = 0;
    int bad = 0;
    if (lane == 0) {
      const uint8_t* w = stage[wave];
      while (nops < OPB && out_pos < total) {
        int64_t rel = in_pos - win_base;
        // headers are <= 5 bytes; never read stage[] past WIN — break to
        // refill the window at the new position instead
        if (rel + 5 > WIN) break;
        if (in_pos >= in_len) { bad = 3; break; }          // truncated
        uint8_t tag = w[rel];
        switch (tag & 3) {
          case 0: {  // literal
            int64_t len = (int64_t)(tag >> 2) + 1;
            int adv = 1;
            if (len > 60) {
              int n_extra = (int)(len - 60);
              len = 0;
              for (int i = 0; i < n_extra; ++i)
                len |= (int64_t)w[rel + 1 + i] << (8 * i);
              len += 1;
              adv = 1 + n_extra;
            }
            if (in_pos + adv + len > in_len ||
                out_pos + len > total) { bad = 2; break; }
            op_src[wave][nops] = in_pos + adv;   // absolute input offset
            op_dst[wave][nops] = out_pos;
            op_len[wave][nops] = (int32_t)len;
            in_pos += adv + len;                 // skip literal payload
            out_pos += len;
            break;
          }
          case 1: {
            int32_t len = ((tag >> 2) & 0x7) + 4;
            int64_t off = ((int64_t)(tag >> 5) << 8) | w[rel + 1];
            op_src[wave][nops] = off;
            op_dst[wave][nops] = out_pos;
            op_len[wave][nops] = -len;
            in_pos += 2;
            if (out_pos + len > total) { bad = 2; break; }
            out_pos += len;
            if (off <= 0 || off > op_dst[wave][nops]) bad = 2;
            break;
          }
          case 2: {
            int32_t len = (int32_t)(tag >> 2) + 1;
            int64_t off = (int64_t)w[rel + 1] | ((int64_t)w[rel + 2] << 8);
            op_src[wave][nops] = off;
            op_dst[wave][nops] = out_pos;
            op_len[wave][nops] = -len;
            in_pos += 3;
            if (out_pos + len > total) { bad = 2; break; }
            out_pos += len;
            if (off <= 0 || off > op_dst[wave][nops]) bad = 2;
            break;
          }
          default: {
            int32_t len = (int32_t)(tag >> 2) + 1;
            int64_t off = (int64_t)w[rel + 1] | ((int64_t)w[rel + 2] << 8) |
                          ((int64_t)w[rel + 3] << 16) |
                          ((int64_t)w[rel + 4] << 24);
            op_src[wave][nops] = off;
            op_dst[wave][nops] = out_pos;
            op_len[wave][nops] = -len;
            in_pos += 5;
            if (out_pos + len > total) { bad = 2; break; }
            out_pos += len;
            if (off <= 0 || off > op_dst[wave][nops]) bad = 2;
            break;
          }
        }
        if (bad) break;
        ++nops;
      }
      if (bad) status[page] = bad;
    }
    bad = wave_bcast(bad);
    if (bad) return;
    nops = wave_bcast(nops);
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- execute phase (all lanes) ----
    for (int k = 0; k < nops; ++k) {
      int32_t len = op_len[wave][k];
      int64_t d0 = op_dst[wave][k];
      if (len >= 0) {  // literal: copy from input
        const uint8_t* s = in + op_src[wave][k];
        uint8_t* d = dst_base + d0;
        int32_t vec = len & ~3;
        for (int32_t i = lane * 4; i < vec; i += PSA_WAVE * 4) {
          uint32_t v = load_u32_unaligned(s + i);
          d[i + 0] = (uint8_t)(v);
          d[i + 1] = (uint8_t)(v >> 8);
          d[i + 2] = (uint8_t)(v >> 16);
          d[i + 3] = (uint8_t)(v >> 24);
        }
        for (int32_t i = vec + lane; i < len; i += PSA_WAVE) d[i] = s[i];
      } else {         // back-copy (possibly overlapping)
        len = -len;
        const int64_t off = op_src[wave][k];
        uint8_t* d = dst_base + d0;
        const uint8_t* s = d - off;
        for (int32_t i = lane; i < len; i += PSA_WAVE)
          d[i] = s[i % off];
      }
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
