// GPU LZ4 block decompression of Parquet pages (codecs LZ4 / LZ4_RAW).
//
// Replaces the Arrow C++ LZ4 decoder the reference calls through
// piece.read() (reference petastorm/arrow_reader_worker.py:358).  The host
// side (gpu/decoder.py) parses the deprecated Hadoop framing
// ([4B BE dlen][4B BE clen][block])* into raw blocks, so this kernel only
// sees plain LZ4 blocks — one 64-lane wave per block.
//
// Same two-phase structure as snappy.hip: the sequence stream is serial, so
// lane 0 parses tokens out of an LDS-staged window and batches descriptors;
// all 64 lanes then replay the batch (literal copies from the input,
// match copies as out[d+i] = out[d-off + i%off], which is order-free because
// every read lands in completed output).
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "common.h"

namespace psa {

__global__ void lz4_decompress_kernel(
    const uint8_t* __restrict__ comp,
    const int64_t* __restrict__ blk_start,
    const int64_t* __restrict__ blk_end,
    uint8_t* __restrict__ out, const int64_t* __restrict__ out_off,
    const int64_t* __restrict__ out_len,
    int32_t* __restrict__ status, int n_blocks) {
  constexpr int WIN = 2048;   // staged window bytes
  constexpr int OPB = 128;    // op descriptors per batch
  constexpr int WPB = 4;      // waves per block (must match launcher)
  __shared__ uint8_t stage[WPB][WIN];
  __shared__ int64_t op_src[WPB][OPB];
  __shared__ int64_t op_dst[WPB][OPB];
  __shared__ int32_t op_len[WPB][OPB];   // negative length = match copy

  const int wave = threadIdx.x / PSA_WAVE;
  const int blk = blockIdx.x * (blockDim.x / PSA_WAVE) + wave;
  if (blk >= n_blocks) return;
  const int lane = lane_id();

  const uint8_t* in = comp + blk_start[blk];
  const int64_t in_len = blk_end[blk] - blk_start[blk];
  uint8_t* dst_base = out + out_off[blk];
  const int64_t total = out_len[blk];

  int64_t in_pos = 0, out_pos = 0;

  while (true) {
    int64_t win_base = wave_bcast(in_pos);
    if (win_base >= in_len) break;
    if (wave_bcast(out_pos) >= total && win_base >= in_len) break;
    for (int i = lane; i < WIN; i += PSA_WAVE)
      stage[wave][i] = (win_base + i < in_len) ? in[win_base + i] : 0;
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- parse phase (lane 0 only) ----
    int nops = 0;
    int bad = 0;
    int done = 0;
    if (lane == 0) {
      const uint8_t* w = stage[wave];
      // rbyte: LDS within the window, global fallback for long extension
      // runs that spill past it (rare; the window refills between ops)
      auto rbyte = [&](int64_t pos) -> uint8_t {
        int64_t rel = pos - win_base;
        return (rel >= 0 && rel < WIN) ? w[rel] : in[pos];
      };
      while (nops < OPB - 1) {
        if (in_pos >= in_len) { done = 1; break; }
        // refill once the next sequence header might leave the window
        // (token + both 8-byte-ish ext runs + offset stay under 20 in the
        // common case; longer ext runs use the global fallback)
        if (in_pos - win_base + 20 > WIN) break;
        uint8_t token = rbyte(in_pos++);
        int64_t lit = token >> 4;
        if (lit == 15) {
          uint8_t b;
          do {
            if (in_pos >= in_len) { bad = 2; break; }
            b = rbyte(in_pos++);
            lit += b;
          } while (b == 255);
          if (bad) break;
        }
        if (lit > 0) {
          if (in_pos + lit > in_len || out_pos + lit > total) {
            bad = 2; break;
          }
          op_src[wave][nops] = in_pos;
          op_dst[wave][nops] = out_pos;
          op_len[wave][nops] = (int32_t)lit;
          ++nops;
          in_pos += lit;
          out_pos += lit;
        }
        if (in_pos >= in_len) { done = 1; break; }  // last sequence
        if (in_pos + 2 > in_len) { bad = 2; break; }
        int64_t off = (int64_t)rbyte(in_pos) |
                      ((int64_t)rbyte(in_pos + 1) << 8);
        in_pos += 2;
        int64_t mlen = (token & 15) + 4;
        if ((token & 15) == 15) {
          uint8_t b;
          do {
            if (in_pos >= in_len) { bad = 2; break; }
            b = rbyte(in_pos++);
            mlen += b;
          } while (b == 255);
          if (bad) break;
        }
        if (off <= 0 || off > out_pos || out_pos + mlen > total) {
          bad = 2; break;
        }
        op_src[wave][nops] = off;
        op_dst[wave][nops] = out_pos;
        op_len[wave][nops] = (int32_t)(-mlen);
        ++nops;
        out_pos += mlen;
      }
      if (bad) status[blk] = bad;
    }
    bad = wave_bcast(bad);
    if (bad) return;
    nops = wave_bcast(nops);
    done = wave_bcast(done);
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
      } else {         // match copy (possibly overlapping)
        len = -len;
        const int64_t off = op_src[wave][k];
        uint8_t* d = dst_base + d0;
        const uint8_t* s = d - off;
        for (int32_t i = lane; i < len; i += PSA_WAVE)
          d[i] = s[i % off];
      }
    }
    if (done) {
      if (lane == 0 && out_pos != total) status[blk] = 3;  // short output
      return;
    }
  }
  if (lane == 0 && out_pos != total) status[blk] = 3;  // truncated stream
}

void lz4_decompress_batch(torch::Tensor comp, torch::Tensor blk_start,
                          torch::Tensor blk_end, torch::Tensor out,
                          torch::Tensor out_offsets,
                          torch::Tensor out_len, torch::Tensor status) {
  TORCH_CHECK(comp.is_cuda() && out.is_cuda(), "tensors must be on device");
  TORCH_CHECK(comp.scalar_type() == torch::kUInt8);
  TORCH_CHECK(blk_start.scalar_type() == torch::kInt64);
  int n_blocks = (int)blk_start.numel();
  if (n_blocks <= 0) return;
  const int WPB = 4;  // waves per block
  int blocks = (n_blocks + WPB - 1) / WPB;
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lz4_decompress_kernel, dim3(blocks),
                     dim3(WPB * PSA_WAVE), 0, stream,
                     comp.data_ptr<uint8_t>(),
                     blk_start.data_ptr<int64_t>(),
                     blk_end.data_ptr<int64_t>(),
                     out.data_ptr<uint8_t>(),
                     out_offsets.data_ptr<int64_t>(),
                     out_len.data_ptr<int64_t>(),
                     status.data_ptr<int32_t>(), n_blocks);
}

}  // namespace psa
