// Common helpers for petastorm_amd HIP kernels (gfx950 / CDNA4 only).
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  * wavefront = 64 lanes; all cross-lane code hard-codes 64
//  * memory-bound kernels vectorize to >= 8B/lane and use grid-stride loops
//    capped at ~2048 blocks
//  * no CUDA-compat shims: this file is HIP-native and compiled only for
//    gfx950
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define PSA_WAVE 64

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",      \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)

namespace psa {

__device__ __forceinline__ int lane_id() { return threadIdx.x & (PSA_WAVE - 1); }

// Broadcast a value from lane 0 to the whole wave.
template <typename T>
__device__ __forceinline__ T wave_bcast(T v) {
  return __shfl(v, 0, PSA_WAVE);
}

// Broadcast from an arbitrary source lane.
template <typename T>
__device__ __forceinline__ T wave_bcast_from(T v, int src_lane) {
  return __shfl(v, src_lane, PSA_WAVE);
}

// Unaligned little-endian loads from a byte stream (global memory).
__device__ __forceinline__ uint32_t load_u32_unaligned(const uint8_t* p) {
  return (uint32_t)p[0] | ((uint32_t)p[1] << 8) | ((uint32_t)p[2] << 16) |
         ((uint32_t)p[3] << 24);
}

__device__ __forceinline__ uint16_t load_u16_unaligned(const uint8_t* p) {
  return (uint16_t)((uint32_t)p[0] | ((uint32_t)p[1] << 8));
}

// Grid sizing for memory-bound grid-stride kernels: fill the 256 CUs with
// headroom but cap the launch (guide §6 Guideline 11).
inline int grid_for(int64_t total_threads, int block_size) {
  int64_t blocks = (total_threads + block_size - 1) / block_size;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

}  // namespace psa
