// GPU ZSTD decompression: one thread per frame (Parquet page), reusing the
// validated single-pass decoder in zstd_core.h.
//
// ZSTD's FSE/Huffman entropy stages are bit-serial per frame, so frame-level
// parallelism is the unit here (hundreds of pages per row-group batch);
// per-frame workspaces (~140 KB: tables + a 128 KiB literals buffer) live in
// a caller-provided global scratch tensor — far too large for LDS.  The
// host-thread path (zstd_host.cpp) remains the default; this kernel keeps
// the decode on-device for page-rich workloads where the PCIe-free path and
// page parallelism pay.
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include "zstd_core.h"

namespace psa {

__global__ void zstd_decompress_kernel(
    const uint8_t* __restrict__ src, const int64_t* __restrict__ src_off,
    const int64_t* __restrict__ src_len, uint8_t* __restrict__ dst,
    const int64_t* __restrict__ dst_off, const int64_t* __restrict__ dst_len,
    zstd::ZstdWork* __restrict__ works, int32_t* __restrict__ status,
    int n_frames) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n_frames) return;
  long r = zstd::decode_frame(src + src_off[i], (size_t)src_len[i],
                              dst + dst_off[i], (size_t)dst_len[i],
                              &works[i]);
  status[i] = (r == dst_len[i]) ? 0 : (r < 0 ? (int32_t)-r : 100);
}

int64_t zstd_work_bytes() { return (int64_t)sizeof(zstd::ZstdWork); }

void zstd_decompress_batch(torch::Tensor src, torch::Tensor src_off,
                           torch::Tensor src_len, torch::Tensor dst,
                           torch::Tensor dst_off, torch::Tensor dst_len,
                           torch::Tensor work, torch::Tensor status) {
  TORCH_CHECK(src.is_cuda() && dst.is_cuda() && work.is_cuda(),
              "tensors must be on device");
  TORCH_CHECK(src.scalar_type() == torch::kUInt8);
  int n = (int)src_off.numel();
  if (n <= 0) return;
  TORCH_CHECK(work.numel() >= (int64_t)n * (int64_t)sizeof(zstd::ZstdWork),
              "work tensor too small");
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  int threads = 64;
  hipLaunchKernelGGL(zstd_decompress_kernel,
                     dim3((n + threads - 1) / threads), dim3(threads), 0,
                     stream, src.data_ptr<uint8_t>(),
                     src_off.data_ptr<int64_t>(),
                     src_len.data_ptr<int64_t>(), dst.data_ptr<uint8_t>(),
                     dst_off.data_ptr<int64_t>(),
                     dst_len.data_ptr<int64_t>(),
                     reinterpret_cast<zstd::ZstdWork*>(
                         work.data_ptr<uint8_t>()),
                     status.data_ptr<int32_t>(), n);
}

}  // namespace psa
