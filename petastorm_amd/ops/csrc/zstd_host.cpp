// Host-side batch ZSTD decompression of Parquet pages (one frame per
// page), using the from-scratch decoder in zstd_core.h — no libzstd, no
// Arrow.  Runs in the reader's IO prefetch thread with the GIL released;
// pages decompress in parallel on a std::thread pool, the result lands in
// a (pinned) host buffer that uploads once and then flows through the
// SAME device decode path as uncompressed chunks (so byte-array offset
// scans and jpeg/png host parsing keep working on zstd datasets).
//
// Replaces the Arrow C++ zstd decoder behind piece.read() in the
// reference (petastorm/arrow_reader_worker.py:358).
#include <torch/extension.h>

#include <atomic>
#include <thread>
#include <vector>

#include "zstd_core.h"

namespace psa {

void zstd_decompress_host(torch::Tensor src, torch::Tensor src_off,
                          torch::Tensor src_len, torch::Tensor dst,
                          torch::Tensor dst_off, torch::Tensor dst_len,
                          torch::Tensor status) {
  TORCH_CHECK(!src.is_cuda() && !dst.is_cuda(),
              "zstd_decompress_host operates on host tensors");
  TORCH_CHECK(src.scalar_type() == torch::kUInt8);
  TORCH_CHECK(src_off.scalar_type() == torch::kInt64);
  const uint8_t* sp = src.data_ptr<uint8_t>();
  uint8_t* dp = dst.data_ptr<uint8_t>();
  const int64_t* so = src_off.data_ptr<int64_t>();
  const int64_t* sl = src_len.data_ptr<int64_t>();
  const int64_t* dofs = dst_off.data_ptr<int64_t>();
  const int64_t* dl = dst_len.data_ptr<int64_t>();
  int32_t* st = status.data_ptr<int32_t>();
  int n = (int)src_off.numel();
  if (n <= 0) return;

  py::gil_scoped_release release;
  auto work = [&](int i, zstd::ZstdWork* ws) {
    long r = zstd::decode_frame(sp + so[i], (size_t)sl[i], dp + dofs[i],
                                (size_t)dl[i], ws);
    st[i] = (r == dl[i]) ? 0 : (r < 0 ? (int32_t)-r : 100);
  };
  int hw = (int)std::thread::hardware_concurrency();
  int n_threads = n < 2 ? 1 : std::min(n, std::max(2, hw / 2));
  if (n_threads <= 1) {
    auto ws = std::make_unique<zstd::ZstdWork>();
    for (int i = 0; i < n; ++i) work(i, ws.get());
    return;
  }
  std::atomic<int> next(0);
  std::vector<std::thread> threads;
  threads.reserve(n_threads);
  for (int t = 0; t < n_threads; ++t) {
    threads.emplace_back([&]() {
      auto ws = std::make_unique<zstd::ZstdWork>();  // ~140 KB: heap, not stack
      int i;
      while ((i = next.fetch_add(1)) < n) work(i, ws.get());
    });
  }
  for (auto& t : threads) t.join();
}

}  // namespace psa
