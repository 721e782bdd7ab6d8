// ASAN/UBSan fuzz harness for the from-scratch ZSTD decoder
// (ops/csrc/zstd_core.h) — SURVEY §5.2: host code must run under
// sanitizers.  Builds WITHOUT torch: zstd_core.h is freestanding.
//
//   g++ -std=c++17 -O1 -g -fsanitize=address,undefined \
//       -fno-sanitize-recover=all -I ../../petastorm_amd/ops/csrc \
//       zstd_fuzz_main.cpp -o zstd_fuzz && ./zstd_fuzz
//
// Strategy: malformed input must produce an error code, never an
// out-of-bounds access.  Inputs: pure garbage, plausible frame headers
// with garbage bodies, truncations and bit flips of a hand-assembled
// valid-ish raw-block frame, and size-field extremes.  A deterministic
// xorshift PRNG keeps runs reproducible.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <memory>
#include <vector>

#include "zstd_core.h"

static uint64_t rng_state = 0x9E3779B97F4A7C15ull;
static uint32_t xrand() {
  rng_state ^= rng_state << 13;
  rng_state ^= rng_state >> 7;
  rng_state ^= rng_state << 17;
  return (uint32_t)(rng_state & 0xFFFFFFFFu);
}

// Hand-assembled minimal valid frame: magic + FHD(single-segment,
// content size byte) + one raw block.
static std::vector<uint8_t> valid_raw_frame(int payload) {
  std::vector<uint8_t> f;
  const uint8_t magic[4] = {0x28, 0xB5, 0x2F, 0xFD};
  f.insert(f.end(), magic, magic + 4);
  f.push_back(0x20);                    // FHD: single-segment, FCS 1 byte
  f.push_back((uint8_t)payload);        // frame content size
  uint32_t hdr = (uint32_t)(1u | (0u << 1) | ((uint32_t)payload << 3));
  f.push_back((uint8_t)(hdr & 0xFF));   // last block, raw, size
  f.push_back((uint8_t)((hdr >> 8) & 0xFF));
  f.push_back((uint8_t)((hdr >> 16) & 0xFF));
  for (int i = 0; i < payload; ++i) f.push_back((uint8_t)xrand());
  return f;
}

int main() {
  auto ws = std::make_unique<psa::zstd::ZstdWork>();
  std::vector<uint8_t> out(1 << 16);
  long total_ok = 0, total_err = 0;

  // 1) sanity: the valid raw frame must decode
  {
    auto f = valid_raw_frame(100);
    long r = psa::zstd::decode_frame(f.data(), f.size(), out.data(), out.size(),
                                ws.get());
    if (r != 100) {
      fprintf(stderr, "FAIL: valid raw frame returned %ld\n", r);
      return 1;
    }
  }

  // 2) pure garbage buffers (exact-size allocations so ASAN sees any
  //    single-byte overread)
  for (int n = 0; n < 4000; ++n) {
    size_t len = xrand() % 512;
    std::unique_ptr<uint8_t[]> buf(new uint8_t[len ? len : 1]);
    for (size_t i = 0; i < len; ++i) buf[i] = (uint8_t)xrand();
    long r = psa::zstd::decode_frame(buf.get(), len, out.data(), out.size(),
                                ws.get());
    (r >= 0 ? total_ok : total_err)++;
  }

  // 3) garbage behind a real magic number (exercises header parsing)
  for (int n = 0; n < 4000; ++n) {
    size_t len = 4 + xrand() % 300;
    std::unique_ptr<uint8_t[]> buf(new uint8_t[len]);
    buf[0] = 0x28; buf[1] = 0xB5; buf[2] = 0x2F; buf[3] = 0xFD;
    for (size_t i = 4; i < len; ++i) buf[i] = (uint8_t)xrand();
    long r = psa::zstd::decode_frame(buf.get(), len, out.data(), out.size(),
                                ws.get());
    (r >= 0 ? total_ok : total_err)++;
  }

  // 4) truncations and bit flips of the valid frame
  {
    auto f = valid_raw_frame(200);
    for (size_t cut = 0; cut < f.size(); ++cut) {
      std::unique_ptr<uint8_t[]> buf(new uint8_t[cut ? cut : 1]);
      memcpy(buf.get(), f.data(), cut);
      long r = psa::zstd::decode_frame(buf.get(), cut, out.data(), out.size(),
                                  ws.get());
      (r >= 0 ? total_ok : total_err)++;
    }
    for (int n = 0; n < 4000; ++n) {
      std::vector<uint8_t> g = f;
      for (int k = 0; k < 1 + (int)(xrand() % 4); ++k)
        g[xrand() % g.size()] ^= (uint8_t)(1u << (xrand() % 8));
      std::unique_ptr<uint8_t[]> buf(new uint8_t[g.size()]);
      memcpy(buf.get(), g.data(), g.size());
      long r = psa::zstd::decode_frame(buf.get(), g.size(), out.data(),
                                  out.size(), ws.get());
      (r >= 0 ? total_ok : total_err)++;
    }
  }

  // 5) tiny output buffers (bounds on the write side)
  {
    auto f = valid_raw_frame(200);
    for (size_t cap = 0; cap < 220; ++cap) {
      std::unique_ptr<uint8_t[]> small(new uint8_t[cap ? cap : 1]);
      long r = psa::zstd::decode_frame(f.data(), f.size(), small.get(), cap,
                                  ws.get());
      if (cap < 200 && r >= 0) {
        fprintf(stderr, "FAIL: cap=%zu accepted (r=%ld)\n", cap, r);
        return 1;
      }
    }
  }

  printf("zstd fuzz OK: %ld decoded, %ld rejected, 0 sanitizer reports\n",
         total_ok, total_err);
  return 0;
}
