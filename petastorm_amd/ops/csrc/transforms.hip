// Fused layout/normalize transforms.
//
// Replaces the per-row Python TransformSpec work of the reference (e.g. the
// normalize in examples/mnist/pytorch_example.py:92-106 and
// _sanitize_pytorch_types, reference petastorm/pytorch.py:40-70) with one
// fused pass: uint8 NHWC -> float32/bf16 NCHW with per-channel mean/std.
//
// This op is memory-bound (reads N*H*W*C bytes, writes 2-4x that), so the
// MI355X design goal is one coalesced read + one coalesced write per element:
// an LDS-staged tile turns the C-strided NHWC reads into contiguous
// per-channel writes.  (MFMA does not apply: there is no dot product here;
// a matrix-core "transpose by multiply with permutation" would burn 2x the
// bytes it saves.)
#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_bf16.h>
#include "common.h"

namespace psa {

// Each block handles TILE_PIX consecutive pixels of one image: reads
// TILE_PIX*3 contiguous bytes (fully coalesced uchar), stages per channel in
// LDS, writes 3 contiguous float runs (fully coalesced).
template <typename OutT>
__global__ void nhwc_to_nchw_normalize_kernel(
    const uint8_t* __restrict__ in,   // [N, HW, C] contiguous
    OutT* __restrict__ out,           // [N, C, HW] contiguous
    int64_t hw, int c,
    const float* __restrict__ mean,   // [C]
    const float* __restrict__ inv_std,  // [C]
    float scale) {                    // e.g. 1/255 applied before mean/std
  constexpr int TILE_PIX = 2048;
  __shared__ float tile[3][TILE_PIX];

  const int64_t n = blockIdx.y;
  const int64_t pix0 = (int64_t)blockIdx.x * TILE_PIX;
  const int64_t npix = min((int64_t)TILE_PIX, hw - pix0);
  if (npix <= 0) return;

  const uint8_t* src = in + (n * hw + pix0) * c;
  const int64_t nbytes = npix * c;
  // vectorized read: 4 bytes per lane per iteration (guide G13 — scalar
  // byte loads run ~2.5x slower); falls back to bytes when the tile base
  // is not 4-aligned (odd H*W*C images)
  if (((uintptr_t)src & 3) == 0) {
    const uint32_t* src4 = (const uint32_t*)src;
    const int64_t nwords = nbytes >> 2;
    for (int64_t w = threadIdx.x; w < nwords; w += blockDim.x) {
      uint32_t v = src4[w];
      int64_t i = w << 2;
#pragma unroll
      for (int b = 0; b < 4; ++b) {
        int64_t idx = i + b;
        tile[(int)(idx % c)][idx / c] = (float)((v >> (8 * b)) & 0xFF);
      }
    }
    for (int64_t i = (nwords << 2) + threadIdx.x; i < nbytes;
         i += blockDim.x)
      tile[(int)(i % c)][i / c] = (float)src[i];
  } else {
    for (int64_t i = threadIdx.x; i < nbytes; i += blockDim.x)
      tile[(int)(i % c)][i / c] = (float)src[i];
  }
  __syncthreads();
  for (int ch = 0; ch < c; ++ch) {
    const float m = mean[ch], is = inv_std[ch];
    OutT* dst = out + (n * c + ch) * hw + pix0;
    // vectorized write: 4 outputs per lane per iteration
    const int64_t nv = npix & ~(int64_t)3;
    if ((((uintptr_t)dst) & 15) == 0 && sizeof(OutT) == 4) {
      float4* dst4 = (float4*)dst;
      for (int64_t p4 = threadIdx.x; p4 < (nv >> 2); p4 += blockDim.x) {
        int64_t p = p4 << 2;
        float4 v;
        v.x = (tile[ch][p + 0] * scale - m) * is;
        v.y = (tile[ch][p + 1] * scale - m) * is;
        v.z = (tile[ch][p + 2] * scale - m) * is;
        v.w = (tile[ch][p + 3] * scale - m) * is;
        dst4[p4] = v;
      }
      for (int64_t p = nv + threadIdx.x; p < npix; p += blockDim.x)
        dst[p] = (OutT)((tile[ch][p] * scale - m) * is);
    } else {
      for (int64_t p = threadIdx.x; p < npix; p += blockDim.x)
        dst[p] = (OutT)((tile[ch][p] * scale - m) * is);
    }
  }
}

void nhwc_to_nchw_normalize(torch::Tensor in, torch::Tensor out,
                            torch::Tensor mean, torch::Tensor inv_std,
                            double scale) {
  TORCH_CHECK(in.is_cuda() && in.scalar_type() == torch::kUInt8);
  TORCH_CHECK(in.dim() == 4, "in must be [N,H,W,C]");
  TORCH_CHECK(in.is_contiguous() && out.is_contiguous());
  const int64_t N = in.size(0), H = in.size(1), W = in.size(2),
                C = in.size(3);
  TORCH_CHECK(C <= 3, "C <= 3 supported");
  TORCH_CHECK(out.size(0) == N && out.size(1) == C && out.size(2) == H &&
              out.size(3) == W, "out must be [N,C,H,W]");
  const int64_t hw = H * W;
  constexpr int TILE_PIX = 2048;
  dim3 grid((unsigned)((hw + TILE_PIX - 1) / TILE_PIX), (unsigned)N);
  hipStream_t stream = c10::hip::getCurrentHIPStream();
  if (out.scalar_type() == torch::kFloat32) {
    hipLaunchKernelGGL(nhwc_to_nchw_normalize_kernel<float>, grid, dim3(256),
                       0, stream, in.data_ptr<uint8_t>(),
                       out.data_ptr<float>(), hw, (int)C,
                       mean.data_ptr<float>(), inv_std.data_ptr<float>(),
                       (float)scale);
  } else if (out.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(nhwc_to_nchw_normalize_kernel<__hip_bfloat16>, grid,
                       dim3(256), 0, stream, in.data_ptr<uint8_t>(),
                       (__hip_bfloat16*)out.data_ptr(), hw, (int)C,
                       mean.data_ptr<float>(), inv_std.data_ptr<float>(),
                       (float)scale);
  } else {
    TORCH_CHECK(false, "out dtype must be float32 or bfloat16");
  }
}

}  // namespace psa
