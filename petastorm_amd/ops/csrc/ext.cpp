// petastorm_amd HIP extension: python bindings.
//
// Every function here either launches gfx950 HIP kernels (snappy, parquet
// page decode, jpeg, layout transforms) or performs the host half of a GPU
// pipeline stage (jpeg header parse, parquet page-header walk).  There is no
// CPU fallback in this module by design: if it imports, the GPU path is the
// path that runs.
#include <torch/extension.h>

namespace psa {

// snappy.hip
void snappy_decompress_batch(torch::Tensor comp, torch::Tensor comp_start,
                             torch::Tensor comp_end, torch::Tensor out,
                             torch::Tensor out_offsets,
                             torch::Tensor out_len, torch::Tensor status);
// lz4.hip
void lz4_decompress_batch(torch::Tensor comp, torch::Tensor blk_start,
                          torch::Tensor blk_end, torch::Tensor out,
                          torch::Tensor out_offsets, torch::Tensor out_len,
                          torch::Tensor status);

// zstd_host.cpp
void zstd_decompress_host(torch::Tensor src, torch::Tensor src_off,
                          torch::Tensor src_len, torch::Tensor dst,
                          torch::Tensor dst_off, torch::Tensor dst_len,
                          torch::Tensor status);

// zstd.hip
int64_t zstd_work_bytes();
void zstd_decompress_batch(torch::Tensor src, torch::Tensor src_off,
                           torch::Tensor src_len, torch::Tensor dst,
                           torch::Tensor dst_off, torch::Tensor dst_len,
                           torch::Tensor work, torch::Tensor status);

// parquet_decode.hip
void rle_hybrid_decode_batch(torch::Tensor data, torch::Tensor start,
                             torch::Tensor end, torch::Tensor bit_width,
                             torch::Tensor n_values, torch::Tensor out_off,
                             torch::Tensor out, torch::Tensor status);
void byte_array_offsets_batch(torch::Tensor data, torch::Tensor start,
                              torch::Tensor end, torch::Tensor n_values,
                              torch::Tensor out_off, torch::Tensor val_off,
                              torch::Tensor val_len, torch::Tensor status);
void varlen_gather(torch::Tensor src, torch::Tensor src_off,
                   torch::Tensor lengths, torch::Tensor dst,
                   torch::Tensor dst_off);
void plain_fixed_decode_batch(torch::Tensor page_buf,
                              torch::Tensor payload_start,
                              torch::Tensor payload_end,
                              torch::Tensor n_values, torch::Tensor row0,
                              int64_t def_mode, int64_t esize,
                              int64_t fill_pattern, torch::Tensor def_buf,
                              torch::Tensor def_start, torch::Tensor def_len,
                              torch::Tensor out,
                              torch::Tensor valid_out, torch::Tensor status);
void npy_payload_offsets(torch::Tensor data, torch::Tensor val_off,
                         torch::Tensor val_len, torch::Tensor pay_off,
                         torch::Tensor pay_len, torch::Tensor status);
// transforms.hip
void nhwc_to_nchw_normalize(torch::Tensor in, torch::Tensor out,
                            torch::Tensor mean, torch::Tensor inv_std,
                            double scale);
// jpeg_host.cpp
py::dict jpeg_parse_batch(torch::Tensor buf, torch::Tensor val_off,
                          torch::Tensor val_len);
// thrift_pages.cpp
py::dict parquet_walk_pages(torch::Tensor buf, torch::Tensor chunk_off,
                            torch::Tensor chunk_len);
// inflate.hip
void inflate_batch(torch::Tensor src, torch::Tensor seg_off,
                   torch::Tensor seg_len, torch::Tensor seg_first,
                   torch::Tensor seg_count, torch::Tensor dst,
                   torch::Tensor dst_off, torch::Tensor dst_cap,
                   torch::Tensor produced, int64_t mode,
                   torch::Tensor status);
void png_unfilter_batch(torch::Tensor raw, torch::Tensor raw_off,
                        torch::Tensor out, torch::Tensor out_off,
                        torch::Tensor height, torch::Tensor row_bytes,
                        torch::Tensor bpp, torch::Tensor status);
void bswap16(torch::Tensor data);
// png_host.cpp
py::dict png_parse_batch(torch::Tensor buf, torch::Tensor val_off,
                         torch::Tensor val_len);
py::dict byte_array_host_offsets(torch::Tensor buf, torch::Tensor val_start,
                                 torch::Tensor counts);
// jpeg.hip
void jpeg_decode_batch(torch::Tensor data, py::dict meta, torch::Tensor coef,
                       torch::Tensor samples, torch::Tensor out,
                       torch::Tensor out_off, torch::Tensor status);
void delta_binary_packed_batch(torch::Tensor page_buf, torch::Tensor start,
                               torch::Tensor end, torch::Tensor n_values,
                               torch::Tensor out_off, torch::Tensor out,
                               int64_t esize, torch::Tensor status);
void delta_length_byte_array_batch(torch::Tensor page_buf,
                                   torch::Tensor start, torch::Tensor end,
                                   torch::Tensor n_values,
                                   torch::Tensor out_idx,
                                   torch::Tensor val_off,
                                   torch::Tensor val_len,
                                   torch::Tensor status);
void delta_byte_array_lengths_batch(torch::Tensor page_buf,
                                    torch::Tensor start, torch::Tensor end,
                                    torch::Tensor n_values,
                                    torch::Tensor out_idx,
                                    torch::Tensor pre, torch::Tensor sfx,
                                    torch::Tensor suf_data_pos,
                                    torch::Tensor status);
void delta_byte_array_reconstruct_batch(
    torch::Tensor page_buf, torch::Tensor n_values, torch::Tensor out_idx,
    torch::Tensor pre, torch::Tensor sfx, torch::Tensor suf_data_pos,
    torch::Tensor val_off, torch::Tensor out, torch::Tensor status);
void byte_stream_split_batch(torch::Tensor page_buf, torch::Tensor start,
                             torch::Tensor n_values, torch::Tensor out_off,
                             torch::Tensor out, int64_t esize);
void bool_unpack_batch(torch::Tensor page_buf, torch::Tensor start,
                       torch::Tensor n_values, torch::Tensor out_off,
                       torch::Tensor out);
void jpeg_decode_fused_batch(torch::Tensor data, py::dict meta,
                             torch::Tensor coef, torch::Tensor samples,
                             torch::Tensor out, torch::Tensor out_off,
                             torch::Tensor mean, torch::Tensor inv_std,
                             double scale, torch::Tensor status);

}  // namespace psa

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "petastorm_amd MI355X (gfx950) decode kernels";
  m.def("snappy_decompress_batch", &psa::snappy_decompress_batch,
        "Batched snappy page decompression (wave-per-page)");
  m.def("lz4_decompress_batch", &psa::lz4_decompress_batch,
        "batch LZ4 block decompression, one wave per block");
  m.def("zstd_decompress_host", &psa::zstd_decompress_host,
        "batch ZSTD frame decompression on host threads (from-scratch "
        "RFC 8878 decoder, GIL released)");
  m.def("zstd_work_bytes", &psa::zstd_work_bytes,
        "per-frame device workspace size for zstd_decompress_batch");
  m.def("zstd_decompress_batch", &psa::zstd_decompress_batch,
        "batch ZSTD frame decompression on device (thread per frame)");
  m.def("rle_hybrid_decode_batch", &psa::rle_hybrid_decode_batch,
        "Parquet RLE/bit-packed hybrid decode (levels & dict indices)");
  m.def("byte_array_offsets_batch", &psa::byte_array_offsets_batch,
        "PLAIN byte-array page -> per-value (offset, length)");
  m.def("varlen_gather", &psa::varlen_gather,
        "Unaligned variable-length byte gather (funnel-shift copy)");
  m.def("plain_fixed_decode_batch", &psa::plain_fixed_decode_batch,
        "Fused PLAIN data-page decode: def levels + prefix scan + scatter");
  m.def("npy_payload_offsets", &psa::npy_payload_offsets,
        ".npy container -> payload (offset, length)");
  m.def("nhwc_to_nchw_normalize", &psa::nhwc_to_nchw_normalize,
        "Fused uint8 NHWC -> float NCHW normalize (LDS-tiled)");
  m.def("parquet_walk_pages", &psa::parquet_walk_pages,
        "Walk Parquet page headers (thrift compact) in a raw column chunk");
  m.def("inflate_batch", &psa::inflate_batch,
        "Batched DEFLATE inflate (zlib/raw), thread-per-stream");
  m.def("png_unfilter_batch", &psa::png_unfilter_batch,
        "PNG scanline unfilter (wave-per-image, bpp-lane chains)");
  m.def("bswap16", &psa::bswap16, "byte-swap 16-bit samples in place");
  m.def("png_parse_batch", &psa::png_parse_batch,
        "Host-side PNG container parse");
  m.def("byte_array_host_offsets", &psa::byte_array_host_offsets,
        "Host-side PLAIN byte-array offset scan");
  m.def("jpeg_parse_batch", &psa::jpeg_parse_batch,
        "Host-side JPEG header/segment parse");
  m.def("jpeg_decode_batch", &psa::jpeg_decode_batch,
        "GPU baseline JPEG decode (huffman/idct/color kernels)");
  m.def("delta_binary_packed_batch", &psa::delta_binary_packed_batch,
        "DELTA_BINARY_PACKED page decode (wave-per-page, shfl prefix scan)");
  m.def("delta_length_byte_array_batch",
        &psa::delta_length_byte_array_batch,
        "DELTA_LENGTH_BYTE_ARRAY page -> per-value (offset, length)");
  m.def("delta_byte_array_lengths_batch",
        &psa::delta_byte_array_lengths_batch,
        "DELTA_BYTE_ARRAY pass 1: prefix/suffix lengths");
  m.def("delta_byte_array_reconstruct_batch",
        &psa::delta_byte_array_reconstruct_batch,
        "DELTA_BYTE_ARRAY pass 2: materialize front-coded values");
  m.def("byte_stream_split_batch", &psa::byte_stream_split_batch,
        "BYTE_STREAM_SPLIT de-interleave");
  m.def("bool_unpack_batch", &psa::bool_unpack_batch,
        "PLAIN boolean bit-unpack");
  m.def("jpeg_decode_fused_batch", &psa::jpeg_decode_fused_batch,
        "JPEG decode with fused YCbCr->RGB + normalize + NCHW fp32 "
        "epilogue (skips the NHWC uint8 intermediate)");
}
