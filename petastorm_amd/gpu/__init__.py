"""MI355X GPU pipeline: native Parquet row-group decode on-device.

This package is the MI355X-native replacement for the reference's worker
pools + Arrow C++ decode: row-group bytes are read into
pinned host buffers, copied H2D asynchronously, and every hot decode stage
(snappy, Parquet page decode, NdarrayCodec unpack, JPEG, layout transforms)
runs as gfx950 HIP kernels from petastorm_amd/ops.
"""
