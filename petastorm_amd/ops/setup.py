"""Build the petastorm_amd HIP extension in-tree:

    cd petastorm_amd/ops && PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting .so lands next to this file and is loaded by
petastorm_amd.ops (package __init__).  gfx950 (MI355X) is the only target.
"""
import os

from setuptools import setup

os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

SRC = [
    'csrc/ext.cpp',
    'csrc/jpeg_host.cpp',
    'csrc/thrift_pages.cpp',
    'csrc/png_host.cpp',
    'csrc/snappy.hip',
    'csrc/lz4.hip',
    'csrc/zstd_host.cpp',
    'csrc/zstd.hip',
    'csrc/parquet_decode.hip',
    'csrc/jpeg.hip',
    'csrc/transforms.hip',
    'csrc/inflate.hip',
]
SRC = [s for s in SRC if os.path.exists(os.path.join(os.path.dirname(__file__) or '.', s))]

setup(
    name='petastorm_amd_hip',
    ext_modules=[
        CUDAExtension(
            name='_petastorm_amd_hip',
            sources=SRC,
            extra_compile_args={
                'cxx': ['-O3', '-std=c++17'],
                'nvcc': ['-O3', '-std=c++17'],
            },
        )
    ],
    cmdclass={'build_ext': BuildExtension},
)
