"""Build the petastorm_amd HIP extension in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python petastorm_amd/ops/setup.py build_ext --inplace

The resulting .so lands next to this file and is loaded by
petastorm_amd.ops (package __init__).  gfx950 (MI355X) is the only target.
Works from any cwd (sources resolve relative to this file).
"""
import os

_HERE = os.path.dirname(os.path.abspath(__file__))

_SRC_NAMES = [
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


def main():
    from setuptools import setup

    os.environ.setdefault('PYTORCH_ROCM_ARCH', 'gfx950')
    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    # torch's hipify stage resolves sources against the cwd: run the build
    # from this directory so the command works from anywhere
    os.chdir(_HERE)
    src = [s for s in _SRC_NAMES if os.path.exists(os.path.join(_HERE, s))]
    setup(
        name='petastorm_amd_hip',
        ext_modules=[
            CUDAExtension(
                name='_petastorm_amd_hip',
                sources=src,
                extra_compile_args={
                    'cxx': ['-O3', '-std=c++17'],
                    'nvcc': ['-O3', '-std=c++17'],
                },
            )
        ],
        cmdclass={'build_ext': BuildExtension},
    )


if __name__ == '__main__':
    main()
