"""Package metadata for petastorm_amd (source-tree installs).

Console scripts mirror the reference's entry points (reference
setup.py:96-102: petastorm-copy-dataset.py, petastorm-generate-metadata.py,
petastorm-throughput.py).  The HIP extension is built separately/in-tree:

    PYTORCH_ROCM_ARCH=gfx950 python petastorm_amd/ops/setup.py build_ext --inplace
"""
from setuptools import find_packages, setup

setup(
    name='petastorm-amd',
    version='0.1.0',
    description='MI355X-native Parquet data-loading framework with '
                'uber/petastorm capabilities',
    packages=find_packages(include=['petastorm_amd', 'petastorm_amd.*']),
    python_requires='>=3.9',
    install_requires=['numpy', 'pyarrow>=6.0.1', 'fsspec', 'psutil'],
    extras_require={
        'torch': ['torch'],
        'tf': ['tensorflow'],
        'spark': ['pyspark'],
    },
    entry_points={
        'console_scripts': [
            'petastorm-amd-throughput = petastorm_amd.benchmark.cli:main',
            'petastorm-amd-copy-dataset = petastorm_amd.tools.copy_dataset:main',
            'petastorm-amd-generate-metadata = '
            'petastorm_amd.etl.petastorm_generate_metadata:main',
            'petastorm-amd-metadata-util = petastorm_amd.etl.metadata_util:main',
            'petastorm-amd-reencode = petastorm_amd.tools.reencode_dataset:main',
        ],
    },
)
