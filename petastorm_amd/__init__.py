"""petastorm_amd: an MI355X-native Parquet data-loading framework with the
capabilities of uber/petastorm.

Public API parity: /root/reference/petastorm/__init__.py:15-19.
"""

from petastorm_amd.errors import NoDataAvailableError  # noqa: F401
from petastorm_amd.reader import make_reader, make_batch_reader  # noqa: F401
from petastorm_amd.transform import TransformSpec  # noqa: F401

__version__ = '0.1.0'
