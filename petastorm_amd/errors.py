"""Framework-level exceptions.

Parity: /root/reference/petastorm/errors.py:16-17 (NoDataAvailableError) plus
the decode/extension errors this framework adds for the GPU path.
"""


class PetastormAmdError(Exception):
    """Base class for all framework errors."""


class NoDataAvailableError(PetastormAmdError):
    """Raised when a shard would receive zero row-groups.

    Reference raises this when ``shard_count`` exceeds the number of row
    groups (petastorm/reader.py:583-585).
    """


class DecodeFieldError(PetastormAmdError):
    """Raised when decoding a single field of a row fails.

    Reference: petastorm/utils.py:48-49 (DecodeFieldError wraps codec errors).
    """


class GpuExtensionNotAvailable(PetastormAmdError):
    """Raised when a GPU op is requested but the HIP extension is missing.

    The HIP extension must fail loudly on a GPU box rather than silently
    falling back to an eager/CPU path.
    """


class PetastormMetadataError(PetastormAmdError):
    """Missing/invalid dataset metadata (reference etl/dataset_metadata.py
    exception of the same name)."""


class PetastormMetadataGenerationError(PetastormAmdError):
    """Metadata (re)generation failed (reference name parity)."""
