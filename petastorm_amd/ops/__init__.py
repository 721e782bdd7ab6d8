"""Loader for the in-tree HIP extension.

The extension is built in-tree (``python petastorm_amd/ops/setup.py
build_ext --inplace``) so the .so travels with the repo snapshot to GPU
boxes.  On a machine with a GPU, a missing extension is a hard error — GPU
ops must never silently fall back to an eager/CPU path.
"""

import glob
import importlib.util
import os

from petastorm_amd.errors import GpuExtensionNotAvailable

_HERE = os.path.dirname(os.path.abspath(__file__))
_ext = None
_load_error = None


def _try_load():
    global _ext, _load_error
    if _ext is not None:
        return _ext
    candidates = glob.glob(os.path.join(_HERE, '_petastorm_amd_hip*.so'))
    if not candidates:
        _load_error = 'extension .so not found under {} (run: PYTORCH_ROCM_ARCH=gfx950 ' \
                      'python petastorm_amd/ops/setup.py build_ext --inplace)'.format(_HERE)
        return None
    try:
        import torch  # noqa: F401 - the extension links against torch libs
        spec = importlib.util.spec_from_file_location('_petastorm_amd_hip',
                                                      candidates[0])
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _ext = mod
        return _ext
    except Exception as e:  # noqa: BLE001
        _load_error = 'failed to load {}: {}'.format(candidates[0], e)
        return None


def available():
    return _try_load() is not None


def ext():
    """The extension module; raises loudly when missing."""
    mod = _try_load()
    if mod is None:
        raise GpuExtensionNotAvailable(
            'petastorm_amd HIP extension is not available: {}'.format(_load_error))
    return mod
