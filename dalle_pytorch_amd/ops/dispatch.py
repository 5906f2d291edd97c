"""HIP extension loading & dispatch policy.

The extension is compiled in-tree (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) so the ``.so`` travels with the repo snapshot to
GPU boxes. Policy:

* CUDA/ROCm tensor + extension present  -> HIP kernel path.
* CUDA/ROCm tensor + extension missing  -> RuntimeError (loud failure); set
  ``DALLE_AMD_ALLOW_EAGER=1`` to explicitly permit the eager path on GPU
  (used only for oracle comparisons in tests).
* CPU tensor -> eager oracle path.
"""

import importlib
import os

_HIP = None
_TRIED = False


def hip_module():
    """Return the loaded HIP extension module, or None."""
    global _HIP, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            _HIP = importlib.import_module('dalle_pytorch_amd._hip')
        except ImportError:
            _HIP = None
    return _HIP


def hip_available() -> bool:
    return hip_module() is not None


def allow_eager_on_gpu() -> bool:
    return os.environ.get('DALLE_AMD_ALLOW_EAGER', '0') == '1'


def using_eager_fallback(tensor) -> bool:
    """Decide eager vs HIP for this tensor; raise if on GPU without the ext."""
    if not tensor.is_cuda:
        return True
    if hip_available():
        return False
    if allow_eager_on_gpu():
        return True
    raise RuntimeError(
        'dalle_pytorch_amd: tensor is on GPU but the gfx950 HIP extension '
        '(dalle_pytorch_amd._hip) is not built. Run `python setup.py '
        'build_ext --inplace` (or set DALLE_AMD_ALLOW_EAGER=1 to force the '
        'eager path, e.g. for oracle tests).')
