"""Native op dispatch layer.

Every op has (a) a hand-written HIP/CDNA4 kernel in nornicdb_amd/csrc
(the only GPU path — there is no CUDA/Metal/Vulkan fallback, by design,
cf. reference pkg/gpu's 4-backend probe which we deliberately do NOT
replicate) and (b) a plain PyTorch CPU reference used for CPU-only test
runs and as the numerics oracle.

On a GPU box the native extension is REQUIRED: ops raise RuntimeError
rather than silently falling back to eager PyTorch.
"""

import torch

try:
    from nornicdb_amd import _C  # built in-tree by `python setup.py build_ext --inplace`

    HAS_NATIVE = True
except ImportError as _e:  # pragma: no cover
    _C = None
    HAS_NATIVE = False
    _IMPORT_ERROR = _e


def require_native():
    """Raise loudly if the HIP extension is missing on a GPU box."""
    if not HAS_NATIVE:
        raise RuntimeError(
            "nornicdb_amd._C native HIP extension is not built. "
            "Run `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_IMPORT_ERROR!r}"
        )
    return _C


def native_or_none():
    return _C


from .vector import l2_normalize_, fill_random_unit_  # noqa: E402,F401
from .knn import knn_search, knn_search_exact  # noqa: E402,F401
