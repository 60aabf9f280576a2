"""HIP kernel bindings (gfx950).

The compiled extension ``mi355x_scale.ops._C`` is built in-tree by
``setup.py build_ext --inplace``. On a GPU box the HIP path is mandatory:
ops raise if the extension is missing (so GPU tests can never silently
fall back to eager PyTorch); on CPU the pure-torch reference
implementations run (they are also the numerics references in tests).
"""
from __future__ import annotations


import os as _os

try:
    if _os.environ.get("MI355X_OPS_EXT") == "_C_asan":
        # SURVEY §5.2 sanitizer pass: the AddressSanitizer device build
        # (setup.py MI355X_ASAN=1) substitutes for _C so the whole GPU
        # kernel test suite runs instrumented (tools/gpu_sanitize.sh)
        from . import _C_asan as _C  # type: ignore
    else:
        from . import _C  # type: ignore
    HAVE_EXT = True
except ImportError:
    _C = None
    HAVE_EXT = False


def require_ext() -> None:
    if not HAVE_EXT:
        raise RuntimeError(
            "mi355x_scale.ops._C is not built — run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace` "
            "(the HIP path is mandatory on GPU; no silent eager fallback)"
        )


from .preprocess import normalize_images  # noqa: E402,F401
