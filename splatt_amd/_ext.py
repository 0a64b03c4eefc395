"""Loader for the native extension (C++ core + gfx950 HIP kernels).

The extension is built in-tree (`python setup.py build_ext --inplace`) so the
.so travels with the repo snapshot. On a GPU box the HIP ops refuse to run
without it — there is no silent eager fallback for device paths.
"""
from __future__ import annotations

import torch  # noqa: F401  (libtorch symbols must be loaded first)

try:
    from splatt_amd import _C  # type: ignore
except ImportError as e:  # pragma: no cover
    _C = None
    _IMPORT_ERROR = e
else:
    _IMPORT_ERROR = None


def native() -> "object":
    """Return the native module, raising loudly if it is missing."""
    if _C is None:
        raise ImportError(
            "splatt_amd._C native extension is not built. Run "
            "`python setup.py build_ext --inplace` at the repo root. "
            f"(original error: {_IMPORT_ERROR})"
        )
    return _C


def have_native() -> bool:
    return _C is not None
