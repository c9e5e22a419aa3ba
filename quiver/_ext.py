"""Loader for the native torch_quiver extension.

The .so is built in-tree (repo root) by build_ext.py so it travels with
gpurun snapshots.  On a GPU box a missing native extension is a hard error —
no silent eager fallback.
"""
import os
import sys

import torch  # noqa: F401  — must be imported before the extension

_repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _repo_root not in sys.path:
    sys.path.insert(0, _repo_root)

try:
    import torch_quiver as _native
except ImportError as e:  # pragma: no cover
    if torch.cuda.is_available():
        raise ImportError(
            "torch_quiver native extension not built; run "
            "`python build_ext.py` at the repo root") from e
    _native = None

_HAS_NATIVE = _native is not None


def _require():
    if _native is None:
        raise RuntimeError(
            "torch_quiver native extension unavailable "
            "(build with `python build_ext.py`)")
    return _native


def __getattr__(name):
    return getattr(_require(), name)
