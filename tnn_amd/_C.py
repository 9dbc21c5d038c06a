"""Loader for the in-tree HIP extension ``tnn_amd._hip``.

The extension is built in-tree (``python setup.py build_ext --inplace`` or
``__graft_entry__.build()``) so the compiled ``.so`` travels with the source
tree. On a GPU machine the HIP kernels are the ONLY compute path for the hot
ops — if the extension is missing there we raise loudly instead of silently
falling back to eager PyTorch (that would invalidate every benchmark).

On CPU-only machines (CI) ops use the pure-PyTorch fp32 reference branches
inside :mod:`tnn_amd.ops.functional` (the numerics oracle the GPU tests
compare against), so the extension is optional there.
"""

from __future__ import annotations

import importlib

_ext = None
_load_error: Exception | None = None

try:
    _ext = importlib.import_module("tnn_amd._hip")
except Exception as e:  # pragma: no cover - exercised only when unbuilt
    _load_error = e


def available() -> bool:
    return _ext is not None


def ext():
    """Return the HIP extension module, raising loudly if absent.

    Called from every GPU dispatch path; never from CPU paths.
    """
    if _ext is None:
        raise RuntimeError(
            "tnn_amd._hip extension is not built but a CUDA/HIP tensor reached "
            "a tnn_amd op. Build it in-tree first: `python setup.py build_ext "
            f"--inplace` (original import error: {_load_error!r})"
        )
    return _ext


def ext_any():
    """The extension module for host-side features (the C++ image codec)
    — legitimately callable on CPU-only machines, same loud failure when
    the extension is unbuilt."""
    if _ext is None:
        raise RuntimeError(
            "tnn_amd._hip extension is not built (needed for the in-tree "
            "image codec). Build it: `python setup.py build_ext --inplace` "
            f"(original import error: {_load_error!r})"
        )
    return _ext
