"""HIP extension loader.

The CDNA4 kernels live in ``csrc/`` and are built IN-TREE (setup.py
build_ext --inplace) to ``pertgnn/_C*.so`` so the binary travels with the
repo snapshot to GPU boxes.  On a machine with a GPU the HIP path is
mandatory: ops raise if a CUDA tensor reaches them without the extension
(no silent eager fallback on GPU).  Set ``PERTGNN_FORCE_EAGER=1`` to
explicitly allow the eager path on GPU (used by parity tests only).
"""
from __future__ import annotations

import importlib
import os

_EXT = None
_TRIED = False


def _load():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        _EXT = importlib.import_module("pertgnn._C")
    except ImportError:
        _EXT = None
    return _EXT


def ext():
    """Return the loaded HIP extension module or None."""
    return _load()


def has_hip() -> bool:
    return _load() is not None


def force_eager() -> bool:
    return os.environ.get("PERTGNN_FORCE_EAGER", "0") == "1"


def require_ext():
    m = _load()
    if m is None:
        raise RuntimeError(
            "pertgnn HIP extension (pertgnn._C) is not built but a CUDA tensor "
            "reached a pertgnn op. Build it with `python setup.py build_ext "
            "--inplace` (PYTORCH_ROCM_ARCH=gfx950). Refusing to silently fall "
            "back to eager on a GPU; set PERTGNN_FORCE_EAGER=1 only for parity "
            "tests."
        )
    return m


def use_hip(t) -> bool:
    """Decide the execution path for a tensor: HIP kernels on GPU (mandatory
    unless PERTGNN_FORCE_EAGER), eager on CPU."""
    if not t.is_cuda:
        return False
    if force_eager():
        return False
    require_ext()
    return True
