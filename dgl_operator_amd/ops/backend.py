"""Native extension loader.

The HIP/CDNA4 extension is built IN-TREE as ``dgl_operator_amd/_C.so``
(see csrc/build.py / __graft_entry__.py::build). On a GPU box the HIP path is
mandatory: any op asked to run on a CUDA (ROCm) tensor without the extension
raises, rather than silently falling back to eager PyTorch.
"""
from __future__ import annotations

import importlib
from typing import Optional

_EXT = None
_TRIED = False


def load_extension(required: bool = False):
    """Return the native module, importing it on first use."""
    global _EXT, _TRIED
    if _EXT is None and not _TRIED:
        _TRIED = True
        try:
            import torch  # noqa: F401  (libtorch symbols must be loaded first)

            _EXT = importlib.import_module("dgl_operator_amd._C")
        except ImportError as e:
            _EXT = None
            _IMPORT_ERROR[0] = e
    if _EXT is None and required:
        raise RuntimeError(
            "dgl_operator_amd native extension (_C.so) is not built — the HIP "
            "kernel path is required on GPU. Build it with "
            "`python -m dgl_operator_amd.csrc.build` or __graft_entry__.build(). "
            f"Original import error: {_IMPORT_ERROR[0]}"
        )
    return _EXT


_IMPORT_ERROR: list = [None]


def has_extension() -> bool:
    return load_extension(required=False) is not None


def ext_for(tensor) -> Optional[object]:
    """Extension handle for ops on ``tensor``: required on GPU, optional on CPU."""
    if tensor.is_cuda:
        return load_extension(required=True)
    return None
