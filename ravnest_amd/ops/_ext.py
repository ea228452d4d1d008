"""HIP extension loader.

The CDNA4 kernels live in ravnest_amd/csrc/*.hip, built IN-TREE to
ravnest_amd/_C*.so by `python setup.py build_ext --inplace` (driven by
__graft_entry__.build()). On a GPU box the extension is REQUIRED: ops
raise instead of silently falling back to eager (the driver records which
.so files the GPU processes actually load).
"""
from __future__ import annotations

_EXT = None
_TRIED = False


def get_ext(required: bool = False):
    global _EXT, _TRIED
    if not _TRIED:
        _TRIED = True
        try:
            from ravnest_amd import _C  # type: ignore
            _EXT = _C
        except ImportError:
            _EXT = None
    if required and _EXT is None:
        raise RuntimeError(
            "ravnest_amd HIP extension (_C) is not built. Run "
            "`python setup.py build_ext --inplace` (or __graft_entry__."
            "build()) — GPU ops refuse to fall back to eager.")
    return _EXT


def has_ext() -> bool:
    return get_ext(required=False) is not None
