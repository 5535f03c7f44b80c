"""Loader for the in-tree HIP extension (mi355x._C).

The extension is compiled for gfx950 only (see mi355x/csrc/). On a GPU box
the ops layer REQUIRES it: any op invoked on a CUDA tensor without the
extension raises immediately instead of silently falling back to stock
PyTorch kernels. CPU tensors always use the plain-torch fp32 reference
implementations (that is the numerics reference the GPU kernels are tested
against, and the path BASELINE config 1 runs on).
"""

from __future__ import annotations

_ext_mod = None
_ext_err: Exception | None = None


def ext():
    """Return the compiled extension module, raising loudly if absent."""
    global _ext_mod, _ext_err
    if _ext_mod is not None:
        return _ext_mod
    if _ext_err is not None:
        raise RuntimeError(
            "mi355x HIP extension (mi355x._C) failed to import; GPU ops are "
            "unavailable. Build it in-tree with `python setup.py build_ext "
            "--inplace` (or __graft_entry__.build())."
        ) from _ext_err
    try:
        from mi355x import _C as m  # built in-tree: mi355x/_C.*.so
    except ImportError as e:
        _ext_err = e
        raise RuntimeError(
            "mi355x HIP extension (mi355x._C) is not built; refusing to run "
            "GPU ops on stock PyTorch kernels. Build it with `python setup.py "
            "build_ext --inplace` (or __graft_entry__.build())."
        ) from e
    _ext_mod = m
    return m


def have_ext() -> bool:
    try:
        ext()
        return True
    except RuntimeError:
        return False
