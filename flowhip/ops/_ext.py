"""Loader for the in-tree HIP extension (`flowhip._C`).

The extension is built in-tree by `setup.py build_ext --inplace` (or
`__graft_entry__.build()`), producing `flowhip/_C.cpython-*.so` compiled for
gfx950. On a GPU box the HIP path is mandatory: ops called on CUDA tensors
raise if the extension is missing, so a silent eager fallback can never
masquerade as the native path. Set FLOWHIP_ALLOW_FALLBACK=1 to permit the
torch reference path on GPU (debugging only), or FLOWHIP_FORCE_REF=1 to force
it (used by GPU oracle tests).
"""

import os

_EXT = None
_EXT_ERR = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from flowhip import _C  # type: ignore
        _EXT = _C
    except ImportError as e:  # pragma: no cover - exercised only sans build
        _EXT_ERR = e
    return _EXT


def ext():
    """Return the extension module or None if not built."""
    return _load()


def force_ref():
    return os.environ.get("FLOWHIP_FORCE_REF", "0") == "1"


def use_hip(tensor):
    """Decide whether the HIP path should run for `tensor`.

    Returns True (use HIP), False (use torch reference). Raises on a CUDA
    tensor with no extension available unless explicitly allowed.
    """
    if not tensor.is_cuda:
        return False
    if force_ref():
        return False
    if _load() is not None:
        return True
    if os.environ.get("FLOWHIP_ALLOW_FALLBACK", "0") == "1":
        return False
    raise RuntimeError(
        "flowhip._C HIP extension is not built but a flowhip op was called on "
        "a CUDA tensor. Build it in-tree with `python setup.py build_ext "
        "--inplace` (PYTORCH_ROCM_ARCH=gfx950). Original import error: %r"
        % (_EXT_ERR,)
    )
