"""MI355X-native compute ops.

Every hot op has two paths:

* a hand-written CDNA4 HIP kernel in ``byol_amd/ops/csrc`` (built in-tree as
  ``byol_amd/_C*.so`` for gfx950) — the ONLY path used on GPU;
* a plain PyTorch fp32 reference of identical semantics — the numerics oracle
  used on CPU and by the unit tests.

Policy: on a GPU box the HIP extension is mandatory.  If a tensor lives on a
``cuda`` device and the extension is missing, ops raise instead of silently
falling back to eager ATen kernels.
"""

import os

_EXT = None
_EXT_ERR: str = ""


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        from .. import _C  # in-tree built extension
        _EXT = _C
    except ImportError as e:  # remember why, for the loud failure path
        _EXT_ERR = str(e)
        _EXT = None
    return _EXT


def extension():
    """The HIP extension module, or None (CPU-only environments)."""
    return _try_load()


def require_extension(what: str):
    ext = _try_load()
    if ext is None:
        raise RuntimeError(
            f"byol_amd HIP extension required for {what} on GPU but not "
            f"built (import error: {_EXT_ERR}). Build it with "
            f"`python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950).")
    return ext


def has_extension() -> bool:
    return _try_load() is not None
