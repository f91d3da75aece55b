"""Custom-op layer: hand-written HIP/CDNA4 kernels behind torch tensors.

The compiled extension ``theanompi_amd._C`` (built in-tree by setup.py via
``hipcc --offload-arch=gfx950``) provides every GPU math kernel the
framework uses.  On a GPU box the extension is
REQUIRED: ops raise immediately rather than fall back to any library
implementation (no MIOpen, no eager-PyTorch conv).  On CPU the pure-torch
reference implementations in ``functional.py`` run instead — they define
the numerics the HIP kernels are tested against.
"""

import torch

_C = None
_C_ERR = None


def _try_load():
    global _C, _C_ERR
    if _C is not None:
        return _C
    try:
        from theanompi_amd import _C as ext  # built by setup.py build_ext --inplace
        _C = ext
    except ImportError as e:
        _C_ERR = e
        _C = None
    return _C


def ext():
    """Return the HIP extension module, or raise loudly on a GPU box."""
    m = _try_load()
    if m is None:
        raise RuntimeError(
            "theanompi_amd._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_C_ERR}")
    return m


def has_ext():
    return _try_load() is not None
