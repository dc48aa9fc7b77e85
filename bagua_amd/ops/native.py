"""Loader for the native C++/HIP extension ``bagua_amd._C``.

The extension is built in-tree (setup.py build_ext --inplace) for gfx950
only. Policy: on a GPU box the native path is MANDATORY — ops raise if the
extension is missing, so a silent eager fallback can never masquerade as
the HIP path. On CPU-only machines (CI) everything falls back to torch
reference implementations.
"""

import logging

import torch

logger = logging.getLogger(__name__)

_lib = None
_tried = False


def lib():
    """The native module, or None when unavailable on CPU."""
    global _lib, _tried
    if not _tried:
        _tried = True
        try:
            from bagua_amd import _C  # built in-tree

            _lib = _C
        except ImportError as e:
            if torch.cuda.is_available():
                raise ImportError(
                    "bagua_amd._C native extension not built but a GPU is "
                    "present. Build with `python setup.py build_ext "
                    "--inplace` (PYTORCH_ROCM_ARCH=gfx950). Refusing to "
                    "fall back to eager on GPU.") from e
            logger.info("native extension unavailable (CPU-only): %s", e)
            _lib = None
    return _lib


def available() -> bool:
    return lib() is not None


def require():
    if lib() is None:
        raise RuntimeError(
            "bagua_amd._C native extension required for the GPU path; "
            "build with `python setup.py build_ext --inplace`.")
    return _lib
