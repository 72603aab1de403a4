"""Loader for the compiled HIP extension (_hipops.so).

The extension is built IN-TREE by `setup.py build_ext --inplace` (or
__graft_entry__.build()) with PYTORCH_ROCM_ARCH=gfx950 so the .so travels
with the repo snapshot to GPU boxes.

Policy (deliberate, see package docstring): on a CUDA/ROCm device the HIP
kernels are the ONLY execution path. If the extension is missing while a GPU
op is requested, we raise — never fall back to eager PyTorch on GPU. CPU
tensors never touch this module.
"""

from __future__ import annotations

import importlib

_ext = None
_tried = False


def _find() -> object:
    # The built .so lives next to the package (in-tree build_ext --inplace
    # places distributed_vgg_f_amd/_hipops*.so).
    return importlib.import_module("distributed_vgg_f_amd._hipops")


def ext():
    """Return the loaded HIP extension module, raising loudly if absent."""
    global _ext, _tried
    if _ext is None:
        try:
            _ext = _find()
        except ImportError as e:
            raise RuntimeError(
                "distributed_vgg_f_amd: the HIP extension (_hipops) is not "
                "built/importable, but a GPU op was requested. There is no "
                "eager fallback on GPU by design. Build it in-tree with "
                "`python setup.py build_ext --inplace` "
                f"(PYTORCH_ROCM_ARCH=gfx950). Original error: {e}"
            ) from e
    return _ext


def available() -> bool:
    global _tried
    try:
        ext()
        return True
    except RuntimeError:
        return False
