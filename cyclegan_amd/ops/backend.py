"""HIP extension loading and dispatch policy.

The compute path on GPU is the in-tree HIP extension (gfx950 kernels under
``cyclegan_amd/ops/_hip``). On CPU (CI boxes without a GPU) a pure-PyTorch
reference implementation of every op is used instead — the same code doubles
as the numerics oracle for the kernel unit tests.

Policy (fail-loud): if a tensor is on a CUDA(=HIP) device and the extension
cannot be loaded, ops raise instead of silently falling back to stock
PyTorch kernels. Set ``CYGAN_ALLOW_TORCH_FALLBACK=1`` to permit the stock
path on GPU (used by oracle tests only).
"""

from __future__ import annotations

import os
import glob

import torch

_EXT = None
_EXT_ERR: str | None = None
_TRIED = False


def _find_ext_path() -> str | None:
    here = os.path.dirname(os.path.abspath(__file__))
    cands = sorted(glob.glob(os.path.join(here, "_hip", "_cyclegan_hip*.so")))
    return cands[0] if cands else None


def load_ext(required: bool = False):
    """Load the in-tree HIP extension; returns the module or None."""
    global _EXT, _EXT_ERR, _TRIED
    if _EXT is not None:
        return _EXT
    if _TRIED and not required:
        return None
    _TRIED = True
    path = _find_ext_path()
    if path is None:
        _EXT_ERR = "no built _cyclegan_hip*.so found under cyclegan_amd/ops/_hip (run `python setup.py build_ext --inplace`)"
        if required:
            raise RuntimeError(_EXT_ERR)
        return None
    try:
        import importlib.util

        spec = importlib.util.spec_from_file_location("_cyclegan_hip", path)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        _EXT = mod
        return _EXT
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = f"failed to load {path}: {e!r}"
        if required:
            raise RuntimeError(_EXT_ERR) from e
        return None


def allow_fallback() -> bool:
    return os.environ.get("CYGAN_ALLOW_TORCH_FALLBACK", "0") == "1"


def use_hip(*tensors: torch.Tensor) -> bool:
    """Decide the execution path for an op given its tensors.

    Returns True → run the HIP kernel; False → run the torch reference.
    Raises if on GPU without the extension and fallback is not allowed.
    """
    on_gpu = any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))
    if not on_gpu:
        return False
    if os.environ.get("CYGAN_FORCE_TORCH", "0") == "1":
        # explicit oracle mode for GPU A/B tests
        return False
    ext = load_ext()
    if ext is not None:
        return True
    if allow_fallback():
        return False
    raise RuntimeError(
        f"cyclegan_amd: tensor on GPU but HIP extension unavailable ({_EXT_ERR}); "
        "refusing silent fallback. Build with `python setup.py build_ext --inplace` "
        "or set CYGAN_ALLOW_TORCH_FALLBACK=1."
    )


def ext():
    return load_ext(required=True)
