"""Loader for the in-tree HIP extension (dist_mnist_amd/_C*.so).

Policy (task requirement): on a GPU box the HIP kernels ARE the compute path.
If a CUDA/HIP tensor reaches one of our ops and the extension is missing, we
raise loudly instead of silently falling back to eager PyTorch. CPU tensors use
plain PyTorch reference implementations (used by the no-GPU test tier and the
gloo plumbing config).
"""

from __future__ import annotations

import glob
import os

_EXT = None
_EXT_ERR: str | None = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    pkg_dir = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sos = sorted(glob.glob(os.path.join(pkg_dir, "_C*.so")))
    if not sos:
        _EXT_ERR = (
            f"dist_mnist_amd native extension not found under {pkg_dir} "
            "(expected dist_mnist_amd/_C*.so). Build it in-tree with "
            "`python setup.py build_ext --inplace` (hipcc, gfx950)."
        )
        return
    try:
        import importlib.util

        spec = importlib.util.spec_from_file_location("dist_mnist_amd._C", sos[0])
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)  # type: ignore[union-attr]
        _EXT = mod
    except Exception as e:  # pragma: no cover - load failure is environmental
        _EXT_ERR = f"failed to load {sos[0]}: {e!r}"


def available() -> bool:
    _try_load()
    return _EXT is not None


def ext():
    """Return the native module, raising loudly if it is not loadable."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "A CUDA tensor reached a dist_mnist_amd op but the HIP extension "
            "is not loaded: " + (_EXT_ERR or "unknown error")
        )
    return _EXT
