"""Backend dispatch for the hot columnar operators.

Policy: on a GPU (ROCm) the hand-written gfx950 HIP extension MUST load —
ops raise loudly if it is missing (no silent eager fallback; set
ANOVOS_AMD_ALLOW_TORCH_FALLBACK=1 to override for debugging). On CPU the
torch reference implementations run (CI container has no GPU). Library
GEMMs (correlation fallback, PCA projection) intentionally go through
rocBLAS via torch.matmul — that is the sanctioned library path; the fused
kernels in ops/hip/ cover the bespoke hot ops.
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_TRIED = False


def _try_load_ext():
    global _EXT, _EXT_TRIED
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        import importlib

        _EXT = importlib.import_module("anovos_amd.ops.hip.anovos_hip")
    except ImportError:
        try:
            # in-tree built .so lives next to the sources
            import glob
            import importlib.util
            import os.path as osp

            here = osp.join(osp.dirname(__file__), "hip")
            sos = glob.glob(osp.join(here, "anovos_hip*.so"))
            if sos:
                spec = importlib.util.spec_from_file_location("anovos_hip", sos[0])
                mod = importlib.util.module_from_spec(spec)
                spec.loader.exec_module(mod)
                _EXT = mod
        except Exception:
            _EXT = None
    return _EXT


def hip_ext():
    """Return the loaded HIP extension module or None."""
    return _try_load_ext()


def require_hip():
    ext = _try_load_ext()
    if ext is None and not os.environ.get("ANOVOS_AMD_ALLOW_TORCH_FALLBACK"):
        raise RuntimeError(
            "anovos_amd HIP extension (anovos_hip*.so) is not built/loadable "
            "but a GPU tensor reached a hot op. Build it with "
            "`python __graft_entry__.py build` (or setup in anovos_amd/ops/hip). "
            "Set ANOVOS_AMD_ALLOW_TORCH_FALLBACK=1 to debug with eager torch."
        )
    return ext


def use_hip(t: torch.Tensor) -> bool:
    """True if this tensor should go through the HIP kernels."""
    if not t.is_cuda:
        return False
    return require_hip() is not None
