"""Dispatch layer between the python ops and the HIP extension.

Policy (keeps GPU runs honest — see repo docs): on a CUDA/ROCm device the
HIP kernels are the compute path.  If the extension is missing on a GPU box
the ops raise instead of silently falling back to eager PyTorch, unless
``DWT_AMD_ALLOW_EAGER=1`` is set (used only for debugging).
"""
from __future__ import annotations

import os
import warnings

_ext = None
_tried = False


def _load():
    global _ext, _tried
    if _tried:
        return _ext
    _tried = True
    try:
        import glob
        import importlib.util
        import os.path as osp

        here = osp.dirname(__file__)
        sos = sorted(glob.glob(osp.join(here, "_dwt_hip*.so")))
        if sos:
            spec = importlib.util.spec_from_file_location("dwt_amd.kernels._dwt_hip", sos[0])
            mod = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(mod)
            _ext = mod
    except Exception as exc:  # pragma: no cover - build issues surface here
        warnings.warn(f"dwt_amd HIP extension failed to load: {exc}")
        _ext = None
    return _ext


def available() -> bool:
    import torch
    if not torch.cuda.is_available():
        return False
    return _load() is not None


def ext():
    e = _load()
    if e is None:
        raise RuntimeError(
            "dwt_amd HIP extension is not built. Run `python -m dwt_amd.kernels.build` "
            "(or the repo's __graft_entry__.build()) to compile it for gfx950."
        )
    return e


def require_or_warn(opname: str) -> None:
    """On a GPU box, a missing extension is an error, not a silent fallback."""
    if os.environ.get("DWT_AMD_ALLOW_EAGER") == "1":
        warnings.warn(f"dwt_amd: running {opname} in eager fallback (DWT_AMD_ALLOW_EAGER=1)")
        return
    raise RuntimeError(
        f"dwt_amd: op '{opname}' called on a GPU but the HIP extension is not "
        "loaded. Build it with `python -m dwt_amd.kernels.build`, or set "
        "DWT_AMD_ALLOW_EAGER=1 to debug with the eager path."
    )


# The concrete HIP-backed implementations are registered here once the
# extension exists; ops import these names.

def whiten_multi(x, gamma, beta, running_means, running_vars, cfg):
    from . import hip_ops
    return hip_ops.whiten_multi(x, gamma, beta, running_means, running_vars, cfg)


def batch_norm_multi(x, gamma, beta, running_means, running_vars, cfg):
    from . import hip_ops
    return hip_ops.batch_norm_multi(x, gamma, beta, running_means, running_vars, cfg)


def mec_loss(x, y):
    from . import hip_ops
    return hip_ops.mec_loss(x, y)


def entropy_loss(x):
    from . import hip_ops
    return hip_ops.entropy_loss(x)


def ce_loss(x, target):
    from . import hip_ops
    return hip_ops.ce_loss(x, target)
