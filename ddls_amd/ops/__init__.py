"""HIP/CDNA4 kernel extension loading.

Kernels live in ``ddls_amd/ops/hip/*.hip`` (pure HIP, gfx950-only — no CUDA
compat paths) and are compiled in-tree via torch.utils.cpp_extension (which
drives hipcc under PYTORCH_ROCM_ARCH=gfx950), so the built .so travels with
the repo snapshot to GPU boxes.

Policy: on a GPU, the HIP path is the path that runs.  If the extension is
missing on a CUDA-capable machine, ops raise loudly; set
DDLS_AMD_ALLOW_TORCH_FALLBACK=1 to override (debug only).
"""
from __future__ import annotations

import os
from typing import Optional

_EXT = None
_EXT_ERR: Optional[Exception] = None

_HERE = os.path.dirname(os.path.abspath(__file__))
_BUILD_DIR = os.path.join(_HERE, "_build")
_SOURCES = [os.path.join(_HERE, "hip", "meanpool.hip"),
            os.path.join(_HERE, "hip", "meanpool_bwd.hip"),
            os.path.join(_HERE, "hip", "meanpool_mfma.hip"),
            os.path.join(_HERE, "hip", "ppo_loss.hip"),
            os.path.join(_HERE, "hip", "policy_head.hip"),
            os.path.join(_HERE, "hip", "lookahead.hip"),
            os.path.join(_HERE, "hip", "env_step.hip"),
            os.path.join(_HERE, "hip", "cached_step.hip"),
            os.path.join(_HERE, "hip", "bindings.hip")]


def build_extensions(verbose: bool = False):
    """Compile every HIP extension for gfx950 (driver 'does it build' check)."""
    global _EXT, _EXT_ERR
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD_DIR, exist_ok=True)
    from torch.utils.cpp_extension import load

    from ..utils.debug import debug_build_flags, debug_enabled, \
        enable_debug_mode
    if debug_enabled():
        # serialize kernels/copies + XNACK faults BEFORE the HIP context
        enable_debug_mode()
    dbg = debug_build_flags()
    _EXT = load(name="ddls_amd_kernels" + ("_debug" if dbg else ""),
                sources=_SOURCES,
                build_directory=_BUILD_DIR,
                extra_cflags=["-O3"] + dbg,
                extra_cuda_cflags=["-O3"] + dbg,
                verbose=verbose)
    _EXT_ERR = None
    return _EXT


def get_extension(required: bool = False):
    """Return the compiled extension module (building it on first use)."""
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    if _EXT_ERR is not None and not required:
        return None
    try:
        build_extensions(verbose=False)
    except Exception as e:  # build toolchain missing etc.
        _EXT_ERR = e
        if required:
            raise RuntimeError(
                "ddls_amd HIP extension unavailable on a GPU machine — the "
                "native kernel path is mandatory on GPUs (set "
                "DDLS_AMD_ALLOW_TORCH_FALLBACK=1 only for debugging)") from e
        return None
    return _EXT


def hip_ops_enabled_for(tensor) -> bool:
    """True if the HIP kernel path should serve this (inference) tensor."""
    import torch
    if not tensor.is_cuda:
        return False
    if os.environ.get("DDLS_AMD_DISABLE_HIP", "0") == "1":
        return False
    if torch.is_grad_enabled():
        # training path: the fused autograd Function owns the kernels (the
        # INPUT may be a constant while the PARAMS still need grads)
        return False
    allow_fallback = os.environ.get("DDLS_AMD_ALLOW_TORCH_FALLBACK", "0") == "1"
    ext = get_extension(required=not allow_fallback)
    return ext is not None
