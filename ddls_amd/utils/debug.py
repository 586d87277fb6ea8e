"""ROCm debug / sanitizer hooks (the MI355X equivalents of the
CUDA compute-sanitizer story — there is no compute-sanitizer on ROCm, so
debugging kernel misbehaviour uses serialization + blocking launches +
XNACK page-fault reporting + device asserts).

Enable with ``DDLS_AMD_DEBUG=1`` in the environment (picked up at import by
``ddls_amd.ops``) or call :func:`enable_debug_mode` BEFORE the first HIP
context is created.  ``DDLS_AMD_DEBUG_BUILD=1`` additionally compiles the
in-tree kernels with ``-g -O1 -DDDLS_AMD_DEVICE_ASSERT`` so in-kernel
bounds/invariant asserts fire (abort with file:line on the GPU).
"""
from __future__ import annotations

import os

# env knobs set by enable_debug_mode (ROCm runtime debugging):
#   AMD_SERIALIZE_KERNEL=3  serialize kernel launch + wait (isolates the
#                           faulting kernel in host order)
#   AMD_SERIALIZE_COPY=3    same for async copies
#   HIP_LAUNCH_BLOCKING=1   every launch synchronous (CUDA_LAUNCH_BLOCKING)
#   HSA_XNACK=1             retry-on-fault addressing: page faults report a
#                           precise faulting address instead of aborting
#   AMD_LOG_LEVEL=3         HIP runtime API/arg logging (very verbose)
_DEBUG_ENV = {
    "AMD_SERIALIZE_KERNEL": "3",
    "AMD_SERIALIZE_COPY": "3",
    "HIP_LAUNCH_BLOCKING": "1",
    "HSA_XNACK": "1",
}


def debug_enabled() -> bool:
    return os.environ.get("DDLS_AMD_DEBUG", "0") == "1"


def enable_debug_mode(verbose_runtime: bool = False) -> dict:
    """Export the ROCm serialization/XNACK debug knobs; returns what was set.
    Must run before the first hipLaunchKernel of the process."""
    applied = {}
    for k, v in _DEBUG_ENV.items():
        os.environ[k] = v
        applied[k] = v
    if verbose_runtime:
        os.environ["AMD_LOG_LEVEL"] = "3"
        applied["AMD_LOG_LEVEL"] = "3"
    return applied


def debug_build_flags() -> list:
    """Extra hipcc flags for a debug build of the in-tree kernels."""
    if os.environ.get("DDLS_AMD_DEBUG_BUILD", "0") != "1":
        return []
    return ["-g", "-O1", "-DDDLS_AMD_DEVICE_ASSERT"]
