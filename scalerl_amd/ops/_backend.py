"""HIP kernel backend: in-tree build (hipcc, gfx950 only) + ctypes loader.

The kernels are plain HIP (no torch extension ABI): one shared library
``_hip_ops.so`` built from ``csrc/*.hip`` with
``hipcc --offload-arch=gfx950``.  Python passes raw device pointers
(``tensor.data_ptr()``) and the current torch HIP stream, so kernels land on
the same stream as surrounding torch ops.

Policy (no silent fallbacks): on a CUDA/ROCm device every op REQUIRES the
library — a missing .so raises immediately.  CPU tensors use the pure
PyTorch reference implementations (also the test oracles).
"""

from __future__ import annotations

import ctypes
import os
import subprocess
import sys
from typing import Optional

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_THIS_DIR, "csrc")
_SO_PATH = os.path.join(_THIS_DIR, "_hip_ops.so")

_LIB: Optional[ctypes.CDLL] = None
_LOAD_ERROR: Optional[str] = None


def hip_sources():
    return sorted(
        os.path.join(_CSRC, f) for f in os.listdir(_CSRC) if f.endswith(".hip"))


def build(verbose: bool = True, arch: str = "gfx950") -> str:
    """Compile csrc/*.hip → _hip_ops.so in-tree.  Idempotent (mtime check)."""
    srcs = hip_sources()
    hdr = os.path.join(_CSRC, "common.h")
    if os.path.exists(_SO_PATH):
        newest = max(os.path.getmtime(p) for p in srcs + [hdr])
        if os.path.getmtime(_SO_PATH) >= newest:
            return _SO_PATH
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [hipcc, f"--offload-arch={arch}", "-O3", "-std=c++17", "-shared",
           "-fPIC", "-lrocblas", "-o", _SO_PATH] + srcs
    if verbose:
        print("[scalerl_amd.ops] building:", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return _SO_PATH


def _declare(lib: ctypes.CDLL) -> None:
    c = ctypes
    P, L, F, I = c.c_void_p, c.c_long, c.c_float, c.c_int
    sigs = {
        "impala_fused_loss": [P, P, P, P, P, P, P, F, F, F, F, F, L, L, L,
                              P, P, P, P, P],
        "vtrace_from_log_rhos": [P, P, P, P, P, F, F, F, L, L, P, P, P],
        "gae_scan": [P, P, P, P, F, L, L, P, P, P],
        "discounted_returns": [P, P, P, L, L, P, P],
        "nstep_fold": [P, P, F, L, L, L, P, P, P, P],
        "fused_rmsprop": [P, P, P, P, L, F, F, F, F, F, P],
        "fused_adam": [P, P, P, P, L, F, F, F, F, F, L, P],
        "fused_polyak": [P, P, L, F, P],
        "grad_clip_by_norm": [P, L, P, F, P],
        "fused_td_loss": [P, P, P, P, P, P, P, P, P, F, L, L, L, I, F,
                          P, P, P, P],
        "per_update": [P, L, P, P, L, P],
        "per_sample": [P, L, L, P, L, P, P, P],
        "per_leaf_min": [P, L, L, P, P],
        "lstm_pointwise_fwd": [P, P, P, P, L, L, P],
        "lstm_pointwise_bwd": [P, P, P, P, P, P, P, L, L, P],
        "masked_lstm_seq_fwd": [P, P, P, P, P, P, P, P, P, P, L, L, L, P],
        "masked_lstm_seq_bwd": [P, P, P, P, P, P, P, P, P, P, L, L, L, P],
    }
    for name, argtypes in sigs.items():
        fn = getattr(lib, name)
        fn.argtypes = argtypes
        fn.restype = c.c_int


def lib() -> ctypes.CDLL:
    """The loaded kernel library.  Raises if unavailable (no fallback)."""
    global _LIB, _LOAD_ERROR
    if _LIB is not None:
        return _LIB
    if not os.path.exists(_SO_PATH):
        raise RuntimeError(
            f"scalerl_amd HIP kernel library not found at {_SO_PATH}. "
            f"Build it with `python -c 'from scalerl_amd.ops import _backend; "
            f"_backend.build()'` or `python setup.py build_ext --inplace`. "
            f"GPU execution without the native kernels is not supported.")
    try:
        _LIB = ctypes.CDLL(_SO_PATH)
    except OSError as e:
        _LOAD_ERROR = str(e)
        raise RuntimeError(f"failed to load {_SO_PATH}: {e}") from e
    _declare(_LIB)
    return _LIB


def available() -> bool:
    return os.path.exists(_SO_PATH)


def current_stream() -> ctypes.c_void_p:
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def check(ret: int, name: str) -> None:
    if ret != 0:
        import torch
        hint = ""
        if ret == -2:
            hint = " (action-space size exceeds kernel MAX_A)"
        elif ret == -3:
            hint = " (rollout length exceeds LDS budget)"
        raise RuntimeError(
            f"HIP kernel {name} failed with code {ret}{hint}; "
            f"device={torch.cuda.get_device_name() if torch.cuda.is_available() else 'n/a'}")
