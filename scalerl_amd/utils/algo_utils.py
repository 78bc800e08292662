"""Algorithm-side helpers (parity with ``scalerl/utils/algo_utils.py``,
minus the Accelerate-specific unwrapping, which has no equivalent here —
process groups are explicit)."""

from __future__ import annotations

from typing import Any, Dict

import torch


def chkpt_attribute_to_device(state: Dict[str, Any], device) -> Dict[str, Any]:
    """Move every tensor in a (possibly nested) checkpoint dict to device."""
    out = {}
    for k, v in state.items():
        if isinstance(v, torch.Tensor):
            out[k] = v.to(device)
        elif isinstance(v, dict):
            out[k] = chkpt_attribute_to_device(v, device)
        else:
            out[k] = v
    return out


def compile_model(model: torch.nn.Module, mode: str = "default"):
    """torch.compile wrapper (algo_utils.py:78-92).  NOTE: the MI355X hot
    paths here use hand-written HIP kernels + hipGraph capture instead of a
    tracing compiler (north-star constraint); this exists for parity and
    for cold paths."""
    return torch.compile(model, mode=mode)


def remove_compile_prefix(state_dict: Dict[str, Any]) -> Dict[str, Any]:
    """Strip torch.compile's `_orig_mod.` prefix (algo_utils.py:95-106)."""
    return {k.replace("_orig_mod.", "", 1): v for k, v in state_dict.items()}


def unwrap_optimizer(optimizer):
    """Reference-parity unwrap (algo_utils.py:10-39 unwraps Accelerate's
    AcceleratedOptimizer).  There is no Accelerate here; any wrapper
    exposing ``.optimizer`` is unwrapped, plain optimizers pass through."""
    seen = set()
    while hasattr(optimizer, "optimizer") and id(optimizer) not in seen:
        seen.add(id(optimizer))
        optimizer = optimizer.optimizer
    return optimizer


def get_device(device: str = "auto") -> torch.device:
    """Reference utils.get_device semantics (utils.py:6)."""
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"
    return torch.device(device)


def calculate_mean(values) -> float:
    """Mean of an iterable, 0.0 when empty (utils.py:28 semantics)."""
    values = list(values)
    return float(sum(values) / len(values)) if values else 0.0
