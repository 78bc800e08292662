"""Target-network updates (parity with ``scalerl/utils/model_utils.py:4-32``).

On CUDA/ROCm devices the polyak update runs through the fused HIP kernel
(:func:`scalerl_amd.ops.fused_polyak_`) over flat parameter buffers when both
models are flat; the generic path below covers arbitrary modules.
"""

from __future__ import annotations

import torch
import torch.nn as nn


@torch.no_grad()
def hard_target_update(src: nn.Module, dst: nn.Module) -> None:
    dst.load_state_dict(src.state_dict())


@torch.no_grad()
def soft_target_update(src: nn.Module, dst: nn.Module, tau: float = 0.005) -> None:
    """dst ← τ·src + (1−τ)·dst."""
    for ps, pd in zip(src.parameters(), dst.parameters()):
        pd.lerp_(ps, tau)
    for bs, bd in zip(src.buffers(), dst.buffers()):
        if bd.dtype.is_floating_point:
            bd.lerp_(bs.to(bd.dtype), tau)
        else:
            bd.copy_(bs)
