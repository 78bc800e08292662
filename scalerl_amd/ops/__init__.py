"""HIP kernel ops with pure-PyTorch CPU references.

Dispatch policy (no silent fallbacks): CUDA/ROCm tensors require the
in-tree kernel library ``_hip_ops.so`` (built by ``_backend.build()``);
CPU tensors run the reference implementations, which double as the
numerics oracles in ``tests/``.
"""

from ._backend import available as kernels_available
from ._backend import build as build_kernels
from .lstm import MaskedLSTM
from .optim import FusedAdam, FusedRMSprop, clip_grad_norm_, fused_polyak_
from .per import SumTree
from .scans import discounted_returns, gae, nstep_fold
from .td import fused_td_loss, per_is_weights, td_loss_reference
from .ppo import ppo_fused_loss, ppo_loss_reference
from .vtrace import (VTraceReturns, action_log_probs, impala_loss,
                     impala_loss_reference, vtrace_from_log_rhos,
                     vtrace_reference)

__all__ = [
    "kernels_available", "build_kernels", "MaskedLSTM", "FusedAdam",
    "FusedRMSprop", "clip_grad_norm_", "fused_polyak_", "SumTree",
    "discounted_returns", "gae", "nstep_fold", "fused_td_loss",
    "per_is_weights", "td_loss_reference", "VTraceReturns",
    "action_log_probs", "impala_loss", "impala_loss_reference",
    "vtrace_from_log_rhos", "vtrace_reference", "ppo_fused_loss",
    "ppo_loss_reference",
]
