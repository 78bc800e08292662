"""Fused flat-buffer optimizers (HIP) with torch-semantics CPU references.

Used by :class:`scalerl_amd.parallel.flat.FlatParamModel`-based learners:
all parameters live in one contiguous fp32 buffer, so an optimizer step is
ONE kernel and gradient all-reduce is ONE RCCL call.  Semantics match
torch.optim.RMSprop / Adam (reference relies on those — impala_atari.py:99,
share_optim.py:65-122 — so checkpoints interop).
"""

from __future__ import annotations

import ctypes
from typing import Optional

import torch

from . import _backend

_c = ctypes.c_void_p


class FusedRMSprop:
    """RMSProp over a flat param/grad pair."""

    def __init__(self, param: torch.Tensor, lr: float, alpha: float = 0.99,
                 eps: float = 0.01, momentum: float = 0.0,
                 weight_decay: float = 0.0):
        assert param.dtype == torch.float32 and param.is_contiguous()
        self.param = param
        self.square_avg = torch.zeros_like(param)
        self.momentum_buf = torch.zeros_like(param) if momentum else None
        self.lr, self.alpha, self.eps = lr, alpha, eps
        self.momentum, self.weight_decay = momentum, weight_decay

    @torch.no_grad()
    def step(self, grad: torch.Tensor, lr: Optional[float] = None) -> None:
        lr = self.lr if lr is None else lr
        if self.param.is_cuda:
            ret = _backend.lib().fused_rmsprop(
                _c(self.param.data_ptr()), _c(grad.data_ptr()),
                _c(self.square_avg.data_ptr()),
                _c(self.momentum_buf.data_ptr()) if self.momentum_buf is not None else None,
                self.param.numel(), lr, self.alpha, self.eps, self.momentum,
                self.weight_decay, _backend.current_stream())
            _backend.check(ret, "fused_rmsprop")
            return
        g = grad if self.weight_decay == 0 else grad + self.weight_decay * self.param
        self.square_avg.mul_(self.alpha).addcmul_(g, g, value=1 - self.alpha)
        upd = g / (self.square_avg.sqrt() + self.eps)
        if self.momentum_buf is not None:
            self.momentum_buf.mul_(self.momentum).add_(upd)
            upd = self.momentum_buf
        self.param.add_(upd, alpha=-lr)

    def state_dict(self):
        return {"square_avg": self.square_avg,
                "momentum_buf": self.momentum_buf,
                "hyper": dict(lr=self.lr, alpha=self.alpha, eps=self.eps,
                              momentum=self.momentum,
                              weight_decay=self.weight_decay)}

    def load_state_dict(self, sd):
        self.square_avg.copy_(sd["square_avg"])
        if self.momentum_buf is not None and sd.get("momentum_buf") is not None:
            self.momentum_buf.copy_(sd["momentum_buf"])


class FusedAdam:
    """Adam over a flat param/grad pair (bias-corrected, torch semantics)."""

    def __init__(self, param: torch.Tensor, lr: float, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        assert param.dtype == torch.float32 and param.is_contiguous()
        self.param = param
        self.exp_avg = torch.zeros_like(param)
        self.exp_avg_sq = torch.zeros_like(param)
        self.lr, self.betas, self.eps = lr, betas, eps
        self.weight_decay = weight_decay
        self.step_count = 0

    @torch.no_grad()
    def step(self, grad: torch.Tensor, lr: Optional[float] = None) -> None:
        lr = self.lr if lr is None else lr
        self.step_count += 1
        b1, b2 = self.betas
        if self.param.is_cuda:
            ret = _backend.lib().fused_adam(
                _c(self.param.data_ptr()), _c(grad.data_ptr()),
                _c(self.exp_avg.data_ptr()), _c(self.exp_avg_sq.data_ptr()),
                self.param.numel(), lr, b1, b2, self.eps, self.weight_decay,
                self.step_count, _backend.current_stream())
            _backend.check(ret, "fused_adam")
            return
        g = grad if self.weight_decay == 0 else grad + self.weight_decay * self.param
        self.exp_avg.mul_(b1).add_(g, alpha=1 - b1)
        self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** self.step_count
        bc2 = 1 - b2 ** self.step_count
        denom = (self.exp_avg_sq / bc2).sqrt_().add_(self.eps)
        self.param.addcdiv_(self.exp_avg / bc1, denom, value=-lr)

    def state_dict(self):
        return {"exp_avg": self.exp_avg, "exp_avg_sq": self.exp_avg_sq,
                "step": self.step_count,
                "hyper": dict(lr=self.lr, betas=self.betas, eps=self.eps,
                              weight_decay=self.weight_decay)}

    def load_state_dict(self, sd):
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.step_count = int(sd.get("step", 0))


@torch.no_grad()
def fused_polyak_(dst: torch.Tensor, src: torch.Tensor, tau: float) -> None:
    """dst ← τ·src + (1−τ)·dst over flat buffers."""
    if dst.is_cuda:
        ret = _backend.lib().fused_polyak(
            _c(dst.data_ptr()), _c(src.data_ptr()), dst.numel(), tau,
            _backend.current_stream())
        _backend.check(ret, "fused_polyak")
    else:
        dst.lerp_(src, tau)


@torch.no_grad()
def clip_grad_norm_(grad: torch.Tensor, max_norm: float,
                    scratch: Optional[torch.Tensor] = None) -> Optional[torch.Tensor]:
    """In-place global-norm clip of a flat grad buffer, no host sync.

    Returns the device scratch holding ||g||² (pre-clip) for optional logging.
    """
    if max_norm <= 0:
        return None
    if grad.is_cuda:
        if scratch is None:
            scratch = torch.zeros(1, device=grad.device, dtype=torch.float32)
        else:
            scratch.zero_()
        ret = _backend.lib().grad_clip_by_norm(
            _c(grad.data_ptr()), grad.numel(), _c(scratch.data_ptr()),
            max_norm, _backend.current_stream())
        _backend.check(ret, "grad_clip_by_norm")
        return scratch
    norm = grad.norm()
    if norm > max_norm:
        grad.mul_(max_norm / (norm + 1e-6))
    return norm.pow(2).reshape(1)
