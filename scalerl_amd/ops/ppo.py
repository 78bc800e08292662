"""Fused PPO clip loss (HIP on GPU, composed torch reference on CPU)."""

from __future__ import annotations

import ctypes
from typing import Tuple

import torch
import torch.nn.functional as F

from . import _backend

_c = ctypes.c_void_p


def ppo_loss_reference(logits, actions, old_logp, adv, returns, values,
                       clip_eps=0.2, vcoef=0.5, ecoef=0.01):
    """Autograd-capable oracle (matches runtime/ppo.py's composed ops).

    Returns (total, pg, v_loss, entropy) — all means."""
    logp_all = F.log_softmax(logits, dim=-1)
    logp = logp_all.gather(1, actions.unsqueeze(1)).squeeze(1)
    ratio = torch.exp(logp - old_logp)
    s1 = ratio * adv
    s2 = torch.clamp(ratio, 1 - clip_eps, 1 + clip_eps) * adv
    pg = -torch.min(s1, s2).mean()
    v_loss = F.mse_loss(values, returns)
    ent = -(logp_all.exp() * logp_all).sum(-1).mean()
    total = pg + vcoef * v_loss - ecoef * ent
    return total, pg.detach(), v_loss.detach(), ent.detach()


def _declare(lib):
    if getattr(lib, "_ppo_declared", False):
        return lib
    c = ctypes
    lib.ppo_fused_loss.argtypes = [c.c_void_p] * 6 + [c.c_float] * 3 + \
        [c.c_long, c.c_long] + [c.c_void_p] * 4
    lib.ppo_fused_loss.restype = c.c_int
    lib._ppo_declared = True
    return lib


class _PPOFusedLossFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, logits, values, actions, old_logp, adv, returns,
                clip_eps, vcoef, ecoef):
        lib = _declare(_backend.lib())
        N, A = logits.shape
        lg = logits.contiguous()
        vals = values.contiguous()
        grad_logits = torch.empty_like(lg)
        grad_values = torch.empty_like(vals)
        loss_out = torch.zeros(3, device=lg.device)
        # locals keep the cast temps alive past the launch (see ops/td.py)
        act = actions.contiguous().long()
        olp = old_logp.contiguous().float()
        advc = adv.contiguous().float()
        retc = returns.contiguous().float()
        ret = lib.ppo_fused_loss(
            _c(lg.data_ptr()), _c(act.data_ptr()),
            _c(olp.data_ptr()),
            _c(advc.data_ptr()),
            _c(retc.data_ptr()),
            _c(vals.data_ptr()), clip_eps, vcoef, ecoef, N, A,
            _c(grad_logits.data_ptr()), _c(grad_values.data_ptr()),
            _c(loss_out.data_ptr()), _backend.current_stream())
        _backend.check(ret, "ppo_fused_loss")
        ctx.save_for_backward(grad_logits, grad_values)
        ctx.mark_non_differentiable(loss_out)
        total = loss_out[0] + vcoef * loss_out[1] - ecoef * loss_out[2]
        return total, loss_out

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, g_total, g_comps):
        grad_logits, grad_values = ctx.saved_tensors
        return (grad_logits * g_total, grad_values * g_total) + (None,) * 7


def ppo_fused_loss(logits, values, actions, old_logp, adv, returns, *,
                   clip_eps=0.2, vcoef=0.5, ecoef=0.01
                   ) -> Tuple[torch.Tensor, torch.Tensor]:
    """(total, components[pg, v_loss, entropy]) with autograd into
    logits/values.  GPU → fused kernel; CPU → composed reference."""
    if logits.is_cuda:
        return _PPOFusedLossFn.apply(logits, values, actions, old_logp, adv,
                                     returns, clip_eps, vcoef, ecoef)
    total, pg, v, ent = ppo_loss_reference(logits, actions, old_logp, adv,
                                           returns, values, clip_eps, vcoef,
                                           ecoef)
    return total, torch.stack([pg, v, ent])
