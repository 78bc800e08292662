"""Fused DQN TD loss (double/vanilla target + Huber/MSE + PER IS weights +
|TD| priorities) — HIP on GPU, composed reference on CPU.

Reference semantics: dqn_agent.py:155-171 (double-DQN target + MSE),
apex/worker.py:134-161 (IS weights, priority update).
"""

from __future__ import annotations

import ctypes
from typing import Tuple

import torch

from . import _backend

_c = ctypes.c_void_p


def td_loss_reference(q, q_next_online, q_next_target, actions, rewards,
                      discounts, weights=None, huber=False, huber_delta=1.0):
    """Autograd-capable reference.  Returns (loss, |td| detached)."""
    if q_next_online is not None:
        astar = q_next_online.argmax(dim=1, keepdim=True)
    else:
        astar = q_next_target.argmax(dim=1, keepdim=True)
    target = rewards + discounts * q_next_target.gather(1, astar).squeeze(1)
    pred = q.gather(1, actions.unsqueeze(1)).squeeze(1)
    td = pred - target.detach()
    if huber:
        a = td.abs()
        l = torch.where(a <= huber_delta, 0.5 * td * td,
                        huber_delta * (a - 0.5 * huber_delta))
    else:
        l = td * td
    if weights is not None:
        l = l * weights
    return l.mean(), td.abs().detach()


def per_is_weights(prios, p_total, p_min, replay_size, beta):
    """(N*P)^-beta / max_w with P = p/total (replay_buffer.py:370-381)."""
    w = (replay_size * (prios / p_total)) ** (-beta)
    w_max = (replay_size * (p_min / p_total)) ** (-beta)
    return w / w_max


class _FusedTDLossFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, q, q_next_online, q_next_target, actions, rewards,
                discounts, prios, p_total, p_min, beta, replay_size, huber,
                huber_delta):
        B, A = q.shape
        qc = q.contiguous().float()
        grad_q = torch.zeros_like(qc)
        td_abs = torch.empty(B, device=q.device, dtype=torch.float32)
        loss_out = torch.zeros(1, device=q.device, dtype=torch.float32)
        # bind every cast/contiguous temp to a local that outlives the
        # launch: a data_ptr() taken from an unreferenced temp is freed
        # before the ctypes call runs, and the next argument's allocation
        # can reuse+overwrite the block (the r2 dgrad corruption)
        qno = (q_next_online.contiguous().float()
               if q_next_online is not None else None)
        qnt = q_next_target.contiguous().float()
        act = actions.contiguous().long()
        rew = rewards.contiguous().float()
        dis = discounts.contiguous().float()
        ret = _backend.lib().fused_td_loss(
            _c(qc.data_ptr()),
            _c(qno.data_ptr()) if qno is not None else None,
            _c(qnt.data_ptr()),
            _c(act.data_ptr()),
            _c(rew.data_ptr()),
            _c(dis.data_ptr()),
            _c(prios.data_ptr()) if prios is not None else None,
            _c(p_total.data_ptr()) if p_total is not None else None,
            _c(p_min.data_ptr()) if p_min is not None else None,
            float(beta), int(replay_size), B, A, int(huber),
            float(huber_delta), _c(grad_q.data_ptr()), _c(td_abs.data_ptr()),
            _c(loss_out.data_ptr()), _backend.current_stream())
        _backend.check(ret, "fused_td_loss")
        ctx.save_for_backward(grad_q)
        ctx.mark_non_differentiable(td_abs)
        return loss_out[0], td_abs

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, g_loss, g_td):
        (grad_q,) = ctx.saved_tensors
        return (grad_q * g_loss,) + (None,) * 12


def fused_td_loss(q, q_next_online, q_next_target, actions, rewards,
                  discounts, *, prios=None, p_total=None, p_min=None,
                  beta: float = 0.4, replay_size: int = 0,
                  huber: bool = False, huber_delta: float = 1.0
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (loss scalar w/ autograd into q, |td| priorities [B])."""
    if q.is_cuda:
        return _FusedTDLossFn.apply(q, q_next_online, q_next_target, actions,
                                    rewards, discounts, prios, p_total, p_min,
                                    beta, replay_size, huber, huber_delta)
    weights = None
    if prios is not None:
        weights = per_is_weights(prios, p_total, p_min, replay_size, beta)
    return td_loss_reference(q, q_next_online, q_next_target, actions,
                             rewards, discounts, weights, huber, huber_delta)
