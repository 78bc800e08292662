"""V-trace and the fused IMPALA loss.

CPU path = pure-PyTorch reference (the numerics oracle, reimplementing
``scalerl/algorithms/impala/vtrace.py:78-172`` and ``loss_fn.py:1-23``).
CUDA path = single fused HIP kernel (csrc/vtrace_loss.hip) producing the
loss values AND the analytic gradients w.r.t. learner logits/values in one
launch.
"""

from __future__ import annotations

import collections
import ctypes
import torch
import torch.nn.functional as F

from . import _backend

VTraceReturns = collections.namedtuple("VTraceReturns", ["vs", "pg_advantages"])


def action_log_probs(logits: torch.Tensor, actions: torch.Tensor) -> torch.Tensor:
    """log pi(a | s) from raw logits (vtrace.py:31-40)."""
    return -F.nll_loss(
        F.log_softmax(logits.flatten(0, -2), dim=-1),
        actions.flatten(), reduction="none").view_as(actions)


@torch.no_grad()
def vtrace_reference(log_rhos, discounts, rewards, values, bootstrap_value,
                     clip_rho_threshold=1.0, clip_c_threshold=1.0,
                     clip_pg_rho_threshold=1.0) -> VTraceReturns:
    """Sequential-scan reference (vtrace.py:134-172 semantics), any device."""
    rhos = torch.exp(log_rhos)
    clipped_rhos = torch.clamp(rhos, max=clip_rho_threshold)
    cs = torch.clamp(rhos, max=clip_c_threshold)
    values_tp1 = torch.cat([values[1:], bootstrap_value.unsqueeze(0)], dim=0)
    deltas = clipped_rhos * (rewards + discounts * values_tp1 - values)
    T = deltas.shape[0]
    acc = torch.zeros_like(bootstrap_value)
    out = []
    for t in range(T - 1, -1, -1):
        acc = deltas[t] + discounts[t] * cs[t] * acc
        out.append(acc)
    vs_minus_v = torch.stack(list(reversed(out)), dim=0)
    vs = vs_minus_v + values
    vs_tp1 = torch.cat([vs[1:], bootstrap_value.unsqueeze(0)], dim=0)
    clipped_pg_rhos = torch.clamp(rhos, max=clip_pg_rho_threshold)
    pg_advantages = clipped_pg_rhos * (rewards + discounts * vs_tp1 - values)
    return VTraceReturns(vs=vs, pg_advantages=pg_advantages)


@torch.no_grad()
def vtrace_from_log_rhos(log_rhos, discounts, rewards, values, bootstrap_value,
                         clip_rho_threshold=1.0, clip_c_threshold=1.0,
                         clip_pg_rho_threshold=1.0) -> VTraceReturns:
    """Device-dispatching V-trace: HIP kernel on GPU, reference on CPU."""
    if not log_rhos.is_cuda:
        return vtrace_reference(log_rhos, discounts, rewards, values,
                                bootstrap_value, clip_rho_threshold,
                                clip_c_threshold, clip_pg_rho_threshold)
    T, B = log_rhos.shape
    args = [x.contiguous().float() for x in
            (log_rhos, discounts, rewards, values, bootstrap_value)]
    vs = torch.empty_like(args[0])
    pg_adv = torch.empty_like(args[0])
    ret = _backend.lib().vtrace_from_log_rhos(
        *[ctypes.c_void_p(a.data_ptr()) for a in args],
        clip_rho_threshold, clip_c_threshold, clip_pg_rho_threshold,
        T, B, ctypes.c_void_p(vs.data_ptr()), ctypes.c_void_p(pg_adv.data_ptr()),
        _backend.current_stream())
    _backend.check(ret, "vtrace_from_log_rhos")
    return VTraceReturns(vs=vs, pg_advantages=pg_adv)


def impala_loss_reference(behavior_logits, target_logits, actions, rewards,
                          discounts, values, bootstrap_value, *,
                          clip_rho_threshold=1.0, clip_c_threshold=1.0,
                          clip_pg_rho_threshold=1.0, baseline_cost=0.5,
                          entropy_cost=0.0006):
    """Composed autograd reference of the fused loss (CPU/any device).

    Returns (total, pg, baseline, entropy) — sums, reference reduction
    (loss_fn.py uses torch.sum everywhere).
    """
    with torch.no_grad():
        log_rhos = (action_log_probs(target_logits, actions)
                    - action_log_probs(behavior_logits, actions))
        vtr = vtrace_reference(log_rhos, discounts, rewards, values.detach(),
                               bootstrap_value, clip_rho_threshold,
                               clip_c_threshold, clip_pg_rho_threshold)
    ce = -action_log_probs(target_logits, actions)
    pg_loss = torch.sum(ce * vtr.pg_advantages)
    baseline_loss = 0.5 * torch.sum((vtr.vs - values) ** 2)
    logp = F.log_softmax(target_logits, dim=-1)
    entropy_loss = torch.sum(logp.exp() * logp)  # sum p log p (neg entropy)
    total = pg_loss + baseline_cost * baseline_loss + entropy_cost * entropy_loss
    return total, pg_loss.detach(), baseline_loss.detach(), entropy_loss.detach()


class _ImpalaFusedLossFn(torch.autograd.Function):
    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.float32)
    def forward(ctx, target_logits, values, behavior_logits, actions, rewards,
                discounts, bootstrap_value, clip_rho, clip_c, clip_pg_rho,
                baseline_cost, entropy_cost, want_vs):
        T, B, A = target_logits.shape
        tl = target_logits.contiguous().float()
        vals = values.contiguous().float()
        bl = behavior_logits.contiguous().float()
        acts = actions.contiguous().long()
        grad_logits = torch.empty_like(tl)
        grad_values = torch.empty_like(vals)
        loss_out = torch.zeros(3, device=tl.device, dtype=torch.float32)
        vs_out = torch.empty_like(vals) if want_vs else None
        c = ctypes.c_void_p
        # locals keep the cast temps alive past the launch (see ops/td.py)
        rew = rewards.contiguous().float()
        dis = discounts.contiguous().float()
        bv = bootstrap_value.contiguous().float()
        ret = _backend.lib().impala_fused_loss(
            c(bl.data_ptr()), c(tl.data_ptr()), c(acts.data_ptr()),
            c(rew.data_ptr()),
            c(dis.data_ptr()),
            c(vals.data_ptr()),
            c(bv.data_ptr()),
            clip_rho, clip_c, clip_pg_rho, baseline_cost, entropy_cost,
            T, B, A, c(grad_logits.data_ptr()), c(grad_values.data_ptr()),
            c(loss_out.data_ptr()),
            c(vs_out.data_ptr()) if vs_out is not None else None,
            _backend.current_stream())
        _backend.check(ret, "impala_fused_loss")
        ctx.save_for_backward(grad_logits, grad_values)
        total = (loss_out[0] + baseline_cost * loss_out[1]
                 + entropy_cost * loss_out[2])
        ctx.mark_non_differentiable(loss_out)
        if vs_out is not None:
            ctx.mark_non_differentiable(vs_out)
            return total, loss_out, vs_out
        return total, loss_out, loss_out.new_zeros(0)

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, g_total, g_loss_out, g_vs):
        grad_logits, grad_values = ctx.saved_tensors
        return (grad_logits * g_total, grad_values * g_total, None, None,
                None, None, None, None, None, None, None, None, None)


def impala_loss(behavior_logits, target_logits, actions, rewards, discounts,
                values, bootstrap_value, *, clip_rho_threshold=1.0,
                clip_c_threshold=1.0, clip_pg_rho_threshold=1.0,
                baseline_cost=0.5, entropy_cost=0.0006, want_vs=False):
    """IMPALA total loss with autograd.  GPU → fused HIP kernel;
    CPU → composed reference.

    Returns (total_loss, components[3] = raw pg/baseline/entropy sums,
    vs or empty).
    """
    if target_logits.is_cuda:
        return _ImpalaFusedLossFn.apply(
            target_logits, values, behavior_logits, actions, rewards,
            discounts, bootstrap_value, clip_rho_threshold, clip_c_threshold,
            clip_pg_rho_threshold, baseline_cost, entropy_cost, want_vs)
    total, pg, bl, ent = impala_loss_reference(
        behavior_logits, target_logits, actions, rewards, discounts, values,
        bootstrap_value, clip_rho_threshold=clip_rho_threshold,
        clip_c_threshold=clip_c_threshold,
        clip_pg_rho_threshold=clip_pg_rho_threshold,
        baseline_cost=baseline_cost, entropy_cost=entropy_cost)
    comps = torch.stack([pg, bl, ent])
    vs = torch.zeros(0)
    if want_vs:
        with torch.no_grad():
            log_rhos = (action_log_probs(target_logits, actions)
                        - action_log_probs(behavior_logits, actions))
            vs = vtrace_reference(log_rhos, discounts, rewards,
                                  values.detach(), bootstrap_value,
                                  clip_rho_threshold, clip_c_threshold,
                                  clip_pg_rho_threshold).vs
    return total, comps, vs
