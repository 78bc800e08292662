"""GAE / discounted-return / n-step scan ops (HIP on GPU, reference on CPU).

Reference semantics: GAE — standard (config-only in the reference,
rl_args.py:338-340); returns — parallel_a3c.py:274-276 / generation.py:142-147;
n-step — replay_buffer.py:230-273.
"""

from __future__ import annotations

import ctypes

import torch

from . import _backend


@torch.no_grad()
def gae_reference(rewards, values, bootstrap_value, discounts, lam: float):
    T = rewards.shape[0]
    adv = torch.zeros_like(bootstrap_value)
    advantages = torch.empty_like(rewards)
    values_tp1 = torch.cat([values[1:], bootstrap_value.unsqueeze(0)], dim=0)
    deltas = rewards + discounts * values_tp1 - values
    for t in range(T - 1, -1, -1):
        adv = deltas[t] + discounts[t] * lam * adv
        advantages[t] = adv
    return advantages, advantages + values


@torch.no_grad()
def gae(rewards, values, bootstrap_value, discounts, lam: float = 0.95):
    """advantages, returns — [T,B] each."""
    if not rewards.is_cuda:
        return gae_reference(rewards, values, bootstrap_value, discounts, lam)
    T, B = rewards.shape
    r = rewards.contiguous().float()
    v = values.contiguous().float()
    bv = bootstrap_value.contiguous().float()
    d = discounts.contiguous().float()
    advantages = torch.empty_like(r)
    returns = torch.empty_like(r)
    c = ctypes.c_void_p
    ret = _backend.lib().gae_scan(
        c(r.data_ptr()), c(v.data_ptr()), c(bv.data_ptr()), c(d.data_ptr()),
        lam, T, B, c(advantages.data_ptr()), c(returns.data_ptr()),
        _backend.current_stream())
    _backend.check(ret, "gae_scan")
    return advantages, returns


@torch.no_grad()
def discounted_returns(rewards, discounts, bootstrap_value=None):
    """R_t = r_t + discount_t * R_{t+1}, seeded with bootstrap (or 0)."""
    if not rewards.is_cuda:
        T = rewards.shape[0]
        acc = (bootstrap_value.clone() if bootstrap_value is not None
               else torch.zeros_like(rewards[0]))
        out = torch.empty_like(rewards)
        for t in range(T - 1, -1, -1):
            acc = rewards[t] + discounts[t] * acc
            out[t] = acc
        return out
    T, B = rewards.shape
    r = rewards.contiguous().float()
    d = discounts.contiguous().float()
    out = torch.empty_like(r)
    c = ctypes.c_void_p
    bv = (bootstrap_value.contiguous().float()
          if bootstrap_value is not None else None)
    ret = _backend.lib().discounted_returns(
        c(r.data_ptr()), c(d.data_ptr()),
        c(bv.data_ptr()) if bv is not None else None,
        T, B, c(out.data_ptr()), _backend.current_stream())
    _backend.check(ret, "discounted_returns")
    return out


@torch.no_grad()
def nstep_fold(rewards, dones, gamma: float, n: int):
    """Per-(t,b): folded n-step reward, done-inside-window flag, steps used.

    folded_r[t] = sum_{k<m} gamma^k r[t+k] with m = min(n, steps to first
    done or end of chunk); matches MultiStepReplayBuffer insert folding.
    """
    if not rewards.is_cuda:
        T, B = rewards.shape
        fr = torch.zeros_like(rewards)
        fd = torch.zeros_like(rewards)
        su = torch.zeros(T, B, dtype=torch.int32)
        for t in range(T):
            for b in range(B):
                acc, g, m, done = 0.0, 1.0, 0, 0.0
                for k in range(n):
                    if t + k >= T:
                        break
                    acc += g * float(rewards[t + k, b])
                    g *= gamma
                    m = k + 1
                    if float(dones[t + k, b]) != 0.0:
                        done = 1.0
                        break
                fr[t, b], fd[t, b], su[t, b] = acc, done, m
        return fr, fd, su
    T, B = rewards.shape
    r = rewards.contiguous().float()
    d = dones.contiguous().float()
    fr = torch.empty_like(r)
    fd = torch.empty_like(r)
    su = torch.empty(T, B, dtype=torch.int32, device=r.device)
    c = ctypes.c_void_p
    ret = _backend.lib().nstep_fold(
        c(r.data_ptr()), c(d.data_ptr()), gamma, n, T, B,
        c(fr.data_ptr()), c(fd.data_ptr()), c(su.data_ptr()),
        _backend.current_stream())
    _backend.check(ret, "nstep_fold")
    return fr, fd, su
