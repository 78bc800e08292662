"""Property-based tests (hypothesis) for the scan ops and the sum tree —
the CPU oracles these properties pin down are the same code the GPU
kernels are tested against."""

import hypothesis.strategies as st
import numpy as np
import torch
from hypothesis import given, settings

from scalerl_amd.ops import (SumTree, discounted_returns, gae, nstep_fold,
                             vtrace_reference)


@settings(max_examples=30, deadline=None)
@given(T=st.integers(1, 20), B=st.integers(1, 5),
       gamma=st.floats(0.0, 0.999), seed=st.integers(0, 10_000))
def test_discounted_returns_satisfies_bellman(T, B, gamma, seed):
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(T, B, generator=g)
    d = torch.full((T, B), gamma)
    out = discounted_returns(r, d)
    for t in range(T - 1):
        torch.testing.assert_close(out[t], r[t] + gamma * out[t + 1],
                                   rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(out[T - 1], r[T - 1], rtol=1e-4, atol=1e-4)


@settings(max_examples=30, deadline=None)
@given(T=st.integers(1, 16), B=st.integers(1, 4), seed=st.integers(0, 10_000),
       lam=st.floats(0.0, 1.0))
def test_gae_advantage_bellman_recursion(T, B, seed, lam):
    g = torch.Generator().manual_seed(seed)
    r = torch.randn(T, B, generator=g)
    v = torch.randn(T, B, generator=g)
    bv = torch.randn(B, generator=g)
    d = torch.full((T, B), 0.95)
    adv, ret = gae(r, v, bv, d, lam=lam)
    v_tp1 = torch.cat([v[1:], bv.unsqueeze(0)])
    delta = r + d * v_tp1 - v
    acc = torch.zeros(B)
    for t in range(T - 1, -1, -1):
        acc = delta[t] + 0.95 * lam * acc
        torch.testing.assert_close(adv[t], acc, rtol=1e-4, atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(T=st.integers(1, 12), n=st.integers(1, 5), seed=st.integers(0, 9999))
def test_nstep_fold_window_invariants(T, n, seed):
    g = torch.Generator().manual_seed(seed)
    r = torch.rand(T, 1, generator=g)
    dones = (torch.rand(T, 1, generator=g) < 0.25).float()
    fr, fd, su = nstep_fold(r, dones, 0.9, n)
    for t in range(T):
        m = int(su[t, 0])
        assert 1 <= m <= min(n, T - t)
        # window stops at the first done or the horizon
        inner_dones = [float(dones[t + k, 0]) for k in range(m - 1)]
        assert all(d == 0.0 for d in inner_dones)
        expect = sum(0.9 ** k * float(r[t + k, 0]) for k in range(m))
        assert abs(float(fr[t, 0]) - expect) < 1e-5


@settings(max_examples=20, deadline=None)
@given(seed=st.integers(0, 9999), k=st.integers(1, 60))
def test_vtrace_on_policy_reduces_to_lambda_return(seed, k):
    """log_rhos = 0 → vs equals the Monte-Carlo/bootstrap return."""
    g = torch.Generator().manual_seed(seed)
    T, B = (k % 12) + 1, 2
    r = torch.randn(T, B, generator=g)
    v = torch.randn(T, B, generator=g)
    bv = torch.randn(B, generator=g)
    d = torch.full((T, B), 0.9)
    vtr = vtrace_reference(torch.zeros(T, B), d, r, v, bv)
    expect = discounted_returns(r, d, bv)
    torch.testing.assert_close(vtr.vs, expect, rtol=1e-4, atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(seed=st.integers(0, 9999), n=st.integers(1, 100))
def test_sumtree_total_equals_leaf_sum(seed, n):
    g = torch.Generator().manual_seed(seed)
    t = SumTree(128)
    prios = torch.rand(n, generator=g) + 0.01
    t.update(torch.arange(n), prios, max_idx=n)
    assert abs(float(t.total) - float(prios.sum())) < 1e-3
    # point update keeps the invariant
    t.update(torch.tensor([0]), torch.tensor([5.0]), max_idx=n)
    expect = float(prios[1:].sum()) + 5.0
    assert abs(float(t.total) - expect) < 1e-3
    assert float(t.min_leaf()) <= float(prios.min()) + 5.0
