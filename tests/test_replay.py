"""Replay buffers: ring semantics, n-step folds, PER priorities, state dicts."""

import pytest
import torch

from scalerl_amd.data import (MultiStepReplayBuffer, PrioritizedReplayBuffer,
                              ReplayBuffer)


def test_ring_wraparound():
    b = ReplayBuffer(8, (2,), gamma=0.9)
    for i in range(12):
        b.add([i, i], i % 2, float(i), [i + 1, i + 1], 0.0)
    assert len(b) == 8
    # oldest 4 overwritten: rewards present are 4..11
    present = set(b.reward.tolist())
    assert present == {float(i) for i in range(4, 12)}


def test_discount_column():
    b = ReplayBuffer(4, (1,), gamma=0.5)
    b.add([0.0], 0, 1.0, [1.0], 0.0)
    b.add([0.0], 0, 1.0, [1.0], 1.0)
    assert b.discount[0].item() == pytest.approx(0.5)
    assert b.discount[1].item() == pytest.approx(0.0)


def test_nstep_buffer_matches_manual_fold():
    m = MultiStepReplayBuffer(100, (1,), n_steps=3, gamma=0.5)
    rewards = [1.0, 2.0, 4.0, 8.0]
    for i, r in enumerate(rewards):
        m.add([float(i)], 0, r, [float(i + 1)], i == 3)
    # t0: 1 + 0.5*2 + 0.25*4 = 3; t1: 2+2+2=... 2 + 0.5*4 + 0.25*8 = 6
    assert m.reward[0].item() == pytest.approx(3.0)
    assert m.reward[1].item() == pytest.approx(6.0)
    # t1 window hits done at i=3 → discount 0
    assert m.discount[1].item() == pytest.approx(0.0)
    # t0 window is 3 full steps, no done → gamma^3
    assert m.discount[0].item() == pytest.approx(0.125)
    # next_obs of t0 = obs after 3 steps = [3]
    assert m.next_obs[0].item() == pytest.approx(3.0)


def test_nstep_chunk_matches_single(device="cpu"):
    torch.manual_seed(0)
    T, B, n = 12, 3, 3
    rewards = torch.rand(T, B)
    dones = (torch.rand(T, B) < 0.15).float()
    obs = torch.arange(T * B, dtype=torch.float32).reshape(T, B, 1)
    actions = torch.randint(0, 4, (T, B))
    boot = torch.full((n, B, 1), 999.0)

    chunked = MultiStepReplayBuffer(100, (1,), n_steps=n, gamma=0.9)
    chunked.add_chunk(obs, actions, rewards, dones, boot)

    single = MultiStepReplayBuffer(100, (1,), n_steps=n, gamma=0.9,
                                   num_envs=B)
    # feed column-major to mimic per-env streams; chunk layout is [T,B] so
    # compare per (t, b) after sorting by obs id
    for b in range(B):
        for t in range(T):
            nxt = obs[t + 1, b] if t + 1 < T else boot[0, b]
            single.add(obs[t, b], int(actions[t, b]), float(rewards[t, b]),
                       nxt, float(dones[t, b]), env_id=b)
    # the chunked buffer folds every row (windows truncate at chunk end with
    # correct discounts); compare the entries both structures share: windows
    # fully inside the chunk and not crossing a done in a different way
    def table(buf):
        d = {}
        for i in range(len(buf)):
            d[float(buf.obs[i])] = (round(float(buf.reward[i]), 5),
                                    round(float(buf.discount[i]), 5))
        return d
    tc, ts = table(chunked), table(single)
    # all single-path entries with full windows must match the chunked fold
    matched = 0
    for k, v in ts.items():
        if k in tc and ts[k][1] > 0:  # full (non-terminal) windows
            t = int(k) // B
            if t + n <= T:  # window inside chunk
                assert tc[k] == v, (k, tc[k], v)
                matched += 1
    assert matched > 5


def test_per_buffer_roundtrip_and_state():
    p = PrioritizedReplayBuffer(32, (2,), alpha=0.5, seed=0)
    for i in range(16):
        p.add([i, i], 0, float(i), [i, i], 0.0)
    batch, idx, prio, total, pmin = p.sample_with_priorities(8)
    assert batch["obs"].shape == (8, 2)
    p.update_priorities(idx, torch.rand(8) * 5)
    sd = p.state_dict()
    q = PrioritizedReplayBuffer(32, (2,), alpha=0.5)
    q.load_state_dict(sd)
    assert float(q.tree.total) == pytest.approx(float(p.tree.total))
    assert len(q) == len(p)


def test_per_with_nstep_composes():
    """PER + n_steps>1 fold together (reference: PrioritizedReplayBuffer
    subclasses MultiStepReplayBuffer) — ADVICE r1 item 3."""
    n, gamma = 3, 0.9
    p = PrioritizedReplayBuffer(64, (1,), alpha=0.6, n_steps=n,
                                gamma=gamma, seed=0)
    rewards = [1.0, 2.0, 4.0, 8.0, 16.0]
    for i, r in enumerate(rewards):
        p.add([float(i)], 0, r, [float(i + 1)], 0.0)
    # first full window: r0 + g*r1 + g^2*r2
    want = rewards[0] + gamma * rewards[1] + gamma ** 2 * rewards[2]
    assert float(p.reward[0]) == pytest.approx(want)
    assert float(p.discount[0]) == pytest.approx(gamma ** n)
    assert float(p.next_obs[0, 0]) == 3.0  # obs n steps ahead
    # priorities assigned for the folded insert
    batch, idx, prio, total, pmin = p.sample_with_priorities(2)
    assert (prio > 0).all()


def test_segment_tree_reference_api():
    """SumSegmentTree/MinSegmentTree parity surface
    (reference segment_tree.py:7-197): point updates, range reduce,
    prefix-sum descent — cross-checked against numpy."""
    import numpy as np
    from scalerl_amd.data.segment_tree import (MinSegmentTree,
                                               SumSegmentTree)
    rng = np.random.default_rng(0)
    cap = 64
    st, mt = SumSegmentTree(cap), MinSegmentTree(cap)
    vals = np.zeros(cap)
    # touch every slot once (MinSegmentTree's neutral is +inf), then
    # random overwrites
    idxs = list(range(cap)) + [int(rng.integers(cap)) for _ in range(200)]
    for i in idxs:
        v = float(rng.random() * 5)
        st[i] = v
        mt[i] = v
        vals[i] = v
    assert st.sum() == pytest.approx(vals.sum())
    assert st.sum(5, 20) == pytest.approx(vals[5:20].sum())
    assert mt.min() == pytest.approx(vals.min())
    assert mt.min(3, 40) == pytest.approx(vals[3:40].min())
    assert st[7] == pytest.approx(vals[7])
    # prefix descent: largest i with cumsum[:i] <= p
    for p in (0.0, 0.3 * vals.sum(), 0.9999 * vals.sum()):
        i = st.find_prefixsum_idx(p)
        cs = np.cumsum(vals)
        want = int(np.searchsorted(cs, p, side="right"))
        assert i == min(want, cap - 1), (p, i, want)
