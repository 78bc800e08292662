"""Prioritized-replay sum tree: exactness + sampling distribution + GPU parity."""

import pytest
import torch

from scalerl_amd.ops import SumTree, per_is_weights


def test_sumtree_update_total():
    t = SumTree(10)
    t.update(torch.arange(5), torch.tensor([1.0, 2.0, 3.0, 4.0, 5.0]), max_idx=5)
    assert abs(float(t.total) - 15.0) < 1e-6
    t.update(torch.tensor([2]), torch.tensor([10.0]), max_idx=5)
    assert abs(float(t.total) - 22.0) < 1e-6
    assert abs(float(t.min_leaf()) - 1.0) < 1e-6


def test_sumtree_sample_proportional():
    torch.manual_seed(0)
    t = SumTree(8)
    prios = torch.tensor([1.0, 0.0, 0.0, 9.0])
    t.update(torch.arange(4), prios, max_idx=4)
    counts = torch.zeros(4)
    for _ in range(200):
        idx, p = t.sample(16)
        for i in idx:
            counts[int(i)] += 1
    frac = counts / counts.sum()
    assert abs(frac[3] - 0.9) < 0.05
    assert abs(frac[0] - 0.1) < 0.05
    assert counts[1] == 0 and counts[2] == 0


def test_is_weights_formula():
    prios = torch.tensor([1.0, 4.0])
    w = per_is_weights(prios, p_total=torch.tensor(5.0),
                       p_min=torch.tensor(1.0), replay_size=100, beta=0.5)
    # w_i = (N p_i/total)^-b; max at p_min → w(p_min)=1
    assert abs(float(w[0]) - 1.0) < 1e-6
    assert abs(float(w[1]) - 0.5) < 1e-6


@pytest.mark.gpu
def test_sumtree_gpu_matches_cpu():
    g = torch.Generator().manual_seed(0)
    prios = torch.rand(1000, generator=g) + 0.01
    idx = torch.arange(1000)
    cpu = SumTree(1024)
    gpu = SumTree(1024, device="cuda:0")
    cpu.update(idx, prios, max_idx=1000)
    gpu.update(idx, prios, max_idx=1000)
    assert abs(float(cpu.total) - float(gpu.total)) < 1e-2
    assert abs(float(cpu.min_leaf()) - float(gpu.min_leaf())) < 1e-6
    # duplicate-index batched update must stay consistent
    dup_idx = torch.tensor([5, 5, 5, 7])
    dup_p = torch.tensor([1.0, 2.0, 3.0, 4.0])
    gpu.update(dup_idx, dup_p, max_idx=1000)
    leaf5 = float(gpu.tree[gpu.M + 5])
    assert leaf5 in (1.0, 2.0, 3.0)  # one of the writes wins...
    # ...and the root equals the true sum of leaves regardless
    root = float(gpu.total)
    leafsum = float(gpu.tree[gpu.M:gpu.M + 1024].sum())
    assert abs(root - leafsum) < 1e-2


@pytest.mark.gpu
def test_sumtree_gpu_sample_distribution():
    torch.manual_seed(1)
    t = SumTree(16, device="cuda:0")
    prios = torch.tensor([1.0, 0.0, 0.0, 9.0])
    t.update(torch.arange(4), prios, max_idx=4)
    counts = torch.zeros(4)
    for _ in range(100):
        idx, p = t.sample(64)
        for i in idx.cpu():
            counts[int(i)] += 1
    frac = counts / counts.sum()
    assert abs(frac[3] - 0.9) < 0.05
    assert counts[1] == 0 and counts[2] == 0
