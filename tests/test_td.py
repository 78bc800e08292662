"""Fused TD loss vs composed reference (CPU golden + GPU parity)."""

import pytest
import torch

from scalerl_amd.ops import fused_td_loss, td_loss_reference


def _inputs(B=32, A=6, device="cpu", seed=0):
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(B, A, generator=g)
    qno = torch.randn(B, A, generator=g)
    qnt = torch.randn(B, A, generator=g)
    a = torch.randint(0, A, (B,), generator=g)
    r = torch.randn(B, generator=g)
    d = 0.99 * (torch.rand(B, generator=g) > 0.1).float()
    return [x.to(device) for x in (q, qno, qnt, a, r, d)]


def test_td_reference_double_vs_vanilla_differ():
    q, qno, qnt, a, r, d = _inputs()
    l_double, _ = td_loss_reference(q, qno, qnt, a, r, d)
    l_vanilla, _ = td_loss_reference(q, None, qnt, a, r, d)
    assert l_double.item() != pytest.approx(l_vanilla.item())


def test_td_reference_golden_single():
    # hand-computed: Q(s,a)=2, target = 1 + 0.5*3 = 2.5, td=-0.5, mse=0.25
    q = torch.tensor([[2.0, 0.0]])
    qnt = torch.tensor([[3.0, 1.0]])
    a = torch.tensor([0])
    r = torch.tensor([1.0])
    d = torch.tensor([0.5])
    loss, tda = td_loss_reference(q, None, qnt, a, r, d)
    assert loss.item() == pytest.approx(0.25)
    assert tda.item() == pytest.approx(0.5)


@pytest.mark.gpu
@pytest.mark.parametrize("huber", [False, True])
@pytest.mark.parametrize("per", [False, True])
def test_fused_td_gpu_matches_reference(huber, per):
    dev = "cuda:0"
    q, qno, qnt, a, r, d = _inputs(device=dev)
    prios = p_total = p_min = None
    weights = None
    if per:
        g = torch.Generator().manual_seed(1)
        prios_cpu = torch.rand(32, generator=g) + 0.1
        prios = prios_cpu.to(dev)
        p_total = torch.tensor([10.0], device=dev)
        p_min = torch.tensor([0.1], device=dev)
        from scalerl_amd.ops import per_is_weights
        weights = per_is_weights(prios_cpu, torch.tensor(10.0),
                                 torch.tensor(0.1), 1000, 0.4)
    qg = q.clone().requires_grad_()
    loss, tda = fused_td_loss(qg, qno, qnt, a, r, d, prios=prios,
                              p_total=p_total, p_min=p_min, beta=0.4,
                              replay_size=1000, huber=huber)
    loss.backward()

    qc = q.cpu().requires_grad_()
    loss_ref, tda_ref = td_loss_reference(qc, qno.cpu(), qnt.cpu(), a.cpu(),
                                          r.cpu(), d.cpu(), weights, huber)
    loss_ref.backward()
    torch.testing.assert_close(loss.cpu(), loss_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(tda.cpu(), tda_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(qg.grad.cpu(), qc.grad, rtol=1e-4, atol=1e-5)


def test_ppo_loss_reference_grads():
    from scalerl_amd.ops import ppo_fused_loss
    g = torch.Generator().manual_seed(0)
    N, A = 64, 5
    logits = torch.randn(N, A, generator=g, requires_grad=True)
    values = torch.randn(N, generator=g, requires_grad=True)
    actions = torch.randint(0, A, (N,), generator=g)
    old_logp = torch.log_softmax(torch.randn(N, A, generator=g), -1).gather(
        1, actions.unsqueeze(1)).squeeze(1)
    adv = torch.randn(N, generator=g)
    rets = torch.randn(N, generator=g)
    total, comps = ppo_fused_loss(logits, values, actions, old_logp, adv,
                                  rets)
    total.backward()
    assert logits.grad.abs().sum() > 0 and values.grad.abs().sum() > 0


@pytest.mark.gpu
def test_ppo_fused_loss_matches_reference():
    from scalerl_amd.ops import ppo_fused_loss
    from scalerl_amd.ops.ppo import ppo_loss_reference
    dev = "cuda:0"
    g = torch.Generator().manual_seed(1)
    N, A = 512, 4
    logits = torch.randn(N, A, generator=g)
    values = torch.randn(N, generator=g)
    actions = torch.randint(0, A, (N,), generator=g)
    old_logp = torch.log_softmax(torch.randn(N, A, generator=g), -1).gather(
        1, actions.unsqueeze(1)).squeeze(1)
    adv = torch.randn(N, generator=g)
    rets = torch.randn(N, generator=g)

    lg = logits.to(dev).requires_grad_()
    vg = values.to(dev).requires_grad_()
    total, comps = ppo_fused_loss(lg, vg, actions.to(dev), old_logp.to(dev),
                                  adv.to(dev), rets.to(dev),
                                  clip_eps=0.2, vcoef=0.5, ecoef=0.01)
    total.backward()

    lc = logits.clone().requires_grad_()
    vc = values.clone().requires_grad_()
    total_ref, pg, v, ent = ppo_loss_reference(lc, actions, old_logp, adv,
                                               rets, vc, 0.2, 0.5, 0.01)
    total_ref.backward()
    torch.testing.assert_close(total.cpu(), total_ref.detach(), rtol=1e-3,
                               atol=1e-4)
    torch.testing.assert_close(comps[0].cpu(), pg, rtol=1e-3, atol=1e-4)
    torch.testing.assert_close(comps[1].cpu(), v, rtol=1e-3, atol=1e-4)
    torch.testing.assert_close(comps[2].cpu(), ent, rtol=1e-3, atol=1e-4)
    torch.testing.assert_close(lg.grad.cpu(), lc.grad, rtol=1e-3, atol=1e-5)
    torch.testing.assert_close(vg.grad.cpu(), vc.grad, rtol=1e-3, atol=1e-5)


def test_ppo_kernel_formula_emulation_matches_autograd():
    """Emulate ppo_fused_loss_kernel's per-row math (incl. the min-branch
    gradient selection) in Python and compare against torch autograd —
    validates the kernel's gradient logic without a GPU."""
    g = torch.Generator().manual_seed(3)
    N, A = 256, 6
    clip_eps, vcoef, ecoef = 0.2, 0.5, 0.01
    logits = torch.randn(N, A, generator=g)
    values = torch.randn(N, generator=g)
    actions = torch.randint(0, A, (N,), generator=g)
    old_logp = torch.log_softmax(torch.randn(N, A, generator=g), -1).gather(
        1, actions.unsqueeze(1)).squeeze(1)
    # spread ratios well past both clip boundaries
    old_logp = old_logp + torch.linspace(-1.5, 1.5, N)
    adv = torch.randn(N, generator=g)
    rets = torch.randn(N, generator=g)

    # ---- emulation of the kernel's per-row formulas ----
    lse = torch.logsumexp(logits, dim=1)
    p = torch.softmax(logits, dim=1)
    plogp = (p * (logits - lse.unsqueeze(1))).sum(1)
    logp_a = logits.gather(1, actions.unsqueeze(1)).squeeze(1) - lse
    ratio = torch.exp(logp_a - old_logp)
    rc = ratio.clamp(1 - clip_eps, 1 + clip_eps)
    s1, s2 = ratio * adv, rc * adv
    pg = -torch.min(s1, s2)
    dmin_dlogpa = torch.where(
        (s1 <= s2) | ((ratio > 1 - clip_eps) & (ratio < 1 + clip_eps)),
        -s1, torch.zeros(()))
    onehot = torch.nn.functional.one_hot(actions, A).float()
    grad_logits = (dmin_dlogpa.unsqueeze(1) * (onehot - p)
                   + ecoef * p * ((logits - lse.unsqueeze(1))
                                  - plogp.unsqueeze(1))) / N
    grad_values = vcoef * 2 * (values - rets) / N

    # ---- autograd reference (the composed torch ops) ----
    from scalerl_amd.ops.ppo import ppo_loss_reference
    lg = logits.clone().requires_grad_()
    vg = values.clone().requires_grad_()
    total, *_ = ppo_loss_reference(lg, actions, old_logp, adv, rets, vg,
                                   clip_eps, vcoef, ecoef)
    total.backward()
    torch.testing.assert_close(grad_logits, lg.grad, rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(grad_values, vg.grad, rtol=1e-4, atol=1e-6)
    # loss components too
    assert abs(float(pg.mean()) - float(
        -(torch.min(s1, s2)).mean())) < 1e-6
