"""Fused optimizers vs torch.optim on identical flat buffers."""

import pytest
import torch

from scalerl_amd.ops import FusedAdam, FusedRMSprop, clip_grad_norm_, fused_polyak_


def _run_pair(device, opt_name, steps=5, n=1037):
    g = torch.Generator().manual_seed(0)
    p0 = torch.randn(n, generator=g)
    grads = [torch.randn(n, generator=g) for _ in range(steps)]

    ref_p = p0.clone().requires_grad_()
    if opt_name == "rmsprop":
        ref_opt = torch.optim.RMSprop([ref_p], lr=1e-3, alpha=0.99, eps=0.01,
                                      momentum=0.3)
        fused = FusedRMSprop(p0.clone().to(device), lr=1e-3, alpha=0.99,
                             eps=0.01, momentum=0.3)
    else:
        ref_opt = torch.optim.Adam([ref_p], lr=1e-3)
        fused = FusedAdam(p0.clone().to(device), lr=1e-3)

    for gr in grads:
        ref_p.grad = gr.clone()
        ref_opt.step()
        fused.step(gr.to(device))
    torch.testing.assert_close(fused.param.cpu(), ref_p.detach(),
                               rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("opt_name", ["rmsprop", "adam"])
def test_fused_optim_cpu_matches_torch(opt_name):
    _run_pair("cpu", opt_name)


@pytest.mark.gpu
@pytest.mark.parametrize("opt_name", ["rmsprop", "adam"])
def test_fused_optim_gpu_matches_torch(opt_name):
    _run_pair("cuda:0", opt_name)


def test_polyak_cpu():
    a = torch.zeros(100)
    b = torch.ones(100)
    fused_polyak_(a, b, 0.25)
    torch.testing.assert_close(a, torch.full((100,), 0.25))


@pytest.mark.gpu
def test_polyak_gpu():
    a = torch.zeros(1000, device="cuda:0")
    b = torch.ones(1000, device="cuda:0")
    fused_polyak_(a, b, 0.25)
    torch.testing.assert_close(a.cpu(), torch.full((1000,), 0.25))


def test_clip_grad_norm_cpu():
    g = torch.ones(100) * 10  # norm = 100
    clip_grad_norm_(g, 40.0)
    assert abs(g.norm().item() - 40.0) < 1e-3
    g2 = torch.ones(4)  # norm = 2 < 40, untouched
    clip_grad_norm_(g2, 40.0)
    torch.testing.assert_close(g2, torch.ones(4))


@pytest.mark.gpu
def test_clip_grad_norm_gpu():
    g = (torch.ones(100000, device="cuda:0") * 10)
    clip_grad_norm_(g, 40.0)
    assert abs(g.norm().item() - 40.0) < 1e-2
    g2 = torch.ones(1000, device="cuda:0")
    clip_grad_norm_(g2, 100.0)
    torch.testing.assert_close(g2.cpu(), torch.ones(1000))
