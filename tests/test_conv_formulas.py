"""CPU emulation of the conv kernels' index formulas vs torch autograd.

The MFMA fragment mapping is hardware-validated (test_conv_experimental);
these tests validate the OTHER half of the backward kernels — the
implicit-GEMM index decode and validity masks in csrc/conv_atari.hip —
by executing the exact same formulas in Python on small shapes.  A bug in
the kernels' gather math would show here without needing a GPU."""

import numpy as np
import pytest
import torch
import torch.nn.functional as F

SHAPES = {1: ((4, 12, 12), (5, 4, 8, 8), 4),   # scaled-down conv1 geometry
          2: ((3, 10, 10), (6, 3, 4, 4), 2),
          3: ((4, 9, 9), (6, 4, 3, 3), 1)}


def _out_hw(ih, iw, kh, kw, s):
    return (ih - kh) // s + 1, (iw - kw) // s + 1


def emu_fwd(x, w, stride):
    """The fwd kernel's formula: out[p, k_out] = sum_k A[p,k]*B[k,k_out]
    with A gathered as in conv_fwd_kernel."""
    N, C, IH, IW = x.shape
    K, _, KH, KW = w.shape
    OH, OW = _out_hw(IH, IW, KH, KW, stride)
    M, KDIM = N * OH * OW, C * KH * KW
    out = np.zeros((M, K), dtype=np.float64)
    wf = w.reshape(K, KDIM)
    for p in range(M):
        n, rem = divmod(p, OH * OW)
        oy, ox = divmod(rem, OW)
        patch = np.empty(KDIM)
        for k in range(KDIM):
            c, kr = divmod(k, KH * KW)
            ky, kx = divmod(kr, KW)
            patch[k] = x[n, c, oy * stride + ky, ox * stride + kx]
        out[p] = patch @ wf.T
    return out.reshape(N, OH, OW, K).transpose(0, 3, 1, 2)


def emu_wgrad(x, dy, stride, kshape):
    """conv_wgrad_kernel's formula."""
    K, C, KH, KW = kshape
    N, _, IH, IW = x.shape
    OH, OW = _out_hw(IH, IW, KH, KW, stride)
    KDIM = C * KH * KW
    dw = np.zeros((K, KDIM))
    for p in range(N * OH * OW):
        n, rem = divmod(p, OH * OW)
        oy, ox = divmod(rem, OW)
        for k in range(KDIM):
            c, kr = divmod(k, KH * KW)
            ky, kx = divmod(kr, KW)
            dw[:, k] += dy[n, :, oy, ox] * x[n, c, oy * stride + ky,
                                             ox * stride + kx]
    return dw.reshape(K, C, KH, KW)


def emu_dgrad(dy, w, stride, in_shape):
    """conv_dgrad_kernel's formula incl. the stride-validity mask."""
    C, IH, IW = in_shape
    K, _, KH, KW = w.shape
    N = dy.shape[0]
    OH, OW = dy.shape[2], dy.shape[3]
    dx = np.zeros((N, C, IH, IW))
    for p in range(N * IH * IW):
        n, rem = divmod(p, IH * IW)
        iy, ix = divmod(rem, IW)
        for k in range(K * KH * KW):
            kout, kr = divmod(k, KH * KW)
            ky, kx = divmod(kr, KW)
            ty, tx = iy - ky, ix - kx
            if ty >= 0 and tx >= 0 and ty % stride == 0 and tx % stride == 0:
                oy, ox = ty // stride, tx // stride
                if oy < OH and ox < OW:
                    dx[n, :, iy, ix] += dy[n, kout, oy, ox] * w[kout, :, ky, kx]
    return dx


@pytest.mark.parametrize("layer", [1, 2, 3])
def test_fwd_formula_matches_torch(layer):
    rng = np.random.default_rng(layer)
    in_shape, w_shape, stride = SHAPES[layer]
    x = rng.standard_normal((2, *in_shape))
    w = rng.standard_normal(w_shape) * 0.2
    got = emu_fwd(x, w, stride)
    want = F.conv2d(torch.tensor(x), torch.tensor(w), stride=stride).numpy()
    np.testing.assert_allclose(got, want, rtol=1e-6, atol=1e-8)


@pytest.mark.parametrize("layer", [1, 2, 3])
def test_wgrad_formula_matches_torch(layer):
    rng = np.random.default_rng(10 + layer)
    in_shape, w_shape, stride = SHAPES[layer]
    x = torch.tensor(rng.standard_normal((2, *in_shape)))
    w = torch.tensor(rng.standard_normal(w_shape) * 0.2, requires_grad=True)
    out = F.conv2d(x, w, stride=stride)
    dy = torch.tensor(rng.standard_normal(tuple(out.shape)))
    (out * dy).sum().backward()
    got = emu_wgrad(x.numpy(), dy.numpy(), stride, w_shape)
    np.testing.assert_allclose(got, w.grad.numpy(), rtol=1e-6, atol=1e-8)


@pytest.mark.parametrize("layer", [2, 3])
def test_dgrad_formula_matches_torch(layer):
    rng = np.random.default_rng(20 + layer)
    in_shape, w_shape, stride = SHAPES[layer]
    x = torch.tensor(rng.standard_normal((2, *in_shape)), requires_grad=True)
    w = torch.tensor(rng.standard_normal(w_shape) * 0.2)
    out = F.conv2d(x, w, stride=stride)
    dy = torch.tensor(rng.standard_normal(tuple(out.shape)))
    (out * dy).sum().backward()
    got = emu_dgrad(dy.numpy(), w.numpy(), stride, in_shape)
    np.testing.assert_allclose(got, x.grad.numpy(), rtol=1e-6, atol=1e-8)
