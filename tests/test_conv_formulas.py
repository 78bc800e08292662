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


# ---- v2 kernels (csrc/conv_fwd.hip): exact per-lane/per-fragment emulation
# of the LDS-staged forward index math at the REAL shapes, vs F.conv2d.
# (The v1 emulations above cover the backward kernels' gather formulas,
# which conv_bwd.hip reuses.)

def _emulate_conv1_v2(x_u8, w, bias):
    N = x_u8.shape[0]
    C, IH, IW, OW, OYB, IYB, PITCH, MPX, KDIM = 4, 84, 84, 20, 10, 44, 96, 200, 256
    out = np.zeros((N, 32, 20, 20), dtype=np.float32)
    xb = torch.from_numpy(x_u8.astype(np.float32) / 255.0) \
        .to(torch.bfloat16).float().numpy()
    wb = torch.from_numpy(w.reshape(32, KDIM)).to(torch.bfloat16).float().numpy()
    for n in range(N):
        for ob in range(2):
            oy0 = ob * OYB
            img = np.zeros((C, IYB, PITCH), dtype=np.float32)
            img[:, :, :IW] = xb[n][:, oy0 * 4:oy0 * 4 + IYB, :]
            imgf = img.reshape(-1)
            out_lds = np.zeros((32, MPX), dtype=np.float32)
            for t in range(((MPX + 15) // 16) * 2):
                mt, nt = t >> 1, t & 1
                acc = np.zeros((16, 16), dtype=np.float32)
                for kt in range(KDIM // 32):
                    A = np.zeros((16, 32), dtype=np.float32)
                    B = np.zeros((32, 16), dtype=np.float32)
                    for lr in range(16):
                        px = mt * 16 + lr
                        if px < MPX:
                            oy_l, ox = px // OW, px % OW
                            for g in range(4):
                                k0 = kt * 32 + g * 8
                                c, ky = k0 >> 6, (k0 & 63) >> 3
                                off = (c * IYB + oy_l * 4 + ky) * PITCH + ox * 4
                                A[lr, g * 8:(g + 1) * 8] = imgf[off:off + 8]
                        ch = nt * 16 + lr
                        for g in range(4):
                            k0 = kt * 32 + g * 8
                            B[g * 8:(g + 1) * 8, lr] = wb[ch, k0:k0 + 8]
                    acc += A @ B
                for lr in range(16):
                    for g in range(4):
                        for r in range(4):
                            row = mt * 16 + g * 4 + r
                            if row < MPX:
                                out_lds[nt * 16 + lr, row] = acc[g * 4 + r, lr]
            for ch in range(32):
                v = np.maximum(out_lds[ch] + bias[ch], 0.0)
                out[n, ch].reshape(-1)[oy0 * OW:oy0 * OW + MPX] = v
    return out


def test_conv1_v2_fwd_formula_matches_torch():
    np.random.seed(0)
    x = np.random.randint(0, 256, (2, 4, 84, 84)).astype(np.uint8)
    w = (np.random.randn(32, 4, 8, 8) * 0.1).astype(np.float32)
    b = np.random.randn(32).astype(np.float32)
    xt = torch.from_numpy(x.astype(np.float32) / 255.0).to(torch.bfloat16).float()
    wt = torch.from_numpy(w).to(torch.bfloat16).float()
    ref = F.relu(F.conv2d(xt, wt, torch.from_numpy(b), stride=4)).numpy()
    got = _emulate_conv1_v2(x, w, b)
    np.testing.assert_allclose(got, ref, rtol=2e-2, atol=2e-2)


def test_conv2_v2_fwd_formula_matches_torch():
    np.random.seed(1)
    N, C, IH, IW, OW, OH = 2, 32, 20, 20, 9, 9
    PITCH, MPX, KDIM, KOUT = 24, 81, 512, 64
    x = np.random.randn(N, C, IH, IW).astype(np.float32)
    w = (np.random.randn(KOUT, C, 4, 4) * 0.1).astype(np.float32)
    b = np.random.randn(KOUT).astype(np.float32)
    xb = torch.from_numpy(x).to(torch.bfloat16).float().numpy()
    wb = torch.from_numpy(w.reshape(KOUT, KDIM)).to(torch.bfloat16).float().numpy()
    ref = F.relu(F.conv2d(torch.from_numpy(xb),
                          torch.from_numpy(wb.reshape(KOUT, C, 4, 4)),
                          torch.from_numpy(b), stride=2)).numpy()
    out = np.zeros((N, KOUT, OH, OW), dtype=np.float32)
    for n in range(N):
        img = np.zeros((C, IH, PITCH), dtype=np.float32)
        img[:, :, :IW] = xb[n]
        imgf = img.reshape(-1)
        out_lds = np.zeros((KOUT, MPX), dtype=np.float32)
        MT, NT = (MPX + 15) // 16, KOUT // 16
        for t in range(MT * NT):
            mt, nt = t // NT, t % NT
            acc = np.zeros((16, 16), dtype=np.float32)
            for kt in range(KDIM // 32):
                A = np.zeros((16, 32), dtype=np.float32)
                B = np.zeros((32, 16), dtype=np.float32)
                for lr in range(16):
                    px = mt * 16 + lr
                    if px < MPX:
                        oy, ox = px // OW, px % OW
                        for g in range(4):
                            k0 = kt * 32 + g * 8
                            c, ky = k0 >> 4, (k0 & 15) >> 2
                            base = (c * IH + oy * 2 + ky) * PITCH + ox * 2
                            A[lr, g * 8:g * 8 + 4] = imgf[base:base + 4]
                            A[lr, g * 8 + 4:g * 8 + 8] = \
                                imgf[base + PITCH:base + PITCH + 4]
                    ch = nt * 16 + lr
                    for g in range(4):
                        k0 = kt * 32 + g * 8
                        B[g * 8:(g + 1) * 8, lr] = wb[ch, k0:k0 + 8]
                acc += A @ B
            for lr in range(16):
                for g in range(4):
                    for r in range(4):
                        row = mt * 16 + g * 4 + r
                        if row < MPX:
                            out_lds[nt * 16 + lr, row] = acc[g * 4 + r, lr]
        for ch in range(KOUT):
            out[n, ch].reshape(-1)[:] = np.maximum(out_lds[ch] + b[ch], 0.0)
    np.testing.assert_allclose(out, ref, rtol=2e-2, atol=2e-2)


def test_conv2_dgrad_v3_parity_decomposition_matches_torch():
    """Exact emulation of the stride-decomposed dgrad (conv_bwd.hip
    conv2_dgrad_v3): each input-parity class reduces over its own 2x2
    sub-kernel, K = KOUT*4 instead of the masked KOUT*16."""
    torch.manual_seed(4)
    N, C, KOUT, OH, OW, IH, IW, HP = 3, 32, 64, 9, 9, 20, 20, 10
    x = torch.randn(N, C, IH, IW, requires_grad=True)
    w = torch.randn(KOUT, C, 4, 4) * 0.1
    out = F.conv2d(x.to(torch.bfloat16).float(),
                   w.to(torch.bfloat16).float(), stride=2)
    dout = torch.randn_like(out)
    (out * dout).sum().backward()
    want = x.grad.numpy()

    dyb = dout.to(torch.bfloat16).float().numpy()
    wb = w.to(torch.bfloat16).float().numpy()
    got = np.zeros((N, C, IH, IW), dtype=np.float32)
    for n in range(N):
        for par in range(4):
            py, px = par >> 1, par & 1
            # A [MP, KP], B [KP, C] exactly as the kernel decodes
            MP, KP = HP * HP, KOUT * 4
            A = np.zeros((MP, KP), dtype=np.float32)
            B = np.zeros((KP, C), dtype=np.float32)
            for row in range(MP):
                u, v = row // HP, row % HP
                for k in range(KP):
                    ko, ab = k >> 2, k & 3
                    a, b = ab >> 1, ab & 1
                    oy, ox = u - a, v - b
                    if 0 <= oy < OH and 0 <= ox < OW:
                        A[row, k] = dyb[n, ko, oy, ox]
            for k in range(KP):
                ko, ab = k >> 2, k & 3
                a, b = ab >> 1, ab & 1
                B[k, :] = wb[ko, :, py + 2 * a, px + 2 * b]
            D = (torch.from_numpy(A).to(torch.bfloat16).float().numpy()
                 @ torch.from_numpy(B).to(torch.bfloat16).float().numpy())
            for row in range(MP):
                u, v = row // HP, row % HP
                got[n, :, 2 * u + py, 2 * v + px] = D[row, :]
    np.testing.assert_allclose(got, want, rtol=5e-2, atol=5e-2)


def test_wgrad_v3_panel_decomposition_matches_torch():
    """Exact emulation of the panel-staged wgrad (conv_bwd.hip
    convN_wgrad_v3) for the conv1 instantiation (2 pixel chunks, PXC=224,
    1-channel quarters): panel build + chunked-K GEMM accumulation."""
    torch.manual_seed(7)
    N, C, KH, KW, S, IH, IW, OH, OW, KOUT = 2, 4, 8, 8, 4, 84, 84, 20, 20, 32
    CQ, PCH, PXC = 1, 2, 224
    KQ = CQ * KH * KW
    MPX = OH * OW
    CHUNK = (MPX + PCH - 1) // PCH
    x = torch.randint(0, 256, (N, C, IH, IW), dtype=torch.uint8)
    w = (torch.randn(KOUT, C, KH, KW) * 0.1).requires_grad_()
    x_ref = (x.float() / 255.0).to(torch.bfloat16).float()
    out = F.conv2d(x_ref, w.to(torch.bfloat16).float(), stride=S)
    dout = torch.randn_like(out)
    (out * dout).sum().backward()
    want = w.grad.numpy()

    dyb = dout.to(torch.bfloat16).float().numpy()
    xb = x_ref.numpy()
    got = np.zeros((KOUT, C * KH * KW), dtype=np.float32)
    for q in range(C // CQ):
        c0 = q * CQ
        acc = np.zeros((KOUT, KQ), dtype=np.float32)
        for n in range(N):
            for ch in range(PCH):
                p0g = ch * CHUNK
                nreal = min(CHUNK, MPX - p0g)
                dy = np.zeros((KOUT, PXC), dtype=np.float32)
                dy[:, :nreal] = dyb[n].reshape(KOUT, MPX)[:, p0g:p0g + nreal]
                panel = np.zeros((KQ, PXC), dtype=np.float32)
                for kq in range(KQ):
                    c = kq // (KH * KW)
                    r = kq % (KH * KW)
                    ky, kx = r // KW, r % KW
                    for p in range(nreal):
                        pg = p0g + p
                        oy, ox = pg // OW, pg % OW
                        panel[kq, p] = xb[n, c0 + c, oy * S + ky,
                                          ox * S + kx]
                acc += (torch.from_numpy(dy).to(torch.bfloat16).float()
                        .numpy() @
                        torch.from_numpy(panel.T).to(torch.bfloat16)
                        .float().numpy())
        got[:, q * KQ:(q + 1) * KQ] = acc
    np.testing.assert_allclose(got.reshape(KOUT, C, KH, KW), want,
                               rtol=5e-2, atol=5e-1)
