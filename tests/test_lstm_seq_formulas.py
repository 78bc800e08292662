"""CPU emulation of the C++ LSTM sequence loop (csrc/lstm_seq.hip) vs the
tested per-step reference — validates the orchestration (mask ordering,
carry buffer reuse, GEMM operand mapping) without a GPU.  The per-element
gate math is shared with the already-tested pointwise kernels."""

import torch

from scalerl_amd.ops.lstm import MaskedLSTM, _MaskedLSTMFn


def emu_seq_fwd(xg, w_hh, notdone, h, c):
    """Mirror of masked_lstm_seq_fwd's loop, including buffer roles."""
    T, B, H4 = xg.shape
    H = H4 // 4
    hs = torch.empty(T, B, H)
    hs_in = torch.empty(T, B, H)
    cs_in = torch.empty(T, B, H)
    cs_out = torch.empty(T, B, H)
    gates_all = xg.clone()
    h, c = h.clone(), c.clone()
    for t in range(T):
        nd = notdone[t].unsqueeze(1)
        h = h * nd
        c = c * nd
        hs_in[t], cs_in[t] = h, c
        gemm = h @ w_hh.t()                       # sgemm_nt
        pre = gates_all[t] + gemm                 # pointwise adds xg + gemm
        i = torch.sigmoid(pre[:, 0 * H:1 * H])
        f = torch.sigmoid(pre[:, 1 * H:2 * H])
        g = torch.tanh(pre[:, 2 * H:3 * H])
        o = torch.sigmoid(pre[:, 3 * H:4 * H])
        c = f * c + i * g
        h = o * torch.tanh(c)
        gates_all[t] = torch.cat([i, f, g, o], dim=1)
        hs[t], cs_out[t] = h, c
    return hs, h, c, gates_all, hs_in, cs_in, cs_out


def emu_seq_bwd(gates_all, cs_in, cs_out, d_hs, notdone, w_hh, d_hT, d_cT):
    """Mirror of masked_lstm_seq_bwd's loop with its carry/buffer dance."""
    T, B, H4 = gates_all.shape
    H = H4 // 4
    dgates_all = torch.empty_like(gates_all)
    dh_carry = d_hT.clone()
    dc_carry = d_cT.clone()
    for t in range(T - 1, -1, -1):
        i = gates_all[t][:, 0 * H:1 * H]
        f = gates_all[t][:, 1 * H:2 * H]
        g = gates_all[t][:, 2 * H:3 * H]
        o = gates_all[t][:, 3 * H:4 * H]
        tc = torch.tanh(cs_out[t])
        dh = d_hs[t] + dh_carry
        dc = dc_carry + dh * o * (1 - tc * tc)
        dgates = torch.cat([dc * g * i * (1 - i),
                            dc * cs_in[t] * f * (1 - f),
                            dc * i * (1 - g * g),
                            dh * tc * o * (1 - o)], dim=1)
        dgates_all[t] = dgates
        dc_prev = dc * f
        nd = notdone[t].unsqueeze(1)
        dh_carry = (dgates @ w_hh) * nd           # sgemm_nn then mask
        dc_carry = dc_prev * nd                   # mask then memcpy into carry
    return dgates_all, dh_carry, dc_carry


def test_seq_loop_matches_per_step_function():
    torch.manual_seed(0)
    T, B, I, H = 9, 4, 7, 11
    ml = MaskedLSTM(I, H, num_layers=1)
    x = torch.randn(T, B, I, requires_grad=True)
    notdone = (torch.rand(T, B) > 0.25).float()
    h0 = torch.randn(B, H, requires_grad=True)
    c0 = torch.randn(B, H, requires_grad=True)

    # reference: the tested per-step autograd Function
    hs_ref, hT_ref, cT_ref = _MaskedLSTMFn.apply(
        x, notdone.unsqueeze(-1), h0, c0, ml.weight_ih_l0, ml.weight_hh_l0,
        ml.bias_ih_l0, ml.bias_hh_l0)
    d_hs = torch.randn_like(hs_ref)
    d_hT = torch.randn_like(hT_ref)
    d_cT = torch.randn_like(cT_ref)
    grads_ref = torch.autograd.grad(
        (hs_ref, hT_ref, cT_ref), (x, h0, c0, ml.weight_hh_l0),
        (d_hs, d_hT, d_cT), allow_unused=False, retain_graph=False)

    # emulation of the C++ loops
    with torch.no_grad():
        xg = (x.reshape(T * B, I) @ ml.weight_ih_l0.t()
              + ml.bias_ih_l0 + ml.bias_hh_l0).view(T, B, 4 * H)
        hs, hT, cT, gates_all, hs_in, cs_in, cs_out = emu_seq_fwd(
            xg, ml.weight_hh_l0, notdone, h0, c0)
        torch.testing.assert_close(hs, hs_ref, rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(hT, hT_ref, rtol=1e-5, atol=1e-6)
        torch.testing.assert_close(cT, cT_ref, rtol=1e-5, atol=1e-6)

        dgates_all, dh0, dc0 = emu_seq_bwd(
            gates_all, cs_in, cs_out, d_hs, notdone, ml.weight_hh_l0,
            d_hT, d_cT)
        # dx / dW_hh follow the same big-GEMM epilogue as the Python path
        dx = (dgates_all.reshape(T * B, 4 * H) @ ml.weight_ih_l0).view(
            T, B, I)
        dw_hh = dgates_all.reshape(T * B, 4 * H).t() @ hs_in.reshape(
            T * B, H)
    torch.testing.assert_close(dx, grads_ref[0], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dh0, grads_ref[1], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dc0, grads_ref[2], rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dw_hh, grads_ref[3], rtol=1e-4, atol=1e-5)
