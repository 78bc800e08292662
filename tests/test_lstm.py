"""MaskedLSTM vs torch.nn.LSTM with manual per-step masking (the reference
semantics, atari_model.py:109-120), forward + gradients, CPU and GPU."""

import pytest
import torch
import torch.nn as nn

from scalerl_amd.ops import MaskedLSTM


def _torch_masked_lstm(lstm: nn.LSTM, x, notdone, h0, c0):
    """Per-step loop with `state *= notdone` masking (the oracle)."""
    outs = []
    h, c = h0, c0
    for t in range(x.shape[0]):
        nd = notdone[t].view(1, -1, 1)
        h = h * nd
        c = c * nd
        out, (h, c) = lstm(x[t:t + 1], (h, c))
        outs.append(out)
    return torch.cat(outs, dim=0), (h, c)


def _sync_weights(ml: MaskedLSTM, tl: nn.LSTM):
    with torch.no_grad():
        for k in range(ml.num_layers):
            getattr(ml, f"weight_ih_l{k}").copy_(getattr(tl, f"weight_ih_l{k}"))
            getattr(ml, f"weight_hh_l{k}").copy_(getattr(tl, f"weight_hh_l{k}"))
            getattr(ml, f"bias_ih_l{k}").copy_(getattr(tl, f"bias_ih_l{k}"))
            getattr(ml, f"bias_hh_l{k}").copy_(getattr(tl, f"bias_hh_l{k}"))


@pytest.mark.parametrize("layers", [1, 2])
def test_masked_lstm_matches_torch_cpu(layers):
    torch.manual_seed(0)
    T, B, I, H = 7, 4, 10, 12
    tl = nn.LSTM(I, H, num_layers=layers)
    ml = MaskedLSTM(I, H, num_layers=layers)
    _sync_weights(ml, tl)
    x = torch.randn(T, B, I, requires_grad=True)
    x2 = x.detach().clone().requires_grad_()
    notdone = (torch.rand(T, B) > 0.2).float()
    h0 = torch.zeros(layers, B, H)
    c0 = torch.zeros(layers, B, H)

    out_ref, (h_ref, c_ref) = _torch_masked_lstm(tl, x, notdone, h0, c0)
    out, (h, c) = ml(x2, notdone, (h0, c0))
    torch.testing.assert_close(out, out_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(h, h_ref, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(c, c_ref, rtol=1e-4, atol=1e-5)

    out_ref.sum().backward()
    out.sum().backward()
    torch.testing.assert_close(x2.grad, x.grad, rtol=1e-4, atol=1e-5)
    for k in range(layers):
        torch.testing.assert_close(
            getattr(ml, f"weight_ih_l{k}").grad,
            getattr(tl, f"weight_ih_l{k}").grad, rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(
            getattr(ml, f"weight_hh_l{k}").grad,
            getattr(tl, f"weight_hh_l{k}").grad, rtol=1e-4, atol=1e-5)


@pytest.mark.gpu
def test_masked_lstm_gpu_matches_cpu():
    torch.manual_seed(0)
    T, B, I, H, L = 20, 8, 32, 64, 2
    ml = MaskedLSTM(I, H, num_layers=L)
    x = torch.randn(T, B, I)
    notdone = (torch.rand(T, B) > 0.1).float()
    state = ml.initial_state(B)

    xc = x.clone().requires_grad_()
    out_c, _ = ml(xc, notdone, state)
    out_c.sum().backward()

    mg = MaskedLSTM(I, H, num_layers=L).cuda()
    mg.load_state_dict(ml.state_dict())
    xg = x.cuda().requires_grad_()
    sg = tuple(s.cuda() for s in state)
    out_g, _ = mg(xg, notdone.cuda(), sg)
    out_g.sum().backward()

    torch.testing.assert_close(out_g.cpu(), out_c.detach(), rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(xg.grad.cpu(), xc.grad, rtol=1e-3, atol=1e-4)
    torch.testing.assert_close(mg.weight_hh_l0.grad.cpu(),
                               ml.weight_hh_l0.grad, rtol=1e-3, atol=1e-3)


@pytest.mark.gpu
def test_masked_lstm_seq_path_matches_python_path():
    """SCALERL_LSTM_SEQ C++-driven loop vs the per-step Python path."""
    import scalerl_amd.ops.lstm as lstm_mod
    torch.manual_seed(0)
    T, B, I, H, L = 20, 8, 32, 64, 2
    ml = MaskedLSTM(I, H, num_layers=L).cuda()
    x = torch.randn(T, B, I, device="cuda")
    notdone = (torch.rand(T, B, device="cuda") > 0.1).float()
    state = tuple(s.cuda() for s in ml.initial_state(B))

    def run():
        ml.zero_grad()
        xg = x.clone().requires_grad_()
        out, (h, c) = ml(xg, notdone, state)
        (out.square().mean() + h.sum() * 0.1).backward()
        return (out.detach().cpu(), h.detach().cpu(),
                xg.grad.cpu(), ml.weight_hh_l0.grad.cpu().clone(),
                ml.weight_ih_l1.grad.cpu().clone())

    lstm_mod._USE_SEQ = False
    ref = run()
    lstm_mod._USE_SEQ = True
    try:
        got = run()
    finally:
        lstm_mod._USE_SEQ = None
    for g, r in zip(got, ref):
        torch.testing.assert_close(g, r, rtol=1e-4, atol=1e-5)
