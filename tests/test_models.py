"""Model shapes, dict I/O contract, init, state-dict interop."""

import pytest
import torch
import torch.nn as nn

from scalerl_amd.models import (A3CAtariNet, ActorCriticNet, AtariNet,
                                AtariQNet, QNet, ResNetLSTMPolicy)


def _impala_inputs(T, B, A, C=4, H=84, W=84):
    return {
        "obs": torch.randint(0, 256, (T, B, C, H, W), dtype=torch.uint8),
        "reward": torch.randn(T, B),
        "done": torch.rand(T, B) < 0.1,
        "last_action": torch.randint(0, A, (T, B)),
    }


@pytest.mark.parametrize("use_lstm", [True, False])
def test_atarinet_contract(use_lstm):
    T, B, A = 5, 3, 7
    net = AtariNet((4, 84, 84), A, use_lstm=use_lstm)
    state = net.initial_state(B)
    out, new_state = net(_impala_inputs(T, B, A), state)
    assert out["policy_logits"].shape == (T, B, A)
    assert out["baseline"].shape == (T, B)
    assert out["action"].shape == (T, B)
    assert out["action"].max() < A
    if use_lstm:
        assert new_state[0].shape == (2, B, 512 + A + 1)


def test_atarinet_lstm_reset_on_done():
    """A done at step t must cut recurrent state flow: outputs after the
    reset are independent of the pre-reset history."""
    T, B, A = 6, 1, 4
    net = AtariNet((4, 84, 84), A, use_lstm=True)
    net.eval()
    inputs = _impala_inputs(T, B, A)
    inputs["done"] = torch.zeros(T, B, dtype=torch.bool)
    inputs["done"][3] = True
    state = net.initial_state(B)
    with torch.no_grad():
        out_a, _ = net(inputs, state)
        # change pre-reset history only
        inputs2 = {k: v.clone() for k, v in inputs.items()}
        inputs2["obs"][:3] = torch.randint(0, 256, (3, B, 4, 84, 84),
                                           dtype=torch.uint8)
        out_b, _ = net(inputs2, net.initial_state(B))
    # rows >= 3 depend only on rows >= 3 (done masks h,c AND the reward /
    # last_action inputs at row 3 are identical)
    torch.testing.assert_close(out_a["baseline"][4:], out_b["baseline"][4:],
                               rtol=1e-4, atol=1e-5)


def test_atari_qnet_shapes():
    net = AtariQNet((4, 84, 84), 6, dueling=True)
    q = net(torch.randint(0, 256, (3, 4, 84, 84), dtype=torch.uint8))
    assert q.shape == (3, 6)


def test_a3c_atari_net():
    net = A3CAtariNet(in_channels=1, num_actions=6)
    state = net.initial_state(2)
    logits, value, state = net(torch.randn(2, 1, 42, 42), state)
    assert logits.shape == (2, 6) and value.shape == (2,)


def test_resnet_lstm_policy():
    net = ResNetLSTMPolicy(num_actions=4)
    T, B = 2, 2
    obs = torch.randint(0, 256, (T, B, 4, 128, 128), dtype=torch.uint8)
    state = net.initial_state(B)
    logits, value, state = net(obs, torch.ones(T, B), state)
    assert logits.shape == (T, B, 4) and value.shape == (T, B)


def test_qnet_and_ac_variants():
    q = QNet(4, 2, dueling=True)
    assert q(torch.randn(5, 4)).shape == (5, 2)
    ac = ActorCriticNet(4, 2)
    logits, v = ac(torch.randn(5, 4))
    assert logits.shape == (5, 2) and v.shape == (5,)
    a, lg, vv = ac.get_action(torch.randn(5, 4), greedy=True)
    assert a.shape == (5,)


def test_masked_lstm_state_dict_matches_nn_lstm_keys():
    """Checkpoint interop: MaskedLSTM parameter names follow nn.LSTM."""
    from scalerl_amd.ops import MaskedLSTM
    ours = set(MaskedLSTM(8, 16, num_layers=2).state_dict())
    torchs = set(nn.LSTM(8, 16, num_layers=2).state_dict())
    assert ours == torchs
