"""MFMA conv encoder (forward) vs torch conv oracle.

Validated on MI355X: fragment self-test + all 3 shapes + fused u8
normalize pass against the bf16 torch reference (round 1)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = [pytest.mark.gpu]


def test_mfma_fragment_selftest():
    from scalerl_amd.ops.conv import mfma_selftest
    assert mfma_selftest(), "16x16x32 bf16 fragment map constants are wrong"


@pytest.mark.parametrize("layer,in_shape,w_shape,stride", [
    (1, (4, 84, 84), (32, 4, 8, 8), 4),
    (2, (32, 20, 20), (64, 32, 4, 4), 2),
    (3, (64, 9, 9), (64, 64, 3, 3), 1),
])
def test_conv_fwd_matches_torch(layer, in_shape, w_shape, stride):
    from scalerl_amd.ops.conv import atari_conv_fwd
    torch.manual_seed(0)
    N = 37  # deliberately not a multiple of the 32-row tile
    x = torch.randn(N, *in_shape, device="cuda")
    w = torch.randn(w_shape, device="cuda") * 0.1
    b = torch.randn(w_shape[0], device="cuda") * 0.1
    got = atari_conv_fwd(layer, x, w, b, relu=True).float()
    want = F.relu(F.conv2d(x.to(torch.bfloat16).float(),
                           w.to(torch.bfloat16).float(), b, stride=stride))
    torch.testing.assert_close(got, want, rtol=5e-2, atol=5e-2)


def test_conv1_u8_normalize_fused():
    from scalerl_amd.ops.conv import atari_conv_fwd
    torch.manual_seed(1)
    x = torch.randint(0, 256, (8, 4, 84, 84), dtype=torch.uint8,
                      device="cuda")
    w = torch.randn(32, 4, 8, 8, device="cuda") * 0.1
    got = atari_conv_fwd(1, x, w, None, relu=False).float()
    want = F.conv2d((x.float() / 255.0).to(torch.bfloat16).float(),
                    w.to(torch.bfloat16).float(), stride=4)
    torch.testing.assert_close(got, want, rtol=5e-2, atol=5e-2)


@pytest.mark.skipif(not __import__("os").environ.get("SCALERL_EXPERIMENTAL"),
                    reason="fwd v3 pending hardware validation (r3)")
@pytest.mark.parametrize("layer", [2, 3])
def test_fwd_v3_matches_v2_and_torch(layer):
    """Panel-staged forward vs the validated v2 kernel and torch."""
    from scalerl_amd.ops.conv import atari_conv_fwd, atari_conv_fwd_v3
    torch.manual_seed(11)
    shapes = {2: ((32, 20, 20), (64, 32, 4, 4), 2),
              3: ((64, 9, 9), (64, 64, 3, 3), 1)}
    in_shape, w_shape, stride = shapes[layer]
    N = 21
    x = torch.randn(N, *in_shape, device="cuda")
    w = torch.randn(w_shape, device="cuda") * 0.1
    b = torch.randn(w_shape[0], device="cuda") * 0.1
    want = F.relu(F.conv2d(x.to(torch.bfloat16).float(),
                           w.to(torch.bfloat16).float(), b, stride=stride))
    got3 = atari_conv_fwd_v3(layer, x, w, b).float()
    got2 = atari_conv_fwd(layer, x, w, b).float()
    torch.testing.assert_close(got3, want, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(got3, got2, rtol=1e-2, atol=1e-2)
