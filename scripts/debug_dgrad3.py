#!/usr/bin/env python3
"""On-GPU structural debug for the conv3 dgrad kernel failure (r2 call 1:
89.6% mismatch on layer 3 while layer 2 passes).  Prints the error
structure so the bug can be localized without guessing."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from scalerl_amd.ops.conv import atari_conv_dgrad

C, KH, KW, S, IH, IW, OH, OW, KOUT = 64, 3, 3, 1, 9, 9, 7, 7, 64


def ref_dgrad(dout, w):
    x = torch.zeros(dout.shape[0], C, IH, IW, device=dout.device,
                    requires_grad=True)
    out = F.conv2d(x, w.to(torch.bfloat16).float(), stride=S)
    (out * dout).sum().backward()
    return x.grad


def main():
    torch.manual_seed(0)
    dev = "cuda"

    # 1) full random compare, N=1: where is it wrong?
    N = 1
    dout = torch.randn(N, KOUT, OH, OW, device=dev)
    w = torch.randn(KOUT, C, KH, KW, device=dev) * 0.1
    want = ref_dgrad(dout, w)
    got = atari_conv_dgrad(3, dout, w).float()
    bad = ((got - want).abs() > 0.05)
    print(f"N=1 mismatch: {bad.sum().item()}/{bad.numel()}")
    # mismatch by input pixel (iy, ix) aggregated over c
    by_pix = bad[0].any(dim=0).int()
    print("bad-by-(iy,ix):\n", by_pix.cpu().numpy())
    # mismatch by channel
    by_c = bad[0].flatten(1).any(dim=1).int()
    print("bad-by-c:", by_c.cpu().numpy())

    # 2) delta test: single dout element x single weight element
    for (ko, oy, ox, c, ky, kx) in [(0, 0, 0, 0, 0, 0), (5, 3, 2, 17, 1, 2),
                                    (63, 6, 6, 63, 2, 2), (12, 2, 5, 40, 0, 1)]:
        dout = torch.zeros(1, KOUT, OH, OW, device=dev)
        dout[0, ko, oy, ox] = 1.0
        w = torch.zeros(KOUT, C, KH, KW, device=dev)
        w[ko, c, ky, kx] = 1.0
        got = atari_conv_dgrad(3, dout, w).float()
        nz = got[0].abs().nonzero()
        expect = (c, oy * S + ky, ox * S + kx)
        print(f"delta ko={ko} oy={oy} ox={ox} c={c} ky={ky} kx={kx}: "
              f"expect din[{expect}]=1, got nonzeros "
              f"{nz.cpu().numpy().tolist()[:8]} "
              f"vals {got[0][got[0].abs() > 0].cpu().numpy().tolist()[:8]}")

    # 3) batch-offset test: is the n decode right?
    N = 3
    dout = torch.zeros(N, KOUT, OH, OW, device=dev)
    dout[2, 0, 0, 0] = 1.0
    w = torch.zeros(KOUT, C, KH, KW, device=dev)
    w[0, 0, 0, 0] = 1.0
    got = atari_conv_dgrad(3, dout, w).float()
    for n in range(N):
        nz = got[n].abs().nonzero()
        print(f"n={n}: nonzeros {nz.cpu().numpy().tolist()[:4]}")


if __name__ == "__main__":
    main()
