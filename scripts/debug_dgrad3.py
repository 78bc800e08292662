#!/usr/bin/env python3
"""On-GPU structural debug for the dgrad N=13 failures (r2c1/r2c3/r2c4):
CPU-autograd references, per-image mismatch maps, and run-to-run variance
(random scatter that changes between runs = race; fixed structure =
indexing)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from scalerl_amd.ops.conv import atari_conv_dgrad

CFG = {2: (32, 4, 4, 2, 20, 20, 9, 9, 64),
       3: (64, 3, 3, 1, 9, 9, 7, 7, 64)}


def main():
    for layer in (3, 2):
        C, KH, KW, S, IH, IW, OH, OW, KOUT = CFG[layer]
        torch.manual_seed(1)
        N = 13
        x = torch.randn(N, C, IH, IW, requires_grad=True)
        w = torch.randn(KOUT, C, KH, KW) * 0.1
        out = F.conv2d(x.to(torch.bfloat16).float(),
                       w.to(torch.bfloat16).float(), stride=S)
        dout = torch.randn_like(out)
        (out * dout).sum().backward()
        want = x.grad

        dout_g, w_g = dout.cuda(), w.cuda()
        runs = []
        for r in range(3):
            got = atari_conv_dgrad(layer, dout_g, w_g).float().cpu()
            runs.append(got)
            bad = (got - want).abs() > 0.05
            per_img = bad.flatten(1).sum(1)
            print(f"layer{layer} run{r}: total bad {int(bad.sum())}/"
                  f"{bad.numel()}  per-image {per_img.tolist()}")
        d01 = (runs[0] - runs[1]).abs().max()
        d12 = (runs[1] - runs[2]).abs().max()
        print(f"layer{layer} run-to-run max diff: {d01:.6f} {d12:.6f}")
        # mismatch map of image 0 by (iy, ix): structured or scattered?
        bad0 = ((runs[0] - want).abs() > 0.05)[0].any(dim=0).int()
        print(f"layer{layer} img0 bad-by-(iy,ix):\n{bad0.numpy()}")
        # sync between launches changes anything? (async temp lifetime)
        got_sync = None
        for r in range(2):
            g = atari_conv_dgrad(layer, dout_g.clone(), w_g.clone())
            torch.cuda.synchronize()
            got_sync = g.float().cpu()
        bad = (got_sync - want).abs() > 0.05
        print(f"layer{layer} with-sync+cloned-args: bad {int(bad.sum())}")


def probe2():
    """Decisive probes: (a) identical images — per-block outputs must be
    identical; (b) batch sweep — where does the wrong/right boundary sit?"""
    layer = 3
    C, KH, KW, S, IH, IW, OH, OW, KOUT = CFG[layer]
    torch.manual_seed(2)
    w = torch.randn(KOUT, C, KH, KW).cuda() * 0.1
    d1 = torch.randn(1, KOUT, OH, OW).cuda()
    # (a) identical images
    N = 12
    dout = d1.repeat(N, 1, 1, 1)
    got = atari_conv_dgrad(layer, dout, w).float()
    ref0 = atari_conv_dgrad(layer, d1, w).float()[0]
    per_img = [(got[i] - ref0).abs().max().item() for i in range(N)]
    print(f"identical-images: max|got[i]-got_single| per image {per_img}")
    # (b) batch sweep with distinct images
    torch.manual_seed(3)
    douts = torch.randn(13, KOUT, OH, OW).cuda()
    singles = torch.stack([
        atari_conv_dgrad(layer, douts[i:i + 1], w).float()[0]
        for i in range(13)])
    for B in (2, 4, 7, 13):
        gotB = atari_conv_dgrad(layer, douts[:B], w).float()
        diffs = [round((gotB[i] - singles[i]).abs().max().item(), 3)
                 for i in range(B)]
        print(f"batch={B}: max|batched-single| per image {diffs}")


if __name__ == "__main__":
    probe2()
