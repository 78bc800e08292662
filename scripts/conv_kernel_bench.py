#!/usr/bin/env python3
"""Per-op timing: native MFMA conv kernels vs torch/MIOpen at the bench
shape (N = T+1 x B = 81*256 = 20736 images).  hipEvent-timed."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "1")

import torch
import torch.nn.functional as F

from scalerl_amd.ops.conv import (atari_conv2_dgrad_v3, atari_conv_dgrad,
                                  atari_conv_fwd, atari_conv_fwd_v3,
                                  atari_conv_wgrad, atari_conv_wgrad_v3)

N = int(os.environ.get("CONV_BENCH_N", "20736"))
dev = "cuda"


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    e.synchronize()
    return s.elapsed_time(e) / iters


def main():
    torch.manual_seed(0)
    shapes = {
        1: ((4, 84, 84), (32, 4, 8, 8), 4, (32, 20, 20)),
        2: ((32, 20, 20), (64, 32, 4, 4), 2, (64, 9, 9)),
        3: ((64, 9, 9), (64, 64, 3, 3), 1, (64, 7, 7)),
    }
    for layer, (ins, ws, stride, outs) in shapes.items():
        if layer == 1:
            x = torch.randint(0, 256, (N, *ins), dtype=torch.uint8, device=dev)
            x_t = (x.float() / 255.0).to(torch.bfloat16)
        else:
            x = torch.randn(N, *ins, device=dev, dtype=torch.bfloat16)
            x_t = x
        w = (torch.randn(ws, device=dev) * 0.1)
        wb = w.to(torch.bfloat16)
        b = torch.randn(ws[0], device=dev)
        dout = torch.randn(N, *outs, device=dev, dtype=torch.bfloat16)
        doutf = dout.float()

        t_nat = timeit(lambda: atari_conv_fwd(layer, x, w, b))
        if layer in (2, 3):
            t_f3 = timeit(lambda: atari_conv_fwd_v3(layer, x, w, b))
            print(f"conv{layer} fwd v3: {t_f3:7.2f} ms (panel-staged)")
        t_mio = timeit(lambda: F.relu(F.conv2d(x_t, wb, b.to(torch.bfloat16),
                                               stride=stride)))
        print(f"conv{layer} fwd : native {t_nat:7.2f} ms  miopen {t_mio:7.2f} ms")

        t_nat = timeit(lambda: atari_conv_wgrad(layer, x, dout))
        t_v3 = timeit(lambda: atari_conv_wgrad_v3(layer, x, dout))
        print(f"conv{layer} wgrd v3: {t_v3:7.2f} ms (panel-staged)")
        wg = wb.clone().requires_grad_()
        def mio_wgrad():
            out = F.conv2d(x_t, wg, stride=stride)
            out.backward(dout)
            wg.grad = None
        t_mio = timeit(mio_wgrad)
        print(f"conv{layer} wgrd: native {t_nat:7.2f} ms  miopen(f+w) {t_mio:7.2f} ms")

        if layer in (2, 3):
            t_nat = timeit(lambda: atari_conv_dgrad(layer, dout, w))
            if layer == 2:
                t_v3 = timeit(lambda: atari_conv2_dgrad_v3(dout, w))
                print(f"conv2 dgrd v3: {t_v3:7.2f} ms (stride-decomposed)")
            xg = x_t.clone().requires_grad_()
            def mio_dgrad():
                out = F.conv2d(xg, wb.detach(), stride=stride)
                out.backward(dout)
                xg.grad = None
            t_mio = timeit(mio_dgrad)
            print(f"conv{layer} dgrd: native {t_nat:7.2f} ms  miopen(f+d+w) {t_mio:7.2f} ms")


if __name__ == "__main__":
    main()
