#!/usr/bin/env python
"""Per-shape conv kernel throughput (TFLOP/s) on representative
Inception-v3 layers. Tuning aid: run via gpurun."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tfmesos_amd import ops

SHAPES = [
    # (name, N, C, H, W, K, R, S, stride, pad)
    ("stem3x3 149^2 c32k32", 32, 32, 147, 147, 32, 3, 3, 1, 0),
    ("A 5x5 35^2 c48k64", 32, 48, 35, 35, 64, 5, 5, 1, 2),
    ("A 3x3 35^2 c96k96", 32, 96, 35, 35, 96, 3, 3, 1, 1),
    ("C 1x7 17^2 c160", 32, 160, 17, 17, 160, 1, 7, 1, 0),
    ("E 3x3 8^2 c448k384", 32, 448, 8, 8, 384, 3, 3, 1, 1),
    ("pw 1x1 17^2 c768k192", 32, 768, 17, 17, 192, 1, 1, 1, 0),
    ("pw 1x1 8^2 c2048k320", 32, 2048, 8, 8, 320, 1, 1, 1, 0),
]


def t_ms(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    dev = "cuda:0"
    only = os.environ.get("CONV_ONLY")
    global SHAPES
    if only is not None:
        SHAPES = [SHAPES[int(only)]]
    print("%-24s %9s %9s %9s | TFLOP/s fwd bwdd bwdw" %
          ("shape", "fwd_us", "bwdd_us", "bwdw_us"))
    for name, N, C, H, W, K, R, S, st, pd in SHAPES:
        pads = (pd if R > 1 else 0, pd if S > 1 else 0)
        x = torch.randn(N, C, H, W, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w = (torch.randn(K, R, S, C, device=dev, dtype=torch.bfloat16) * 0.1)
        Ho = (H + 2 * pads[0] - R) // st + 1
        Wo = (W + 2 * pads[1] - S) // st + 1
        y = ops.conv2d(x, w, stride=st, padding=pads, weight_format="krsc")
        dy = torch.randn_like(y).contiguous(memory_format=torch.channels_last)
        ext = ops._ext()
        flops = 2.0 * N * Ho * Wo * K * C * R * S

        f = t_ms(lambda: ext.conv2d_fwd(
            x, w, torch.empty(0, device=dev), st, st, pads[0], pads[1],
            False)) * 1e3
        d = t_ms(lambda: ext.conv2d_bwd_data(
            dy, w, H, W, st, st, pads[0], pads[1])) * 1e3
        g = t_ms(lambda: ext.conv2d_bwd_weight(
            dy, x, R, S, st, st, pads[0], pads[1])) * 1e3
        print("%-24s %9.1f %9.1f %9.1f | %7.0f %7.0f %7.0f" %
              (name, f, d, g, flops / f / 1e6, flops / d / 1e6,
               flops / g / 1e6))


if __name__ == "__main__":
    main()
