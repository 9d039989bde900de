#!/usr/bin/env python
"""mnist head-kernel standalone timings (tuning aid).

Times the fused MFMA head in its variants (with/without dW2, from-ws)
plus the other mnist-step kernels, so a head change's effect is
attributable without re-running the full bench.

Run: gpurun -- 'python tools/headbench.py > gpurun_out/head.txt'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tfmesos_amd import ops


def t_us(fn, iters=3000, warmup=300):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    assert torch.cuda.is_available()
    dev = "cuda:0"
    torch.manual_seed(0)
    B, H, C, P = 100, 100, 10, 784

    x = torch.randn(B, P, device=dev, dtype=torch.bfloat16)
    w1 = torch.randn(P, H, device=dev, dtype=torch.bfloat16) * 0.05
    b1 = torch.randn(H, device=dev, dtype=torch.bfloat16)
    w2 = torch.randn(H, C, device=dev, dtype=torch.bfloat16) * 0.05
    b2 = torch.randn(C, device=dev, dtype=torch.bfloat16)
    y = torch.randint(0, C, (B,), device=dev)
    h = ops.gemm_bias_act(x, w1, b1, act="relu")
    dw2 = torch.zeros(H, C, device=dev)
    db2 = torch.zeros(C, device=dev)

    print("fwd gemm x@w1 relu     %7.2f us" %
          t_us(lambda: ops.gemm_bias_act(x, w1, b1, act="relu")))
    print("head (no dw2, 1 WG)    %7.2f us" %
          t_us(lambda: ops.mlp_head_fused(h, w2, b2, y)))
    print("head (+dw2, 2 WG)      %7.2f us" %
          t_us(lambda: ops.mlp_head_fused(h, w2, b2, y, dw2=dw2, db2=db2)))
    for ns in (2, 3, 4, 6, 9):
        os.environ["TFA_HEAD_NSLICE"] = str(ns)
        print("fused fwd+head ns=%d    %7.2f us" %
              (ns, t_us(lambda: ops.mlp_fwd_head_fused(x, w1, b1, w2, b2, y,
                                                       dw2=dw2, db2=db2))))
    os.environ.pop("TFA_HEAD_NSLICE")
    # launch floor reference: trivial elementwise kernel on a tiny tensor
    tiny = torch.zeros(256, device=dev)
    print("tiny fill (floor ref)  %7.2f us" % t_us(lambda: tiny.fill_(0.0)))

    # per-phase wall_clock64 stamps (100 MHz-class constant clock)
    from tfmesos_amd.ops import _ext
    dbg = torch.zeros(32, dtype=torch.int64, device=dev)
    _ext().head_debug(dbg)
    for _ in range(300):   # steady-state (cold first-call stamps lie)
        ops.mlp_head_fused(h, w2, b2, y, dw2=dw2, db2=db2)
    torch.cuda.synchronize()
    _ext().head_debug(torch.empty(0, device=dev))
    d = dbg.cpu().view(2, 16)
    names = ["entry", "zero", "stage", "logits", "softmax", "dh/dw2",
             "db2", "loss"]
    for wgi in range(2):
        row = d[wgi]
        base = int(row[0])
        out = []
        prev = base
        for ph in range(1, 8):
            v = int(row[ph])
            if v == 0:
                continue
            out.append("%s +%d" % (names[ph], v - prev))
            prev = v
        print("WG%d phase ticks (wall_clock64): %s  total %d"
              % (wgi, "  ".join(out), prev - base))


if __name__ == "__main__":
    main()
