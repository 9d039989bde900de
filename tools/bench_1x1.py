#!/usr/bin/env python
"""1x1-conv routing experiment: are Inception's pointwise convs faster
through the conv kernels or through gemm.hip (they are plain GEMMs over
pixels)? Times fwd / bwd-data / bwd-weight each way, plus torch.matmul
(hipBLASLt) as the library reference.

Run: gpurun -- 'python tools/bench_1x1.py > gpurun_out/b1x1.txt 2>&1'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tfmesos_amd import ops


def t_us(fn, iters=300, warmup=50):
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
    ext = ops._ext()
    N = 32
    # (H==W spatial, C, K) for every distinct Inception 1x1 shape class
    shapes = [(35, 192, 64), (35, 256, 64), (35, 288, 48),
              (17, 768, 128), (17, 768, 192),
              (8, 1280, 320), (8, 2048, 320), (8, 2048, 448)]
    print("%-18s %8s %8s %8s | %8s %8s %8s | %8s %8s %8s"
          % ("shape", "cv_fwd", "cv_bwdd", "cv_bwdw",
             "gm_fwd", "gm_bwdd", "gm_bwdw",
             "tm_fwd", "tm_bwdd", "tm_bwdw"))
    for HW, C, K in shapes:
        P = N * HW * HW
        x4 = torch.randn(N, C, HW, HW, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w4 = torch.randn(K, 1, 1, C, device=dev, dtype=torch.bfloat16)
        dy4 = torch.randn(N, K, HW, HW, device=dev, dtype=torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        x2 = torch.randn(P, C, device=dev, dtype=torch.bfloat16)
        w2 = torch.randn(K, C, device=dev, dtype=torch.bfloat16)
        dy2 = torch.randn(P, K, device=dev, dtype=torch.bfloat16)
        dwbuf = torch.zeros(K, 1, 1, C, device=dev, dtype=torch.float32)
        eb = torch.empty(0, device=dev)

        cv_fwd = t_us(lambda: ext.conv2d_fwd(x4, w4, eb, 1, 1, 0, 0, False))
        cv_bwdd = t_us(lambda: ext.conv2d_bwd_data(dy4, w4, HW, HW,
                                                   1, 1, 0, 0))
        cv_bwdw = t_us(lambda: ext.conv2d_bwd_weight_out(dy4, x4, 1, 1,
                                                         1, 1, 0, 0, dwbuf))
        gm_fwd = t_us(lambda: ops.gemm_bias_act(x2, w2, trans_b=True))
        gm_bwdd = t_us(lambda: ops.gemm_bias_act(dy2, w2))
        gm_bwdw = t_us(lambda: ops.gemm_bias_act(dy2, x2, trans_a=True))
        tm_fwd = t_us(lambda: x2 @ w2.t())
        tm_bwdd = t_us(lambda: dy2 @ w2)
        tm_bwdw = t_us(lambda: dy2.t() @ x2)
        print("%2dx%2d c%-4d k%-4d  %8.1f %8.1f %8.1f | %8.1f %8.1f %8.1f"
              " | %8.1f %8.1f %8.1f"
              % (HW, HW, C, K, cv_fwd, cv_bwdd, cv_bwdw,
                 gm_fwd, gm_bwdd, gm_bwdw, tm_fwd, tm_bwdd, tm_bwdw))


if __name__ == "__main__":
    main()
