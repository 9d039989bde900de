#!/usr/bin/env python
"""Per-op microbenchmark on the MI355X box (tuning aid, not the flagship
bench). Times each mnist-step kernel standalone (1000 launches, stream-
synced once) and compares the GEMMs against torch.matmul (hipBLASLt).

Run: gpurun -- 'python tools/microbench.py > gpurun_out/micro.txt'
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tfmesos_amd import ops


def t_us(fn, iters=2000, warmup=200):
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
    hid_w = torch.randn(P, H, device=dev, dtype=torch.bfloat16)
    hid_b = torch.randn(H, device=dev, dtype=torch.bfloat16)
    sm_w = torch.randn(H, C, device=dev, dtype=torch.bfloat16)
    sm_b = torch.randn(C, device=dev, dtype=torch.bfloat16)
    h = ops.gemm_bias_act(x, hid_w, hid_b, act="relu")
    logits = ops.gemm_bias_act(h, sm_w, sm_b)
    y = torch.randint(0, C, (B,), device=dev)
    _, dlogits = ops.softmax_xent_fused(logits, y)
    gw1 = torch.zeros(P, H, device=dev)
    gb1 = torch.zeros(H, device=dev)
    gw2 = torch.zeros(H, C, device=dev)
    gb2 = torch.zeros(C, device=dev)
    flat = torch.randn(P * H + H + H * C + C, device=dev)
    flat_g = torch.randn_like(flat)
    flat_bf = flat.to(torch.bfloat16)

    rows = [
        ("fwd1 gemm+bias+relu [100x784]@[784x100] (split-K)",
         lambda: ops.gemm_bias_act(x, hid_w, hid_b, act="relu")),
        ("fwd2 gemm+bias [100x100]@[100x10]",
         lambda: ops.gemm_bias_act(h, sm_w, sm_b)),
        ("softmax_xent fused fwd+bwd [100,10]",
         lambda: ops.softmax_xent_fused(logits, y)),
        ("dW2 gemm tn + colsum [100x100]^T@[100x10]",
         lambda: ops.gemm_bias_act(h, dlogits, trans_a=True, out=gw2,
                                   colsum_out=gb2)),
        ("dh gemm nt + relu_bwd [100x10]@[100x10]^T",
         lambda: ops.gemm_bias_act(dlogits, sm_w, trans_b=True,
                                   act="relu_bwd", aux=h)),
        ("dW1 gemm tn + colsum [100x784]^T@[100x100]",
         lambda: ops.gemm_bias_act(x, h, trans_a=True, out=gw1,
                                   colsum_out=gb1)),
        ("mlp_head_fused (fwd2+softmax+dh in one kernel)",
         lambda: ops.mlp_head_fused(h, sm_w, sm_b, y)),
        ("sgd fused apply (79510 params)",
         lambda: ops.fused_sgd(flat, flat_g, lr=0.01, bf16_out=flat_bf)),
        ("torch.matmul ref fwd1 (hipBLASLt)",
         lambda: torch.matmul(x, hid_w)),
        ("torch.matmul ref dW1 (hipBLASLt)",
         lambda: torch.matmul(x.t(), h)),
        ("null kernel (fill 4B) = launch+exec floor",
         lambda: gb2.fill_(0.0)),
    ]
    total = 0.0
    for name, fn in rows:
        us = t_us(fn)
        print("%-52s %8.2f us" % (name, us))
        if not name.startswith(("torch", "null")):
            total += us
    print("%-52s %8.2f us" % ("SUM (step estimate)", total))


if __name__ == "__main__":
    main()
