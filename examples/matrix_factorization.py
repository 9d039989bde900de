#!/usr/bin/env python
"""Nonnegative matrix factorization with manual model parallelism.

Port of the reference's ``examples/matrix_factorization.py``: rank-200
NMF of a 1000x1000 matrix, W placed on ps:0 and H on ps:1 (manual
variable sharding, reference ``:21-28``), Frobenius loss + soft
nonnegativity penalty (``:30-37``), plain gradient descent run through
the worker's session (``:43-49``), 100-iteration driver loop (``:67-72``).

The MI355X path: W and H live in each ps task's executor store; the
worker fetches both factors peer-to-peer, runs the three GEMMs on the
MFMA kernel (bf16) via ``tfmesos_amd.ops``, and ships the factor
updates back to their owning ps — the sparse/sharded PS traffic pattern
of the reference, minus TF.
"""

import argparse
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from tfmesos_amd import cluster  # noqa: E402
from tfmesos_amd import rt  # noqa: E402


def nmf_step(ctx, shapes, lr, lam):
    """Runs ON the worker: fetch W from ps:0 and H from ps:1, one GD
    step of the penalized Frobenius objective, push updates back."""
    import torch as T

    from tfmesos_amd import ops

    W = ctx.fetch("/job:ps/task:0", "W").to(ctx.device)
    H = ctx.fetch("/job:ps/task:1", "H").to(ctx.device)
    X = ctx._ex.store.get("X")
    if X is None:
        g = T.Generator().manual_seed(99)
        n, rank = shapes
        w0, h0 = T.rand(n, rank, generator=g), T.rand(rank, n, generator=g)
        X = (w0 @ h0 / rank).to(ctx.device)
        ctx._ex.store["X"] = X

    # summed Frobenius objective + soft nonneg penalty, as the
    # reference's loss (matrix_factorization.py:30-37)
    if ctx.device != "cpu":
        Wb, Hb = W.bfloat16(), H.bfloat16()
        E = (ops.gemm_bias_act(Wb, Hb).float() - X).bfloat16()
        dW = ops.gemm_bias_act(E, Hb, trans_b=True).float() * 2.0
        dH = ops.gemm_bias_act(Wb, E, trans_a=True).float() * 2.0
        loss = (E.float() ** 2).sum()
    else:
        E = W @ H - X
        dW = E @ H.t() * 2.0
        dH = W.t() @ E * 2.0
        loss = (E * E).sum()
    dW += 2.0 * lam * T.clamp(W, max=0.0)
    dH += 2.0 * lam * T.clamp(H, max=0.0)
    pen = (T.clamp(W, max=0.0) ** 2).sum() + (T.clamp(H, max=0.0) ** 2).sum()
    loss = float(loss + lam * pen)

    W = W - lr * dW
    H = H - lr * dH
    ctx.rpc("/job:ps/task:0", {"op": "put", "key": "W", "value": W.cpu()})
    ctx.rpc("/job:ps/task:1", {"op": "put", "key": "H", "value": H.cpu()})
    return loss


def main(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", "--name", default="nmf")
    parser.add_argument("-m", "--master", default=None)
    parser.add_argument("-Gw", "--worker_gpus", type=float, default=0)
    parser.add_argument("--size", type=int, default=1000)
    parser.add_argument("--rank", type=int, default=200)
    parser.add_argument("--steps", type=int, default=100)
    parser.add_argument("--learning_rate", type=float, default=1e-4)
    parser.add_argument("--nonneg_weight", type=float, default=1.0)
    args = parser.parse_args(argv)

    jobs_def = [
        dict(name="ps", num=2),
        dict(name="worker", num=1, gpus=args.worker_gpus),
    ]
    g = torch.Generator().manual_seed(0)
    W0 = torch.rand(args.size, args.rank, generator=g) / args.rank ** 0.5
    H0 = torch.rand(args.rank, args.size, generator=g) / args.rank ** 0.5

    with cluster(jobs_def, name=args.name, master=args.master) as c:
        sess = rt.Session(c.targets["/job:worker/task:0"], targets=c.targets,
                          secret=c.secret)
        # manual model-parallel placement: W on ps:0, H on ps:1
        # (reference matrix_factorization.py:21-28)
        rt.RemoteCall(c.targets["/job:ps/task:0"], c.secret).put("W", W0)
        rt.RemoteCall(c.targets["/job:ps/task:1"], c.secret).put("H", H0)

        losses = []
        for i in range(args.steps):
            loss = sess.call(nmf_step, (args.size, args.rank),
                             args.learning_rate, args.nonneg_weight,
                             device="/job:worker/task:0")
            losses.append(loss)
            if i % 10 == 0:
                print("iter %3d loss %.6f" % (i, loss))
        print("final loss %.6f (start %.6f)" % (losses[-1], losses[0]))
        sess.close()
        return 0 if losses[-1] < losses[0] else 1


if __name__ == "__main__":
    sys.exit(main())
