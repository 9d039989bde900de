#!/usr/bin/env python
"""In-graph replicated MNIST training (fine-grained mode).

Port of the reference's ``examples/mnist/mnist.py``: ONE client process
builds the model, round-robins parameter shards across the ps tasks
(like ``tf.train.replica_device_setter``, reference ``mnist.py:43``),
creates a train step per worker, and drives every worker concurrently
with one client thread per worker sharing a locked data iterator
(reference ``mnist.py:38,63-72``). Here the remote tasks run the
tfmesos_amd executor (serve mode) instead of bare ``tf.train.Server``s:
each ps task hosts a PStore shard with fused HIP apply; each worker
executes shipped step closures on its own GPU/CPU device.
"""

import argparse
import itertools
import os
import sys
import threading

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from tfmesos_amd import cluster  # noqa: E402
from tfmesos_amd import rt  # noqa: E402
from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch  # noqa: E402


def round_robin_shards(params, n_ps):
    """variable i -> ps (i % n_ps), the replica_device_setter policy."""
    shards = [[] for _ in range(n_ps)]
    for i, (name, t) in enumerate(params):
        shards[i % n_ps].append((name, t))
    return shards


def worker_step(ctx, shard_names, batch_x, batch_y, lr):
    """Runs ON the worker agent: pull shards from every ps, one
    fwd+bwd on this device, push grads back to each ps shard."""
    import torch as T

    from tfmesos_amd.models.mlp import MnistMLP

    params = {}
    for i, names in enumerate(shard_names):
        pulled = ctx.rpc("/job:ps/task:%d" % i,
                         {"op": "ps_pull", "names": names,
                          "dtype": "bf16" if ctx.device != "cpu" else "fp32"})
        params.update(pulled)
    dev = ctx.device
    dtype = T.bfloat16 if dev != "cpu" else T.float32
    params = {n: t.to(dev, dtype) for n, t in params.items()}
    x = batch_x.to(dev, dtype)
    y = batch_y.to(dev)

    grads = {}

    def gview(name):
        g = T.zeros(params[name].shape, dtype=T.float32, device=dev)
        grads[name] = g
        return g

    model = MnistMLP()
    loss = model.fwd_bwd(lambda n: params[n], x, y, gview)
    for i, names in enumerate(shard_names):
        ctx.rpc("/job:ps/task:%d" % i,
                {"op": "ps_push",
                 "grads": {n: grads[n].cpu() for n in names}})
    return float(loss)


def main(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", "--name", default="mnist")
    parser.add_argument("-m", "--master", default=None)
    parser.add_argument("-w", "--nworker", type=int, default=2)
    parser.add_argument("-s", "--nserver", type=int, default=2)
    parser.add_argument("-Gw", "--worker_gpus", type=float, default=0)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--batch_size", type=int, default=100)
    parser.add_argument("--learning_rate", type=float, default=0.01)
    args = parser.parse_args(argv)

    model = MnistMLP()
    shards = round_robin_shards(model.init_params(), args.nserver)
    shard_names = [[n for n, _ in s] for s in shards]

    jobs_def = [
        dict(name="ps", num=args.nserver),
        dict(name="worker", num=args.nworker, gpus=args.worker_gpus),
    ]
    with cluster(jobs_def, name=args.name, master=args.master) as c:
        sess = rt.Session(c.targets["/job:worker/task:0"], targets=c.targets,
                          secret=c.secret)
        # place each parameter shard on its ps (replica_device_setter)
        for i, shard in enumerate(shards):
            sess.call(
                lambda ctx, params, lr: ctx._ex._ps_dispatch(
                    "ps_init", {"params": params, "optimizer": "sgd",
                                "hparams": {"lr": lr}}),
                [(n, t) for n, t in shard], args.learning_rate,
                device="/job:ps/task:%d" % i)

        # one client thread per worker, shared locked data iterator
        # (reference mnist.py:38,63-72)
        lock = threading.Lock()
        seeds = itertools.count()
        losses = []

        def feed():
            with lock:
                s = next(seeds)
            return synthetic_batch(args.batch_size, seed=s)

        def drive(widx):
            dev = "/job:worker/task:%d" % widx
            for _ in range(args.steps):
                x, y = feed()
                loss = sess.call(worker_step, shard_names, x, y,
                                 args.learning_rate, device=dev)
                with lock:
                    losses.append(loss)

        threads = [threading.Thread(target=drive, args=(i,))
                   for i in range(args.nworker)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()

        first = sum(losses[:args.nworker]) / args.nworker
        last = sum(losses[-args.nworker:]) / args.nworker
        print("trained %d global steps; loss %.4f -> %.4f"
              % (len(losses), first, last))
        sess.close()
        return 0 if last < first else 1


if __name__ == "__main__":
    sys.exit(main())
