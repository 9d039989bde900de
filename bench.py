#!/usr/bin/env python
"""Flagship benchmark: mnist_replica 1-ps/N-worker sync SGD, bf16.

Measures the BASELINE.json metric — global steps/sec of the
mnist_replica-equivalent workload (784-100-10 MLP, batch 100/worker,
sync gradient aggregation at the PS, SGD lr=0.01) on N MI355X GPUs:
world==1 runs ps+worker colocated on one GPU; world>1 puts the PS on
rank 0's GPU and workers on ranks 1..N-1, with grad push = RCCL reduce
and param pull = RCCL broadcast over xGMI.

Run directly (N=1) or under torch.distributed.run with one rank per GPU:
  python bench.py --gpus 1 --steps 1000 --warmup 100
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 1000 --warmup 100

Rank 0 prints ONE JSON line with the whole-job metric.

Extra workloads (same PS machinery): --workload nmf (sparse-embedding
2-ps-shard NMF config), --workload inception (conv net, bf16).
"""

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist


def log(msg):
    sys.stderr.write(msg + "\n")
    sys.stderr.flush()


def run_mnist(args, device, rank, world):
    from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
    from tfmesos_amd.ps.replica import SyncReplicaTrainer

    model = MnistMLP(hidden_units=args.hidden)
    # PS shards colocate with workers: all N GPUs compute replicas and
    # the sharded reduce lands on ranks 0..n_ps-1 ("1-ps/N-worker" on N
    # devices, the BASELINE.json config)
    grad_dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    trainer = SyncReplicaTrainer(
        model.init_params(), optimizer=args.optimizer,
        hparams={"lr": args.lr}, device=device, n_ps=args.num_ps,
        colocate_ps=True, grad_dtype=grad_dtype)
    roles = trainer.roles

    act_dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    x, y = synthetic_batch(args.batch, device=device, dtype=act_dtype,
                           seed=1000 + rank)

    pview = (lambda n: trainer.store.view(n, bf16=True)) \
        if act_dtype == torch.bfloat16 else (lambda n: trainer.store.view(n))

    if world == 1 and device.type == "cuda" and not args.no_fuse \
            and model.supports_fused_apply(trainer, x):
        # single-GPU fast path: the SGD apply rides the producing
        # kernels' epilogues (4 launches/step instead of 5) — same
        # update math, measured not skipped
        def one_step():
            model.fwd_bwd_apply(trainer, x, y, lr=args.lr)
    else:
        def one_step():
            if roles.is_worker:
                model.fwd_bwd(pview, x, y, trainer.grad_view)
            trainer.step()

    if world == 1 and device.type == "cuda" and not args.no_graph:
        # single-GPU hot path: capture the whole 8-kernel step as ONE
        # hipGraph replay (no collectives inside)
        from tfmesos_amd.utils.graphstep import GraphedStep
        one_step = GraphedStep(one_step)

    return one_step, {
        "model": "mnist_replica_mlp_784x%dx10" % args.hidden,
        "global_batch": args.batch * roles.n_workers,
        "parallelism": roles.describe(),
        "optimizer": args.optimizer,
        "lr": args.lr,
        "sync": True,
    }


def run_nmf(args, device, rank, world):
    """matrix_factorization config: rank-200 NMF. world==1 runs the
    dense local workload; world>1 runs the sparse-embedding PS path
    (W rows on ps:0, H rows on ps:1, minibatch pull/push)."""
    if world > 1:
        from tfmesos_amd.models.nmf import SparseNMF
        wl = SparseNMF(rank, world, device=device, n=args.nmf_n,
                       factor_rank=args.nmf_rank, batch=args.nmf_batch,
                       lr=args.lr)
        n_ps = 2 if world > 2 else 1
        par = "ps%d+w%d(sparse)" % (n_ps, world - n_ps)
        gbatch = args.nmf_batch * (world - n_ps)
    else:
        from tfmesos_amd.models.nmf import NMFWorkload
        wl = NMFWorkload(n=args.nmf_n, rank=args.nmf_rank, device=device,
                         lr=args.lr, seed=1234 + rank)
        par = "local"
        gbatch = args.nmf_n
    def one_step():
        wl.one_step()
    one_step.finalize = getattr(wl, "finalize", lambda: None)
    return one_step, {
        "model": "nmf_%dx%d_rank%d" % (args.nmf_n, args.nmf_n, args.nmf_rank),
        "global_batch": gbatch,
        "parallelism": par,
        "optimizer": "sgd",
        "lr": args.lr,
        "sync": False if world > 1 else True,
    }


def run_inception(args, device, rank, world):
    """Inception-v3 distributed train config (BASELINE.json): autograd
    model on the implicit-GEMM MFMA conv kernels, PS replica data plane
    (zero-copy flat store wiring)."""
    import torch.nn.functional  # noqa: F401

    from tfmesos_amd import ops
    from tfmesos_amd.models.inception import InceptionV3, synthetic_images
    from tfmesos_amd.ps.module_trainer import ModuleReplicaTrainer

    model = InceptionV3(num_classes=args.classes)
    trainer = ModuleReplicaTrainer(model, optimizer=args.optimizer,
                                   hparams={"lr": args.lr}, device=device,
                                   n_ps=args.num_ps, colocate_ps=True)
    x, y = synthetic_images(args.inc_batch, size=args.inc_size,
                            classes=args.classes, device=device,
                            dtype=torch.bfloat16, seed=1000 + rank)

    def one_step():
        if trainer.roles.is_worker:
            trainer.zero_grad()
            loss = ops.softmax_xent_loss(model(x).contiguous(), y)
            loss.backward()
        trainer.step()

    if world == 1 and device.type == "cuda" and not args.no_graph:
        # capture the whole fwd+bwd+apply step (~600 launches) as one
        # hipGraph replay; GraphedStep self-tunes and keeps eager if
        # replay doesn't clearly win, and we fall back if any captured
        # op refuses graph capture
        try:
            from tfmesos_amd.utils.graphstep import GraphedStep
            one_step = GraphedStep(one_step)
        except Exception as e:
            log("graph capture unavailable, running eager: %r" % (e,))

    return one_step, {
        "model": "inception_v3_%d" % args.inc_size,
        "global_batch": args.inc_batch * trainer.roles.n_workers,
        "parallelism": trainer.roles.describe(),
        "optimizer": args.optimizer,
        "lr": args.lr,
        "sync": True,
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=1000)
    p.add_argument("--warmup", type=int, default=100)
    p.add_argument("--workload", default="mnist",
                   choices=["mnist", "nmf", "inception"])
    p.add_argument("--batch", type=int, default=100)
    p.add_argument("--hidden", type=int, default=100)
    p.add_argument("--optimizer", default="sgd")
    p.add_argument("--num-ps", type=int, default=1,
                   help="PS shards (ranks 0..n_ps-1) when world>1")
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph step capture (world==1)")
    p.add_argument("--no-fuse", action="store_true",
                   help="disable the world==1 fused-apply fast paths")
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--nmf-n", type=int, default=1000)
    p.add_argument("--nmf-rank", type=int, default=200)
    p.add_argument("--nmf-batch", type=int, default=256)
    p.add_argument("--inc-batch", type=int, default=32)
    p.add_argument("--inc-size", type=int, default=299)
    p.add_argument("--classes", type=int, default=1000)
    args = p.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    if torch.cuda.is_available():
        dev_idx = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)
    else:
        device = torch.device("cpu")
        log("WARNING: no GPU visible; CPU dev run (not a valid benchmark)")

    one_step, config = {"mnist": run_mnist, "nmf": run_nmf,
                        "inception": run_inception}[args.workload](
        args, device, rank, world)

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        one_step()
    barrier_sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks = whole-job time
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        steps_per_sec = args.steps / elapsed
        # honest topology label: N=1 numbers have NO transport cost
        # (ps+worker colocated in one process); never conflate them
        # with N>1 points of the scaling curve
        if world == 1:
            transport = "none(colocated-1proc)"
        else:
            transport = {"nccl": "rccl"}.get(dist.get_backend(),
                                             dist.get_backend())
        config["transport"] = transport
        out = {
            "metric": "global steps/sec, mnist_replica 1-ps/N-worker at "
                      "1/2/4/8 MI355X" if args.workload == "mnist"
                      else "global steps/sec, %s" % args.workload,
            "value": steps_per_sec,
            "unit": "steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed * 1000.0 / args.steps,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if device.type == "cuda" else "fp32-cpu-devrun",
            "data": "synthetic",
            "config": config,
        }
        print(json.dumps(out))
    getattr(one_step, "finalize", lambda: None)()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
