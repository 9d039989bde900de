#!/usr/bin/env python
"""Between-graph replica MNIST training — the benchmark workload.

Port of the reference's ``examples/mnist/mnist_replica.py`` (784 ->
hidden(100) relu -> 10 softmax-xent, batch 100, lr 0.01, optional sync
replicas): each task process runs this script; ps tasks serve parameter
shards, workers compute replicas. Differences from the reference, per
the MI355X-first design:

* data plane is RCCL/xGMI collectives via SyncReplicaTrainer (dense
  sharded reduce/apply/broadcast) instead of TF gRPC;
* synthetic MNIST-shaped data, random-init weights (no network);
* ``--sync_replicas`` maps to the sync trainer (aggregate-N-then-apply,
  reference ``mnist_replica.py:148-162``); default is async
  apply-on-arrival like the reference's default mode.

Launch via the CLI (cmd mode substitutes the placeholders):
  tfa_run -w 2 -s 1 -- python examples/mnist/mnist_replica.py \\
      --ps_hosts {ps_hosts} --worker_hosts {worker_hosts} \\
      --job_name {job_name} --worker_index {task_index} --sync_replicas
or standalone under torchrun (one rank per GPU).
"""

import argparse
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, REPO)

import torch  # noqa: E402

from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch  # noqa: E402
from tfmesos_amd.ps.replica import (  # noqa: E402
    AsyncPSServer, AsyncPSWorker, SyncReplicaTrainer, make_pair_groups)
from tfmesos_amd.utils.metrics import StepTimer  # noqa: E402


def parse_args(argv):
    p = argparse.ArgumentParser()
    # reference flag surface (mnist_replica.py:49-78); data flags accepted
    # but synthetic data is used (no dataset downloads in this build)
    p.add_argument("--data_dir", default="/tmp/mnist-data")
    p.add_argument("--download_only", action="store_true")
    p.add_argument("--worker_index", type=int, default=None)
    p.add_argument("--ps_hosts", default=None)
    p.add_argument("--worker_hosts", default=None)
    p.add_argument("--job_name", default=None)
    p.add_argument("--num_gpus", type=int, default=None)
    p.add_argument("--replicas_to_aggregate", type=int, default=None)
    p.add_argument("--hidden_units", type=int, default=100)
    p.add_argument("--train_steps", type=int, default=200)
    p.add_argument("--batch_size", type=int, default=100)
    p.add_argument("--learning_rate", type=float, default=0.01)
    p.add_argument("--sync_replicas", action="store_true")
    p.add_argument("--optimizer", default="sgd",
                   help="sgd|adam|adagrad (reference used Adam)")
    p.add_argument("--checkpoint", default=None,
                   help="save PS state here at the end; resume if present")
    return p.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    if args.download_only:
        print("synthetic data build: nothing to download")
        return 0

    n_ps = len(args.ps_hosts.split(",")) if args.ps_hosts \
        else int(os.environ.get("TFA_NUM_PS", "1"))

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    if device == "cuda:0":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0"))
                              if "LOCAL_RANK" in os.environ else 0)

    model = MnistMLP(hidden_units=args.hidden_units)
    trainer = SyncReplicaTrainer(
        model.init_params(), optimizer=args.optimizer,
        hparams={"lr": args.learning_rate}, device=device, n_ps=n_ps)
    roles = trainer.roles

    if args.checkpoint and os.path.exists(args.checkpoint):
        trainer.load(args.checkpoint)
        print("resumed from %s at step %d"
              % (args.checkpoint, trainer.store.global_step))

    dtype = torch.bfloat16 if device != "cpu" else torch.float32
    x, y = synthetic_batch(args.batch_size, device=device, dtype=dtype,
                           seed=1000 + trainer.rank)
    pview = (lambda n: trainer.store.view(n, bf16=True)) \
        if dtype == torch.bfloat16 else trainer.param

    job = args.job_name or os.environ.get("TFA_JOB_NAME") or \
        ("ps" if roles.is_ps and not roles.is_worker else "worker")
    print("job=%s rank=%d/%d roles=%s device=%s mode=%s"
          % (job, trainer.rank, trainer.world, roles.describe(), device,
             "sync" if args.sync_replicas else "async"))

    timer = StepTimer(report_every=max(20, args.train_steps // 10),
                      prefix="[%s:%d] " % (job, roles.worker_index)).start()
    if args.sync_replicas or trainer.world == 1:
        for step in range(args.train_steps):
            loss = None
            if roles.is_worker:
                loss = model.fwd_bwd(pview, x, y, trainer.grad_view)
            trainer.step()
            timer.step()
            if roles.is_worker and step % 20 == 0:
                print("step %d loss %.4f" % (step, float(loss)))
    else:
        groups = make_pair_groups(roles)
        if roles.is_ps:
            # open-ended like the reference's async mode: serve until
            # every worker sends its stop sentinel (README.rst:68-72)
            AsyncPSServer(trainer, groups).serve()
        else:
            w = AsyncPSWorker(trainer, groups)
            for step in range(args.train_steps):
                loss = model.fwd_bwd(pview, x, y, trainer.grad_view)
                w.step()
                timer.step()
                if step % 20 == 0:
                    print("step %d loss %.4f" % (step, float(loss)))
            w.stop()
    if device != "cpu":
        torch.cuda.synchronize()
    s = timer.summary()
    print("training done in %.3fs (%.1f steps/s)"
          % (s["elapsed_s"], s["steps_per_sec"]))

    # validation xent on the chief (reference prints it at the end,
    # mnist_replica.py:213-226)
    if roles.is_worker and roles.worker_index in (0, -1) or trainer.world == 1:
        vx, vy = synthetic_batch(1000, device=device, dtype=dtype, seed=7)
        val = model.loss_only(pview, vx, vy)
        steps_done = trainer.store.global_step or args.train_steps
        print("after %d steps validation xent = %.5f"
              % (steps_done, float(val)))

    if args.checkpoint:
        trainer.save(args.checkpoint)
    if trainer.world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
