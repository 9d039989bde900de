"""Helper process for distributed replica tests (gloo, CPU).

Usage: RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT in env;
argv: <mode sync|async> <steps> <out_path_prefix> [n_ps]
Rank 0 saves final (shard-synced) master params to <prefix>.pt.
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch  # noqa: E402
from tfmesos_amd.ps.replica import (  # noqa: E402
    AsyncPSServer, AsyncPSWorker, SyncReplicaTrainer)


def main():
    mode, steps, prefix = sys.argv[1], int(sys.argv[2]), sys.argv[3]
    n_ps = int(sys.argv[4]) if len(sys.argv) > 4 else 1
    variant = sys.argv[5] if len(sys.argv) > 5 else ""
    optimizer = "adam" if mode == "sync-adam" else "sgd"
    lr = 0.01 if optimizer == "adam" else 0.1
    if mode == "sync-adam":
        mode = "sync"

    if mode == "sync-tiny":
        # tiny model whose aligned total (512) yields an EMPTY shard at
        # n_ps=3: collectives/applies must skip it (ADVICE round-1 low)
        torch.manual_seed(7)
        params = [("a", torch.randn(10, 10)), ("b", torch.randn(5))]
        trainer = SyncReplicaTrainer(params, optimizer="sgd",
                                     hparams={"lr": 0.1}, device="cpu",
                                     n_ps=n_ps)
        assert any(hi == lo for lo, hi in trainer.shards), trainer.shards
        for _ in range(steps):
            if trainer.roles.is_worker:
                trainer.grad_view("a").fill_(0.01)
                trainer.grad_view("b").fill_(0.02)
            trainer.step()
        trainer.sync_masters()
        if trainer.rank == 0:
            torch.save({n: trainer.store.view(n).clone()
                        for n in trainer.store.names}, prefix + ".pt")
            torch.save(trainer.store.global_step, prefix + ".step")
        if trainer.world > 1:
            import torch.distributed as dist
            dist.barrier()
            dist.destroy_process_group()
        return

    model = MnistMLP()
    trainer = SyncReplicaTrainer(model.init_params(), optimizer=optimizer,
                                 hparams={"lr": lr}, device="cpu",
                                 n_ps=n_ps, colocate_ps=variant == "colocate",
                                 mode="allreduce" if variant == "allreduce"
                                 else "ps")
    roles = trainer.roles
    # every worker gets the same batch as the single-process reference
    x, y = synthetic_batch(50, seed=42)

    if mode == "sync":
        for _ in range(steps):
            if roles.is_worker:
                model.fwd_bwd(trainer.param, x, y, trainer.grad_view)
            trainer.step()
    elif mode == "async":
        from tfmesos_amd.ps.replica import make_pair_groups
        groups = make_pair_groups(roles)
        if roles.is_ps:
            AsyncPSServer(trainer, groups).serve(steps)
        else:
            w = AsyncPSWorker(trainer, groups)
            for _ in range(steps):
                model.fwd_bwd(trainer.param, x, y, trainer.grad_view)
                w.step()
    elif mode == "async-open":
        # open-ended contract: no step count at the server; workers run
        # UNEVEN step counts (steps + worker_index) then send the stop
        # sentinel (reference async mode is open-ended, README.rst:68-72)
        from tfmesos_amd.ps.replica import make_pair_groups
        groups = make_pair_groups(roles)
        if roles.is_ps:
            AsyncPSServer(trainer, groups).serve()
        else:
            w = AsyncPSWorker(trainer, groups)
            for _ in range(steps + roles.worker_index):
                model.fwd_bwd(trainer.param, x, y, trainer.grad_view)
                w.step()
            w.stop()
    else:
        raise SystemExit("bad mode")

    trainer.sync_masters()
    if trainer.rank == 0:
        torch.save({n: trainer.store.view(n).clone()
                    for n in trainer.store.names}, prefix + ".pt")
        torch.save(trainer.store.global_step, prefix + ".step")
    if trainer.world > 1:
        import torch.distributed as dist
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
