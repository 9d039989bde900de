"""Between-graph replica training: dense PS push/pull over RCCL/xGMI.

Replaces the reference's TF PS data plane for replica mode (grad push +
param pull per step over gRPC, reference ``examples/mnist/
mnist_replica.py:116-157``) with collectives sized for the xGMI fabric:

* **push** = one ``dist.reduce`` of a single flat fp32 gradient buffer
  onto the PS rank (every worker has a DIRECT xGMI link to the PS GPU,
  so the asymmetric PS pattern maps well to the point-to-point fabric);
* **apply** = ONE fused HIP kernel over the flat master buffer
  (optimizer update + bf16 shadow refresh in the same pass);
* **pull** = one ``dist.broadcast`` of the flat bf16 parameter buffer
  from the PS rank.

Sync mode replicates ``tf.train.SyncReplicasOptimizer`` semantics
(aggregate exactly N worker gradients, apply once, all workers see the
new params before the next step — reference ``mnist_replica.py:148-162``):
the reduce IS the aggregation barrier (exactly one contribution per
worker per step, stale gradients impossible by construction), and the
broadcast is the token release.

Async mode (the reference's default ``tfrun`` mode) uses point-to-point
isend/irecv pairs per worker with apply-on-arrival at the PS.

Role mapping when launched by torchrun/bench (one rank per GPU):
world==1 -> colocated ps+worker; world>1 -> rank 0 = ps, 1..W-1 workers.
When launched by the tfmesos_amd launcher, roles come from TFA_JOB_NAME.
"""

import os

import torch
import torch.distributed as dist

from tfmesos_amd.ps.store import PStore


def _env(name, default=None):
    v = os.environ.get(name)
    return v if v is not None else default


def init_distributed(device=None):
    """Init torch.distributed from env (torchrun or tfmesos_amd agent).

    Returns (rank, world_size). No-op returning (0, 1) when WORLD_SIZE
    is absent or 1 (single-process mode).
    """
    world = int(_env("WORLD_SIZE", _env("TFA_WORLD_SIZE", "1")))
    if world <= 1:
        return 0, 1
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    rank = int(_env("RANK", _env("TFA_RANK", "0")))
    backend = "nccl" if (device and str(device).startswith("cuda")) else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world


class Roles(object):
    def __init__(self, rank, world):
        self.rank = rank
        self.world = world
        if world == 1:
            self.is_ps = True
            self.is_worker = True
            self.n_workers = 1
            self.ps_rank = 0
            self.worker_index = 0
        else:
            self.ps_rank = 0
            self.is_ps = rank == 0
            self.is_worker = rank != 0
            self.n_workers = world - 1
            self.worker_index = rank - 1 if rank > 0 else -1

    def describe(self):
        if self.world == 1:
            return "ps1+w1(colocated)"
        return "ps1+w%d" % self.n_workers


class SyncReplicaTrainer(object):
    """Dense sync PS trainer over one flat buffer."""

    def __init__(self, params, optimizer="sgd", hparams=None, device="cpu",
                 grad_dtype=torch.float32):
        """params: dict/list of (name, fp32 tensor) — identical on all
        ranks (same seed)."""
        self.device = torch.device(device)
        self.rank, self.world = init_distributed(device)
        self.roles = Roles(self.rank, self.world)
        hparams = dict(hparams or {})

        # every rank computes the same flat layout; only the PS rank
        # keeps optimizer state
        self.store = PStore(device=device)
        self.store.init_params(params, optimizer=optimizer, **hparams)
        if not self.roles.is_ps:
            self.store.state = {}  # workers don't need optimizer state

        self.flat_params_bf16 = self.store.flat_bf16
        self.flat_grad = torch.zeros_like(self.store.flat, dtype=grad_dtype)

        # initial pull so every worker starts from the PS masters
        if self.world > 1:
            dist.broadcast(self.flat_params_bf16, src=self.roles.ps_rank)

    # parameter views (bf16 working copies, refreshed in-place by pull)
    def param(self, name):
        return self.store.view(name, bf16=True)

    def grad_view(self, name):
        start, numel = self.store.offsets[name]
        return self.flat_grad[start:start + numel].view(self.store.shapes[name])

    def zero_grad(self):
        self.flat_grad.zero_()

    def step(self, grad_scale=None):
        """One global step: reduce grads -> PS apply -> broadcast params.

        The worker must have filled ``flat_grad`` (via ``grad_view``)
        before calling. grad_scale defaults to 1/n_workers (grad mean,
        matching SyncReplicasOptimizer's averaging).
        """
        scale = grad_scale if grad_scale is not None \
            else 1.0 / self.roles.n_workers
        if self.world > 1:
            if self.roles.is_ps and not self.roles.is_worker:
                # dist.reduce is in-place: the PS buffer holds last step's
                # sum and would be re-added — contribute zeros instead
                self.flat_grad.zero_()
            dist.reduce(self.flat_grad, dst=self.roles.ps_rank, op=dist.ReduceOp.SUM)
        if self.roles.is_ps:
            self.store.apply_flat(self.flat_grad, grad_scale=scale)
        if self.world > 1:
            dist.broadcast(self.flat_params_bf16, src=self.roles.ps_rank)
        return self.store.global_step

    def save(self, path):
        if self.roles.is_ps:
            self.store.save(path)

    def load(self, path):
        self.store.load(path)
        if self.world > 1:
            dist.broadcast(self.flat_params_bf16, src=self.roles.ps_rank)


def make_pair_groups(roles):
    """One process group per (ps, worker) pair — every rank must call
    new_group for every pair (it is collective). These give async mode
    independent channels: each PS serving thread blocks on its own
    group, which is safe on both gloo and RCCL (one communicator per
    thread)."""
    groups = {}
    for w in range(1, roles.world):
        groups[w] = dist.new_group([roles.ps_rank, w])
    return groups


class AsyncPSWorker(object):
    """Worker side of async (apply-on-arrival) PS exchange.

    Replicates the reference's default async mode (each worker pushes
    grads and pulls params on its own clock, no aggregation barrier —
    README.rst:68-72)."""

    def __init__(self, trainer, pair_groups):
        self.t = trainer
        assert trainer.world > 1, "async mode needs a separate PS rank"
        self.group = pair_groups[trainer.rank]

    def step(self):
        t = self.t
        dist.send(t.flat_grad, dst=t.roles.ps_rank, group=self.group)
        dist.recv(t.flat_params_bf16, src=t.roles.ps_rank, group=self.group)
        return True

    def stop(self):
        """Tell the PS this worker is done (a zero-length sentinel is not
        expressible; the server counts steps instead)."""


class AsyncPSServer(object):
    """PS side: one serving thread per worker, apply-on-arrival."""

    def __init__(self, trainer, pair_groups):
        self.t = trainer
        self.groups = pair_groups
        assert trainer.roles.is_ps and trainer.world > 1

    def _serve_one(self, worker_rank, steps):
        t = self.t
        buf = torch.zeros_like(t.flat_grad)
        group = self.groups[worker_rank]
        for _ in range(steps):
            dist.recv(buf, src=worker_rank, group=group)
            with t.store.lock:
                t.store.apply_flat(buf)
                params = t.store.flat_bf16.clone()
            dist.send(params, dst=worker_rank, group=group)

    def serve(self, steps_per_worker):
        import threading
        threads = [
            threading.Thread(target=self._serve_one, args=(w, steps_per_worker))
            for w in range(1, self.t.world)
        ]
        for th in threads:
            th.start()
        for th in threads:
            th.join()
        return self.t.store.global_step
