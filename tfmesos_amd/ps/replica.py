"""Between-graph replica training: dense PS push/pull over RCCL/xGMI.

Replaces the reference's TF PS data plane for replica mode (grad push +
param pull per step over gRPC, reference ``examples/mnist/
mnist_replica.py:116-157``) with collectives sized for the xGMI fabric:

* **push** = one ``dist.reduce`` per PS shard of the flat fp32 gradient
  buffer onto that shard's PS rank (every worker has a DIRECT xGMI link
  to every PS GPU, so the asymmetric PS pattern maps well to the
  point-to-point fabric, and multiple shards spread root bandwidth
  across links — the xGMI-idiomatic form of the reference's multi-ps
  variable sharding, ``examples/mnist/mnist.py:43``);
* **apply** = ONE fused HIP kernel per shard over the flat master buffer
  (optimizer update + bf16 shadow refresh in the same pass);
* **pull** = one ``dist.broadcast`` per shard of the flat bf16 parameter
  buffer from that shard's PS rank.

Sync mode replicates ``tf.train.SyncReplicasOptimizer`` semantics
(aggregate exactly N worker gradients, apply once, all workers see the
new params before the next step — reference ``mnist_replica.py:148-162``):
the reduce IS the aggregation barrier (exactly one contribution per
worker per step, stale gradients impossible by construction), and the
broadcast is the token release.

Async mode (the reference's default ``tfrun`` mode) uses point-to-point
send/recv pairs per (ps, worker) with apply-on-arrival at the PS.

Role mapping when launched by torchrun/bench (one rank per GPU):
world==1 -> colocated ps+worker; world>1 -> ranks [0, n_ps) are PS
shards, the rest workers. When launched by the tfmesos_amd launcher,
roles come from TFA_JOB_NAME/TFA_NUM_PS.
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
    # launcher-provided backend wins (a mixed CPU-ps/GPU-worker cluster
    # must agree on gloo); otherwise pick by device
    backend = _env("TFA_DIST_BACKEND") or (
        "nccl" if (device and str(device).startswith("cuda")) else "gloo")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(backend=backend, rank=rank, world_size=world)
    return rank, world


def _staged(t):
    """gloo collectives need CPU tensors; nccl wants device-resident."""
    return t.cpu() if (t.is_cuda and dist.get_backend() == "gloo") else t


def reduce_(t, dst):
    s = _staged(t)
    dist.reduce(s, dst=dst, op=dist.ReduceOp.SUM)
    if s is not t:
        t.copy_(s)


def broadcast_(t, src):
    s = _staged(t)
    dist.broadcast(s, src=src)
    if s is not t:
        t.copy_(s)


def all_reduce_(t):
    s = _staged(t)
    dist.all_reduce(s, op=dist.ReduceOp.SUM)
    if s is not t:
        t.copy_(s)


def reduce_batch(slices_dsts):
    """Issue ALL per-shard reduces async, then wait once: shards with
    different PS roots progress concurrently instead of serializing one
    blocking collective per shard in a Python loop (round-1 weak #3 —
    on xGMI each root's reduce uses that root's own links, so the
    shard reduces are bandwidth-disjoint and should overlap)."""
    works = []
    for t, dst in slices_dsts:
        s = _staged(t)
        w = dist.reduce(s, dst=dst, op=dist.ReduceOp.SUM, async_op=True)
        works.append((w, s, t))
    for w, s, t in works:
        w.wait()
        if s is not t:
            t.copy_(s)


def broadcast_batch(slices_srcs):
    """Async-issued per-shard broadcasts, one wait (see reduce_batch)."""
    works = []
    for t, src in slices_srcs:
        s = _staged(t)
        w = dist.broadcast(s, src=src, async_op=True)
        works.append((w, s, t))
    for w, s, t in works:
        w.wait()
        if s is not t:
            t.copy_(s)


class Roles(object):
    """Rank -> role mapping. Ranks [0, n_ps) are PS shards (reference:
    ps tasks come first in the jobs_def and hold param slices), the rest
    are workers. world==1 colocates one ps and one worker.

    ``colocate_ps=True`` makes the PS-shard ranks ALSO workers (the
    launcher packs a ps task and a worker task onto the same GPU when
    resources allow): every rank computes a replica, so N GPUs give N
    workers — "1-ps/N-worker" on N devices."""

    def __init__(self, rank, world, n_ps=1, colocate_ps=False):
        self.rank = rank
        self.world = world
        self.colocated = colocate_ps or world == 1
        if world == 1:
            self.n_ps = 1
            self.is_ps = True
            self.is_worker = True
            self.n_workers = 1
            self.ps_rank = 0
            self.ps_ranks = [0]
            self.worker_index = 0
        elif colocate_ps:
            n_ps = max(1, min(int(n_ps), world))
            self.n_ps = n_ps
            self.ps_ranks = list(range(n_ps))
            self.ps_rank = 0
            self.is_ps = rank < n_ps
            self.is_worker = True
            self.n_workers = world
            self.worker_index = rank
        else:
            n_ps = max(1, min(int(n_ps), world - 1))
            self.n_ps = n_ps
            self.ps_ranks = list(range(n_ps))
            self.ps_rank = 0
            self.is_ps = rank < n_ps
            self.is_worker = rank >= n_ps
            self.n_workers = world - n_ps
            self.worker_index = rank - n_ps if rank >= n_ps else -1

    @property
    def worker_ranks(self):
        if self.world == 1:
            return [0]
        if self.colocated:
            return list(range(self.world))
        return list(range(self.n_ps, self.world))

    def describe(self):
        if self.world == 1:
            return "ps1+w1(colocated)"
        if self.colocated:
            return "ps%d+w%d(colocated)" % (self.n_ps, self.n_workers)
        return "ps%d+w%d" % (self.n_ps, self.n_workers)


def _shard_ranges(total, n_ps, align=256):
    """Even, 256-element-aligned contiguous slices of the flat buffer.

    Small buffers with many PS ranks can yield EMPTY shards (lo == hi,
    e.g. total<=512 with n_ps=3); every consumer skips collectives and
    applies for those (zero-numel reduce/broadcast is error-prone on
    gloo and pointless everywhere)."""
    bounds = [0]
    for i in range(1, n_ps):
        b = (total * i // n_ps + align - 1) // align * align
        bounds.append(min(b, total))
    bounds.append(total)
    return [(bounds[i], bounds[i + 1]) for i in range(n_ps)]


class SyncReplicaTrainer(object):
    """Dense sync PS trainer over one flat buffer, sharded across n_ps."""

    def __init__(self, params, optimizer="sgd", hparams=None, device="cpu",
                 grad_dtype=torch.float32, n_ps=None, colocate_ps=False,
                 mode="ps"):
        """params: dict/list of (name, fp32 tensor) — identical on all
        ranks (same seed).

        mode="ps" (default): sharded reduce -> PS apply -> broadcast.
        mode="allreduce": when every rank hosts a replica, the PS
        exchange degenerates to ONE all_reduce + a replicated
        deterministic apply on every rank (identical masters everywhere,
        no broadcast) — half the collectives, the xGMI-efficient option
        at scale. Only valid with colocated roles.
        """
        self.device = torch.device(device)
        self.rank, self.world = init_distributed(device)
        if n_ps is None:
            n_ps = int(_env("TFA_NUM_PS", "1"))
        self.mode = mode
        if mode == "allreduce":
            colocate_ps = True
        self.roles = Roles(self.rank, self.world, n_ps=n_ps,
                           colocate_ps=colocate_ps)
        hparams = dict(hparams or {})

        # every rank computes the same flat layout; only PS ranks apply
        # (and then only their own shard) — except allreduce mode, where
        # every rank applies the whole buffer
        self.store = PStore(device=device)
        self.store.init_params(params, optimizer=optimizer, **hparams)
        if not self.roles.is_ps and mode != "allreduce":
            self.store.state = {}  # workers don't need optimizer state

        self.shards = _shard_ranges(self.store.flat.numel(), self.roles.n_ps)
        if self.roles.is_ps and self.world > 1:
            self.my_shard = self.shards[self.rank]
        elif self.world == 1:
            self.my_shard = self.shards[0]
        else:
            self.my_shard = None

        self.flat_params_bf16 = self.store.flat_bf16
        self.flat_grad = torch.zeros_like(self.store.flat, dtype=grad_dtype)

        # initial pull so every worker starts from the PS masters
        self._broadcast_params()

    def _broadcast_params(self):
        if self.world == 1:
            return
        broadcast_batch([(self.flat_params_bf16[lo:hi],
                          self.roles.ps_ranks[i])
                         for i, (lo, hi) in enumerate(self.shards)
                         if hi > lo])

    # parameter views (bf16 working copies, refreshed in-place by pull)
    def param(self, name):
        return self.store.view(name, bf16=True)

    def grad_view(self, name):
        start, numel = self.store.offsets[name]
        return self.flat_grad[start:start + numel].view(self.store.shapes[name])

    def zero_grad(self):
        self.flat_grad.zero_()

    def step(self, grad_scale=None, skip_reduce=False):
        """One global step: reduce grads per shard -> PS apply -> broadcast.

        The worker must have filled ``flat_grad`` (via ``grad_view``)
        before calling. grad_scale defaults to 1/n_workers (grad mean,
        matching SyncReplicasOptimizer's averaging). ``skip_reduce``:
        the caller already reduced flat_grad (the backward-overlap
        bucket manager, ps/module_trainer.py) — go straight to apply.
        """
        scale = grad_scale if grad_scale is not None \
            else 1.0 / self.roles.n_workers
        if self.mode == "allreduce":
            if self.world > 1 and not skip_reduce:
                all_reduce_(self.flat_grad)
            self.store.apply_flat(self.flat_grad, grad_scale=scale)
            return self.store.global_step
        if self.world > 1 and not skip_reduce:
            if self.roles.is_ps and not self.roles.is_worker:
                # dist.reduce is in-place: a PURE ps buffer holds last
                # step's sum and would be re-added — contribute zeros
                # (colocated ps ranks contribute their own worker grads)
                self.flat_grad.zero_()
            reduce_batch([(self.flat_grad[lo:hi], self.roles.ps_ranks[i])
                          for i, (lo, hi) in enumerate(self.shards)
                          if hi > lo])
        if self.my_shard is not None and self.my_shard[1] > self.my_shard[0]:
            lo, hi = self.my_shard
            self.store.apply_flat(self.flat_grad, grad_scale=scale,
                                  lo=lo, hi=hi)
        if self.world > 1:
            broadcast_batch([(self.flat_params_bf16[lo:hi],
                              self.roles.ps_ranks[i])
                             for i, (lo, hi) in enumerate(self.shards)
                             if hi > lo])
        return self.store.global_step

    def sync_masters(self):
        """Broadcast every shard's fp32 masters so ALL ranks hold the
        complete master buffer (e.g. for a full checkpoint or eval)."""
        if self.world > 1:
            for i, (lo, hi) in enumerate(self.shards):
                if hi > lo:
                    broadcast_(self.store.flat[lo:hi],
                               src=self.roles.ps_ranks[i])
        return self.store.flat

    def save(self, path):
        """PS-side checkpoint. Single PS writes ``path``; each shard of a
        multi-PS run writes ``path.ps<i>`` (its own masters + opt state).
        The reference delegated this to tf.train.Supervisor's chief
        (``mnist_replica.py:165-183``); here the PS owns its state."""
        if self.world == 1 or (self.roles.is_ps and self.roles.n_ps == 1):
            self.store.save(path)
        elif self.roles.is_ps:
            self.store.save("%s.ps%d" % (path, self.rank), *self.my_shard)

    def load(self, path):
        """Restore and re-broadcast (others wait at the broadcast — the
        chief-inits/others-wait semantic of the reference Supervisor)."""
        if self.world == 1 or self.roles.n_ps == 1:
            if self.roles.is_ps or self.world == 1:
                self.store.load(path)
        elif self.roles.is_ps:
            self.store.load("%s.ps%d" % (path, self.rank), *self.my_shard)
        self._broadcast_params()


def make_pair_groups(roles):
    """One Chan per (ps, worker) pair — ``new_group`` is collective, so
    every rank calls this (with identical pair enumeration). Async mode
    uses these as independent request/reply channels per pair."""
    from tfmesos_amd.ps.chan import make_pair_chans
    return make_pair_chans(roles.ps_ranks, roles.worker_ranks,
                           my_rank=roles.rank)


# async wire protocol: each interaction on a (ps, worker) channel is
# [flag int64] then payload. Replicates the reference's OPEN-ENDED
# async contract (workers train on their own clock until done,
# README.rst:68-72): the PS never needs a step count up front.
_FLAG_STEP = 1
_FLAG_STOP = 0


class AsyncPSWorker(object):
    """Worker side of async (apply-on-arrival) PS exchange.

    Replicates the reference's default async mode (each worker pushes
    grads and pulls params on its own clock, no aggregation barrier —
    README.rst:68-72). Call ``stop()`` when done: the server exits once
    every worker has stopped (open-ended, like the reference)."""

    def __init__(self, trainer, pair_chans):
        self.t = trainer
        assert trainer.roles.is_worker and trainer.world > 1, \
            "async mode needs separate PS rank(s)"
        self.chans = pair_chans
        dev = trainer.flat_grad.device if dist.get_backend() == "nccl" \
            else torch.device("cpu")
        self._flag = {f: torch.tensor([f], dtype=torch.int64, device=dev)
                      for f in (_FLAG_STEP, _FLAG_STOP)}

    def _chan(self, ps_rank):
        return self.chans[(ps_rank, self.t.rank)]

    def step(self):
        t = self.t
        for i, (lo, hi) in enumerate(t.shards):
            if hi <= lo:
                continue
            c = self._chan(t.roles.ps_ranks[i])
            c.send(self._flag[_FLAG_STEP])
            c.send(t.flat_grad[lo:hi])
        for i, (lo, hi) in enumerate(t.shards):
            if hi <= lo:
                continue
            self._chan(t.roles.ps_ranks[i]).recv_into(
                t.flat_params_bf16[lo:hi])
        return True

    def stop(self):
        """Tell every PS shard this worker is done (sentinel flag)."""
        for i, (lo, hi) in enumerate(self.t.shards):
            if hi > lo:
                self._chan(self.t.roles.ps_ranks[i]).send(
                    self._flag[_FLAG_STOP])


class AsyncPSServer(object):
    """PS-shard side: apply-on-arrival over this rank's shard.

    Two serving strategies by backend:

    * **nccl/RCCL**: ONE polling loop over every worker channel (irecv
      on the flag word, ``Work.is_completed`` is a CUDA-event query).
      Single-threaded by design — N threads blocking on N RCCL
      communicators of one device can deadlock.
    * **gloo**: one blocking thread per worker channel (gloo's p2p
      ``Work.is_completed`` does not flip without ``wait()``, so the
      polling loop cannot make progress there; threads on gloo are
      safe — it is the CPU/test backend).
    """

    def __init__(self, trainer, pair_chans):
        self.t = trainer
        self.chans = pair_chans
        assert trainer.roles.is_ps and trainer.world > 1

    def _handle(self, c, buf, full, lo, hi):
        t = self.t
        c.recv_into(buf)
        with t.store.lock:
            full[lo:hi].copy_(buf)
            t.store.apply_flat(full, lo=lo, hi=hi)
            params = t.store.flat_bf16[lo:hi].clone()
        c.send(params)

    def serve(self, steps_per_worker=None):
        """Serve until every worker sends the stop sentinel (default,
        matching the reference's open-ended async mode — the reference
        never needs a step count up front, README.rst:68-72), or until
        each worker has been served ``steps_per_worker`` steps (bounded
        variant for benchmarks)."""
        t = self.t
        lo, hi = t.my_shard if t.my_shard is not None else (0, 0)
        if hi <= lo:
            return t.store.global_step   # empty shard: nothing to serve
        if dist.get_backend() == "nccl":
            self._serve_polling(lo, hi, steps_per_worker)
        else:
            self._serve_threads(lo, hi, steps_per_worker)
        return t.store.global_step

    def _serve_threads(self, lo, hi, steps_per_worker):
        import threading
        t = self.t

        def one(w):
            c = self.chans[(t.rank, w)]
            buf = torch.zeros(hi - lo, dtype=t.flat_grad.dtype,
                              device=t.flat_grad.device)
            full = torch.zeros_like(t.flat_grad)
            flag = torch.zeros(1, dtype=torch.int64)
            served = 0
            while steps_per_worker is None or served < steps_per_worker:
                c.recv_into(flag)
                if int(flag.item()) == _FLAG_STOP:
                    return
                self._handle(c, buf, full, lo, hi)
                served += 1

        threads = [threading.Thread(target=one, args=(w,))
                   for w in t.roles.worker_ranks]
        for th in threads:
            th.start()
        for th in threads:
            th.join()

    def _serve_polling(self, lo, hi, steps_per_worker):
        import time as _time
        t = self.t
        dev = t.flat_grad.device
        buf = torch.zeros(hi - lo, dtype=t.flat_grad.dtype, device=dev)
        full = torch.zeros_like(t.flat_grad)
        flags, pending, served = {}, {}, {}
        chans = {w: self.chans[(t.rank, w)] for w in t.roles.worker_ranks}
        for w, c in chans.items():
            flags[w] = torch.zeros(1, dtype=torch.int64, device=dev)
            pending[w] = c.irecv_into(flags[w])
            served[w] = 0
        while pending:
            progress = False
            for w in list(pending):
                if not pending[w].is_completed():
                    continue
                progress = True
                c = chans[w]
                if int(flags[w].item()) == _FLAG_STOP:
                    del pending[w]
                    continue
                self._handle(c, buf, full, lo, hi)
                served[w] += 1
                if steps_per_worker is not None \
                        and served[w] >= steps_per_worker:
                    del pending[w]   # bounded mode: no more irecvs
                else:
                    pending[w] = c.irecv_into(flags[w])
            if not progress:
                _time.sleep(1e-4)
