"""Unit tests for the RCCL-side single-threaded polling serve loops.

These paths are selected only under an nccl backend (multiple threads
blocking on RCCL communicators of one device can deadlock), so on a
1-GPU test pool they would first execute during the driver's 8-GPU
run. Fake channels script the worker interactions and verify
apply-on-arrival, bounded mode, and stop-sentinel handling without any
process group."""

import torch

from tfmesos_amd.ps.replica import AsyncPSServer, Roles
from tfmesos_amd.ps.sparse import EmbeddingTable, SparsePSServer
from tfmesos_amd.ps.store import PStore


class FakeWork(object):
    def __init__(self, chan):
        self.chan = chan
        self.done = False

    def is_completed(self):
        if not self.done:
            self.done = self.chan._try_fill()
        return self.done

    def wait(self):
        assert self.is_completed()


class FakeChan(object):
    """Scripted peer: ``script`` is a list of tensors the peer sends in
    order; everything the server sends is recorded."""

    def __init__(self, script):
        self.script = list(script)
        self.sent = []
        self._pending = None     # irecv destination awaiting data
        self.device_only = True  # exercise the nccl decision branches

    def _try_fill(self):
        if self._pending is None:
            return True
        if not self.script:
            return False
        self._pending.copy_(self.script.pop(0))
        self._pending = None
        return True

    def irecv_into(self, t):
        assert self._pending is None
        self._pending = t
        return FakeWork(self)

    def recv_into(self, t):
        assert self._pending is None and self.script, "blocking recv starved"
        t.copy_(self.script.pop(0))
        return t

    def recv_new(self, shape, dtype, device):
        t = torch.empty(shape, dtype=dtype, device=device)
        return self.recv_into(t)

    def send(self, t):
        self.sent.append(t.detach().clone())


class _FakeTrainer(object):
    """Just enough of SyncReplicaTrainer for AsyncPSServer."""

    def __init__(self, n=64, n_workers=2):
        self.store = PStore()
        self.store.init_params([("w", torch.zeros(n))], optimizer="sgd",
                               lr=1.0)
        self.flat_grad = torch.zeros_like(self.store.flat)
        self.my_shard = (0, self.store.flat.numel())
        self.world = 1 + n_workers
        self.rank = 0
        self.roles = Roles(0, self.world, n_ps=1)


def _grad(n, v):
    return torch.full((n,), float(v))


def test_async_polling_open_ended_uneven():
    t = _FakeTrainer()
    n = t.store.flat.numel()
    stop = torch.tensor([0], dtype=torch.int64)
    step = torch.tensor([1], dtype=torch.int64)
    # worker 1: two steps then stop; worker 2: one step then stop
    chans = {
        (0, 1): FakeChan([step, _grad(n, 1.0), step, _grad(n, 2.0), stop]),
        (0, 2): FakeChan([step, _grad(n, 4.0), stop]),
    }
    srv = AsyncPSServer.__new__(AsyncPSServer)
    srv.t = t
    srv.chans = chans
    srv._serve_polling(0, n, None)
    assert t.store.global_step == 3
    # lr=1.0 sgd: master = -(1+2+4)
    assert torch.allclose(t.store.view("w"), torch.full_like(t.store.view("w"), -7.0))
    # each interaction replied with the fresh params
    assert len(chans[(0, 1)].sent) == 2
    assert len(chans[(0, 2)].sent) == 1


def test_async_polling_bounded_mode():
    t = _FakeTrainer()
    n = t.store.flat.numel()
    step = torch.tensor([1], dtype=torch.int64)
    chans = {
        (0, 1): FakeChan([step, _grad(n, 1.0)]),
        (0, 2): FakeChan([step, _grad(n, 2.0)]),
    }
    srv = AsyncPSServer.__new__(AsyncPSServer)
    srv.t = t
    srv.chans = chans
    srv._serve_polling(0, n, 1)   # one step per worker, no sentinel
    assert t.store.global_step == 2
    assert torch.allclose(t.store.view("w"), torch.full_like(t.store.view("w"), -3.0))


def test_sparse_polling_pull_push_stop():
    tbl = EmbeddingTable("W", rows=10, dim=8, lr=0.5, seed=3)
    before = tbl.master.clone()
    ids = torch.tensor([1, 3, 1])
    grads = torch.ones(3, 8, dtype=torch.bfloat16)
    pull_hdr = torch.tensor([3, 0], dtype=torch.int64)
    push_hdr = torch.tensor([-3, 0], dtype=torch.int64)
    stop_hdr = torch.tensor([0, 0], dtype=torch.int64)
    chans = {
        (0, 1): FakeChan([pull_hdr, ids, push_hdr, ids, grads, stop_hdr]),
        (0, 2): FakeChan([stop_hdr]),
    }
    srv = SparsePSServer(0, [tbl], [1, 2], chans)
    srv.device = torch.device("cpu")
    srv._serve_polling()
    # the pull replied with the pre-push rows
    rows = chans[(0, 1)].sent[0]
    assert torch.allclose(rows.float(),
                          before[ids].to(torch.bfloat16).float())
    # the push applied -lr*grads with duplicate accumulation
    want = before.clone()
    want.index_add_(0, ids, -0.5 * grads.float())
    assert torch.allclose(tbl.master, want, atol=1e-2)
