"""Sparse-embedding parameter server: row push/pull over point-to-point
torch.distributed (RCCL/xGMI on GPU, gloo on CPU).

The reference's PS architecture implies sparse variable traffic (row
factors in ``examples/matrix_factorization.py``; TF's PS gathers
embedding rows on pull and scatter-adds gradients on push). Here each
named table lives on ONE ps rank (manual model parallelism, like W on
ps:0 / H on ps:1 in the reference, ``matrix_factorization.py:21-28``);
workers pull only the rows a minibatch touches and push sparse row
gradients back:

* **pull**: worker sends the id vector, the PS gathers bf16 rows with
  the HIP gather kernel (``csrc/embedding.hip``) and sends them back;
* **push**: worker sends (ids, row grads); the PS applies SGD by
  scatter-adding ``-lr * grad`` into the fp32 master with the HIP
  scatter-add kernel (duplicate ids accumulate, matching TF's
  sparse-apply semantics) and refreshes the touched bf16 shadow rows.

Transport is variable-length send/recv on a dedicated (ps, worker)
process group per pair — each PS serving thread blocks on its own
channel, safe on both gloo and RCCL.
"""

import threading

import torch
import torch.distributed as dist

from tfmesos_amd import ops


class EmbeddingTable(object):
    """One PS-resident table: fp32 master + bf16 pull shadow."""

    def __init__(self, name, rows, dim, device="cpu", lr=0.01, seed=0):
        self.name = name
        self.rows = rows
        self.dim = dim
        self.lr = lr
        self.device = torch.device(device)
        g = torch.Generator().manual_seed(seed)
        init = torch.rand(rows, dim, generator=g) / dim ** 0.5
        self.master = init.to(self.device)
        self.shadow = self.master.to(torch.bfloat16)
        self.lock = threading.Lock()

    def pull(self, ids):
        """bf16 rows for ids (HIP gather on GPU)."""
        with self.lock:
            return ops.embedding_gather(self.shadow, ids)

    def push(self, ids, grads, lr=None):
        """SGD sparse apply: master[ids] -= lr*grads (duplicates sum),
        then refresh the touched shadow rows."""
        lr = self.lr if lr is None else lr
        with self.lock:
            g = grads.to(self.master.dtype) * (-lr)
            ops.embedding_scatter_add(self.master, ids, g)
            uniq = torch.unique(ids)
            self.shadow.index_copy_(
                0, uniq, self.master.index_select(0, uniq).to(torch.bfloat16))


def make_sparse_pair_groups(ps_ranks, worker_ranks):
    """One group per (ps, worker) pair; collective — all ranks call."""
    groups = {}
    for p in ps_ranks:
        for w in worker_ranks:
            groups[(p, w)] = dist.new_group([p, w])
    return groups


class SparseWorkerClient(object):
    """Worker-side pull/push to the PS rank owning each table."""

    def __init__(self, rank, table_homes, dims, pair_groups, device="cpu"):
        """table_homes: {table_name: ps_rank}; dims: {name: row_dim}."""
        self.rank = rank
        self.homes = table_homes
        self.dims = dims
        self.groups = pair_groups
        self.device = torch.device(device)

    def _chan(self, name):
        p = self.homes[name]
        return p, self.groups[(p, self.rank)]

    def pull(self, name, ids):
        p, g = self._chan(name)
        hdr = torch.tensor([ids.numel(), self._tidx(name)], dtype=torch.int64)
        dist.send(hdr, dst=p, group=g)
        dist.send(ids.cpu(), dst=p, group=g)
        rows = torch.empty(ids.numel(), self.dims[name], dtype=torch.bfloat16)
        dist.recv(rows, src=p, group=g)
        return rows.to(self.device)

    def push(self, name, ids, grads):
        p, g = self._chan(name)
        hdr = torch.tensor([-ids.numel(), self._tidx(name)],
                           dtype=torch.int64)   # count<0 => push
        dist.send(hdr, dst=p, group=g)
        dist.send(ids.cpu(), dst=p, group=g)
        dist.send(grads.to(torch.bfloat16).cpu().contiguous(), dst=p, group=g)

    def _tidx(self, name):
        mine = sorted(n for n, h in self.homes.items()
                      if h == self.homes[name])
        return mine.index(name)

    def done_all(self):
        """One shutdown marker per PS RANK (its serving thread exits on
        the first zero-count header)."""
        for p in sorted(set(self.homes.values())):
            dist.send(torch.tensor([0, 0], dtype=torch.int64), dst=p,
                      group=self.groups[(p, self.rank)])


class SparsePSServer(object):
    """PS-side server: one thread per worker, fixed request protocol
    (n>0: pull n ids; n<0: push |n| id/grad rows; n==0: worker done)."""

    def __init__(self, rank, tables, worker_ranks, pair_groups):
        self.rank = rank
        self.tables = {t.name: t for t in tables}
        self.worker_ranks = worker_ranks
        self.groups = pair_groups

    def _serve_worker(self, w):
        g = self.groups[(self.rank, w)]
        names = sorted(self.tables)   # header carries the table index
        while True:
            hdr = torch.empty(2, dtype=torch.int64)
            dist.recv(hdr, src=w, group=g)
            count = int(hdr[0].item())
            if count == 0:
                return
            table = self.tables[names[int(hdr[1].item())]]
            ids = torch.empty(abs(count), dtype=torch.int64)
            dist.recv(ids, src=w, group=g)
            dev_ids = ids.to(table.device)
            if count > 0:
                rows = table.pull(dev_ids)
                dist.send(rows.cpu().contiguous(), dst=w, group=g)
            else:
                grads = torch.empty(abs(count), table.dim,
                                    dtype=torch.bfloat16)
                dist.recv(grads, src=w, group=g)
                table.push(dev_ids, grads.to(table.device))

    def serve(self):
        threads = [threading.Thread(target=self._serve_worker, args=(w,))
                   for w in self.worker_ranks]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
