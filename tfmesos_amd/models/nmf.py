"""Nonnegative matrix factorization — the sparse-model stand-in.

Port of the reference's NMF example (``examples/matrix_factorization.py``:
X[1000,1000] ~= W[1000,200] @ H[200,1000], Frobenius loss + soft
nonnegativity penalty, plain gradient descent; W and H sharded across 2
ps tasks). The MI355X path runs the three GEMMs on the MFMA kernel and
the factor-row push/pull through the sparse embedding path.
"""

import torch

from tfmesos_amd import ops


class NMFWorkload(object):
    """Single-device NMF training step (bench --workload nmf)."""

    def __init__(self, n=1000, rank=200, device="cpu", lr=1e-3, seed=0,
                 nonneg_weight=1.0):
        self.device = torch.device(device)
        self.lr = lr
        self.lam = nonneg_weight
        g = torch.Generator().manual_seed(seed)
        # synthetic nonnegative X with planted low-rank structure
        w0 = torch.rand(n, rank, generator=g)
        h0 = torch.rand(rank, n, generator=g)
        self.X = (w0 @ h0 / rank).to(self.device)
        self.W = (torch.rand(n, rank, generator=g) / rank ** 0.5).to(self.device)
        self.H = (torch.rand(rank, n, generator=g) / rank ** 0.5).to(self.device)
        self.use_bf16 = self.device.type == "cuda"
        if self.use_bf16:
            self.Xb = self.X.to(torch.bfloat16)
            # persistent bf16 shadows: refreshed inside the fused
            # GEMM->SGD reduce, so the step has NO per-step
            # fp32<->bf16 casts of the big tensors and the factor
            # gradients never materialize
            self.Wb = self.W.to(torch.bfloat16)
            self.Hb = self.H.to(torch.bfloat16)

    def loss(self):
        E = self.W @ self.H - self.X
        pen = (torch.clamp(self.W, max=0.0) ** 2).sum() + \
              (torch.clamp(self.H, max=0.0) ** 2).sum()
        return (E * E).mean() + self.lam * pen / self.X.numel()

    def one_step(self):
        scale = 2.0 / self.X.numel()
        c = 2.0 * self.lam / self.X.numel()
        if self.use_bf16:
            # 5 dispatches/step: E = W@H - X in ONE kernel (fused
            # residual epilogue), then each factor's GEMM+apply pair
            # rides the fused GEMM->SGD path — the split-K reduce IS
            # the optimizer update (shadow refresh + nonneg penalty
            # folded in; the gradients never hit HBM as tensors)
            E = ops.gemm_bias_act(self.Wb, self.Hb, act="sub", aux=self.Xb)
            # paired form: BOTH gradient GEMMs read the pre-update
            # factors (simultaneous update, matching the reference's
            # one-train-op apply), then both fused applies run
            ops.gemm_sgd_pair(
                (E, self.Hb, self.W, self.Wb, False, True),
                (self.Wb, E, self.H, self.Hb, True, False),
                lr=self.lr, grad_scale=scale, neg_decay=c)
            return None
        E = self.W @ self.H - self.X
        dW = E @ self.H.t() * scale
        dH = self.W.t() @ E * scale
        dW += c * torch.clamp(self.W, max=0.0)
        dH += c * torch.clamp(self.H, max=0.0)
        self.W -= self.lr * dW
        self.H -= self.lr * dH
        return None


class SparseNMF(object):
    """Distributed minibatch NMF on the sparse-embedding PS
    (BASELINE.json config "matrix_factorization sparse-embedding PS,
    2-ps/6-worker"): W rows on ps:0, H rows on ps:1 (manual model
    parallelism as in the reference, matrix_factorization.py:21-28);
    each worker step pulls only the minibatch's factor rows, computes
    the minibatch Frobenius gradient on MFMA GEMMs, and pushes sparse
    row grads back (HIP gather/scatter-add on a GPU PS).
    """

    def __init__(self, rank, world, device="cpu", n=1000, factor_rank=200,
                 batch=256, lr=0.05, seed=0):
        from tfmesos_amd.ps.replica import init_distributed
        init_distributed(device)
        from tfmesos_amd.ps.sparse import (
            EmbeddingTable, SparsePSServer, SparseWorkerClient,
            make_sparse_pair_groups)

        assert world > 1
        self.rank, self.world = rank, world
        self.device = torch.device(device)
        self.n, self.r, self.batch, self.lr = n, factor_rank, batch, lr
        n_ps = 2 if world > 2 else 1
        ps_ranks = list(range(n_ps))
        worker_ranks = list(range(n_ps, world))
        self.is_ps = rank < n_ps
        homes = {"W": 0, "H": ps_ranks[-1]}
        groups = make_sparse_pair_groups(ps_ranks, worker_ranks)

        # planted low-rank target factors (same seed everywhere);
        # device-resident so the per-step synthetic minibatch target
        # X = w0[I] @ h0[J]^T is computed where the worker computes —
        # as a host matmul it CPU-bound the whole distributed step on
        # boxes with few/oversubscribed cores
        g = torch.Generator().manual_seed(seed)
        self.w0 = torch.rand(n, factor_rank, generator=g).to(self.device)
        self.h0 = torch.rand(n, factor_rank, generator=g).to(self.device)
        self.scale = 1.0 / factor_rank

        if self.is_ps:
            tables = []
            for name, home in homes.items():
                if home == rank:
                    tables.append(EmbeddingTable(
                        name, n, factor_rank, device=device, lr=lr,
                        seed=seed + hash(name) % 1000))
            self.server = SparsePSServer(rank, tables, worker_ranks, groups)
            import threading
            self._srv_thread = threading.Thread(target=self.server.serve,
                                                daemon=True)
            self._srv_thread.start()
        else:
            self.client = SparseWorkerClient(
                rank, homes, {"W": factor_rank, "H": factor_rank}, groups,
                device=device)
            self._step_gen = torch.Generator().manual_seed(777 + rank)
        self.losses = []

    def one_step(self):
        if self.is_ps:
            return None
        from tfmesos_amd import ops
        b = self.batch
        I = torch.randint(0, self.n, (b,), generator=self._step_gen)
        J = torch.randint(0, self.n, (b,), generator=self._step_gen)
        # both pulls in flight at once (W and H live on different PS
        # ranks — sequential pulls paid two full round trips)
        WI, HJ = self.client.pull_many([("W", I), ("H", J)])  # [b,r] bf16
        Id, Jd = I.to(self.device), J.to(self.device)
        X = self.w0[Id] @ self.h0[Jd].t() * self.scale
        if self.device.type == "cuda":
            E = (ops.gemm_bias_act(WI, HJ, trans_b=True).float() - X)
            Eb = E.to(torch.bfloat16)
            dW = ops.gemm_bias_act(Eb, HJ).float() * 2.0
            dH = ops.gemm_bias_act(Eb, WI, trans_a=True).float() * 2.0
        else:
            WIf, HJf = WI.float(), HJ.float()
            E = WIf @ HJf.t() - X
            dW = 2.0 * (E @ HJf)
            dH = 2.0 * (E.t() @ WIf)
        self.client.push_many([("W", I, dW / self.batch),
                               ("H", J, dH / self.batch)])
        loss = float((E * E).mean())
        self.losses.append(loss)
        return loss

    def finalize(self):
        if self.is_ps:
            self._srv_thread.join(timeout=60)
        else:
            self.client.done_all()
