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

    def loss(self):
        E = self.W @ self.H - self.X
        pen = (torch.clamp(self.W, max=0.0) ** 2).sum() + \
              (torch.clamp(self.H, max=0.0) ** 2).sum()
        return (E * E).mean() + self.lam * pen / self.X.numel()

    def one_step(self):
        if self.use_bf16:
            Wb = self.W.to(torch.bfloat16)
            Hb = self.H.to(torch.bfloat16)
            P = ops.gemm_bias_act(Wb, Hb)                      # [n,n] bf16
            E = (P.float() - self.X).to(torch.bfloat16)
            scale = 2.0 / self.X.numel()
            dW = ops.gemm_bias_act(E, Hb, trans_b=True).float() * scale
            dH = ops.gemm_bias_act(Wb, E, trans_a=True).float() * scale
        else:
            E = self.W @ self.H - self.X
            scale = 2.0 / self.X.numel()
            dW = E @ self.H.t() * scale
            dH = self.W.t() @ E * scale
        c = 2.0 * self.lam / self.X.numel()
        dW += c * torch.clamp(self.W, max=0.0)
        dH += c * torch.clamp(self.H, max=0.0)
        self.W -= self.lr * dW
        self.H -= self.lr * dH
        return None
