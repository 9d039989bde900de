"""PS-side parameter store: flat fp32 masters + fused HIP apply.

Replaces the reference's TF variable ops on ps tasks (placed by
``replica_device_setter``, reference ``examples/mnist/mnist_replica.py:
116-120``). MI355X-first design choices:

* all parameters live in ONE flat fp32 master buffer (296 GB HBM makes
  resident masters free), with named views;
* a matching flat bf16 shadow is refreshed by the SAME fused apply
  kernel that does the optimizer update — one kernel per step for the
  whole model, no per-tensor launch storm;
* optimizer state (momentum/Adam moments/Adagrad accums) is flat too.
"""

import threading

import torch

from tfmesos_amd import ops


def _align(n, a=256):
    return (n + a - 1) // a * a


class PStore(object):

    def __init__(self, device="cpu"):
        self.device = device
        self.lock = threading.RLock()
        self.global_step = 0
        self.names = []
        self.shapes = {}
        self.offsets = {}   # name -> (start, numel)
        self.flat = None    # fp32 master
        self.flat_bf16 = None
        self.opt = "sgd"
        self.hparams = {}
        self.state = {}     # optimizer state buffers (flat)

    # ---------------------------------------------------------------- init

    def init_params(self, params, optimizer="sgd", **hparams):
        """params: list of (name, tensor-like) or dict name->tensor.

        Tensors may be numpy arrays or torch tensors (any float dtype);
        masters are stored fp32.
        """
        if isinstance(params, dict):
            params = list(params.items())
        with self.lock:
            self.names = [n for n, _ in params]
            tensors = []
            for name, t in params:
                if not isinstance(t, torch.Tensor):
                    t = torch.as_tensor(t)
                t = t.float()
                self.shapes[name] = tuple(t.shape)
                tensors.append(t)
            total = 0
            for name, t in zip(self.names, tensors):
                self.offsets[name] = (total, t.numel())
                total += _align(t.numel())
            self.flat = torch.zeros(total, dtype=torch.float32,
                                    device=self.device)
            self.flat_bf16 = torch.zeros(total, dtype=torch.bfloat16,
                                         device=self.device)
            for name, t in zip(self.names, tensors):
                start, numel = self.offsets[name]
                self.flat[start:start + numel].copy_(t.reshape(-1))
            self.flat_bf16.copy_(self.flat.to(torch.bfloat16))
            self.opt = optimizer
            self.hparams = dict(hparams)
            self.state = {}
            if optimizer == "sgd" and self.hparams.get("momentum", 0.0):
                self.state["momentum_buf"] = torch.zeros_like(self.flat)
            elif optimizer == "adam":
                self.state["exp_avg"] = torch.zeros_like(self.flat)
                self.state["exp_avg_sq"] = torch.zeros_like(self.flat)
            elif optimizer == "adagrad":
                init_acc = self.hparams.get("initial_accumulator", 0.1)
                self.state["accum"] = torch.full_like(self.flat, init_acc)
            self.global_step = 0

    def view(self, name, bf16=False):
        start, numel = self.offsets[name]
        buf = self.flat_bf16 if bf16 else self.flat
        return buf[start:start + numel].view(self.shapes[name])

    # ---------------------------------------------------------------- pull

    def pull(self, names=None, dtype="bf16"):
        """Returns {name: tensor} (cpu tensors when called over RPC)."""
        with self.lock:
            names = names or self.names
            bf16 = (dtype or "bf16") == "bf16"
            return {n: self.view(n, bf16=bf16).clone() for n in names}

    def pull_flat(self, bf16=True):
        return self.flat_bf16 if bf16 else self.flat

    # ---------------------------------------------------------------- push

    def grads_to_flat(self, grads):
        """Pack {name: grad} into a flat fp32 buffer aligned with masters."""
        flat_g = torch.zeros_like(self.flat)
        for name, g in grads.items():
            if not isinstance(g, torch.Tensor):
                g = torch.as_tensor(g)
            start, numel = self.offsets[name]
            flat_g[start:start + numel].copy_(
                g.reshape(-1).to(self.device, torch.float32))
        return flat_g

    def push_apply(self, grads):
        """Apply one gradient set (async-mode semantics: apply on arrival,
        like each worker's independent TF apply). Returns the new step."""
        with self.lock:
            flat_g = grads if isinstance(grads, torch.Tensor) \
                else self.grads_to_flat(grads)
            self.apply_flat(flat_g)
            return self.global_step

    def apply_flat(self, flat_grad, grad_scale=1.0, lo=0, hi=None):
        """One fused optimizer apply over the flat buffer (or the [lo,hi)
        slice of it — a multi-PS shard); also refreshes the bf16 shadow
        in the same kernel. grad_scale folds the sync-replica worker-mean
        into the kernel."""
        with self.lock:
            self.global_step += 1
            hp = self.hparams
            lr = hp.get("lr", 0.01)
            hi = self.flat.numel() if hi is None else hi
            p = self.flat[lo:hi]
            g = flat_grad[lo:hi]
            bf = self.flat_bf16[lo:hi]
            st = {k: v[lo:hi] for k, v in self.state.items()}
            if self.opt == "sgd":
                ops.fused_sgd(p, g, lr,
                              momentum=hp.get("momentum", 0.0),
                              weight_decay=hp.get("weight_decay", 0.0),
                              momentum_buf=st.get("momentum_buf"),
                              bf16_out=bf, grad_scale=grad_scale)
            elif self.opt == "adam":
                ops.fused_adam(p, g, st["exp_avg"], st["exp_avg_sq"],
                               self.global_step, lr,
                               beta1=hp.get("beta1", 0.9),
                               beta2=hp.get("beta2", 0.999),
                               eps=hp.get("eps", 1e-8),
                               weight_decay=hp.get("weight_decay", 0.0),
                               bf16_out=bf, grad_scale=grad_scale)
            elif self.opt == "adagrad":
                ops.fused_adagrad(p, g, st["accum"],
                                  lr, eps=hp.get("eps", 1e-10),
                                  weight_decay=hp.get("weight_decay", 0.0),
                                  bf16_out=bf, grad_scale=grad_scale)
            else:
                raise ValueError("unknown optimizer %r" % self.opt)

    # ---------------------------------------------------------- checkpoint

    def save(self, path, lo=0, hi=None):
        """PS-side checkpoint: params + optimizer state + step (the
        reference delegated this to tf.train.Supervisor's chief; here the
        PS owns it — SURVEY.md §5 checkpoint/resume). With lo/hi, saves
        only that shard's slice (multi-PS: each shard owns its slice)."""
        with self.lock:
            full = lo == 0 and (hi is None or hi == self.flat.numel())
            hi = self.flat.numel() if hi is None else hi
            torch.save({
                "names": self.names,
                "shapes": self.shapes,
                "offsets": self.offsets,
                "shard": None if full else (lo, hi),
                "flat": self.flat[lo:hi].cpu(),
                "opt": self.opt,
                "hparams": self.hparams,
                "state": {k: v[lo:hi].cpu() for k, v in self.state.items()},
                "global_step": self.global_step,
            }, path)

    def load(self, path, lo=0, hi=None):
        """Restore a full checkpoint, or copy a shard checkpoint's slice
        into the (already initialized) buffers."""
        with self.lock:
            ck = torch.load(path, map_location="cpu", weights_only=True)
            shard = ck.get("shard")
            if shard is None and lo == 0 and hi is None:
                self.names = ck["names"]
                self.shapes = ck["shapes"]
                self.offsets = ck["offsets"]
                self.opt = ck["opt"]
                self.hparams = ck["hparams"]
                self.global_step = ck["global_step"]
                # copy INTO the existing buffers when the layout matches:
                # trainers hold live views of flat/flat_bf16 (broadcast
                # source, nn.Module .data aliases) — rebinding would
                # orphan them and train/broadcast stale memory
                if (self.flat is not None
                        and self.flat.numel() == ck["flat"].numel()):
                    self.flat.copy_(ck["flat"].to(self.device))
                    self.flat_bf16.copy_(self.flat.to(torch.bfloat16))
                    for k, v in ck["state"].items():
                        if k in self.state:
                            self.state[k].copy_(v.to(self.device))
                        else:
                            self.state[k] = v.to(self.device)
                else:
                    self.flat = ck["flat"].to(self.device)
                    self.flat_bf16 = self.flat.to(torch.bfloat16)
                    self.state = {k: v.to(self.device)
                                  for k, v in ck["state"].items()}
                return
            cklo, ckhi = shard if shard is not None else (lo, hi)
            if self.flat is None:
                raise RuntimeError("shard checkpoint needs an initialized "
                                   "store (call init_params first)")
            self.flat[cklo:ckhi].copy_(ck["flat"].to(self.device))
            self.flat_bf16[cklo:ckhi].copy_(
                self.flat[cklo:ckhi].to(torch.bfloat16))
            for k, v in ck["state"].items():
                self.state[k][cklo:ckhi].copy_(v.to(self.device))
            self.global_step = ck["global_step"]
