"""PS replica training for arbitrary ``nn.Module``s — zero-copy wiring.

Generalizes ``SyncReplicaTrainer`` (hand-rolled fwd/bwd models like the
mnist MLP) to autograd models (the Inception config of BASELINE.json):

* every parameter's ``.data`` IS a named view of the PS flat **bf16
  shadow** — the per-step broadcast refreshes module weights in place,
  no per-tensor copies;
* every parameter's ``.grad`` IS a view of the flat **bf16 gradient
  buffer** — autograd accumulates straight into the reduce buffer, so
  push is one (sharded) ``dist.reduce`` with zero gather kernels;
* apply stays ONE fused HIP kernel per PS shard over fp32 masters
  (bf16 gradients are converted inside the kernel, csrc/apply.hip).
"""

import torch

from tfmesos_amd.ps.replica import SyncReplicaTrainer


class ModuleReplicaTrainer(object):

    def __init__(self, module, optimizer="sgd", hparams=None, device="cpu",
                 n_ps=None):
        self.device = torch.device(device)
        params = [(name, p.detach().float().cpu())
                  for name, p in module.named_parameters()]
        self.t = SyncReplicaTrainer(params, optimizer=optimizer,
                                    hparams=hparams, device=device,
                                    grad_dtype=torch.bfloat16, n_ps=n_ps)
        self.roles = self.t.roles
        self.module = module
        module.to(device=self.device, dtype=torch.bfloat16)
        for name, p in module.named_parameters():
            p.data = self.t.store.view(name, bf16=True)
            p.grad = self.t.grad_view(name)

    def zero_grad(self):
        self.t.flat_grad.zero_()

    def step(self, grad_scale=None):
        """Reduce (sharded) -> fused apply on PS -> broadcast shadows.
        Module weights update in place via the shadow views."""
        return self.t.step(grad_scale=grad_scale)

    @property
    def store(self):
        return self.t.store

    def save(self, path):
        self.t.save(path)

    def load(self, path):
        self.t.load(path)
