"""PS replica training for arbitrary ``nn.Module``s — zero-copy wiring.

Generalizes ``SyncReplicaTrainer`` (hand-rolled fwd/bwd models like the
mnist MLP) to autograd models (the Inception config of BASELINE.json):

* every parameter's ``.data`` IS a named view of the PS flat **bf16
  shadow** — the per-step broadcast refreshes module weights in place,
  no per-tensor copies;
* gradients are gathered into the flat **bf16 gradient buffer** with
  one batched ``_foreach_copy_`` per step (``.grad`` stays ``None``
  between steps so autograd assigns instead of accumulating; params
  with an out-of-band fp32 ``_tfa_raw_grad`` — the conv grad arena —
  are converted in the same batched pass), so push is one (sharded)
  ``dist.reduce``;
* apply stays ONE fused HIP kernel per PS shard over fp32 masters
  (bf16 gradients are converted inside the kernel, csrc/apply.hip).
"""

import torch

from tfmesos_amd.ps.replica import SyncReplicaTrainer


class ModuleReplicaTrainer(object):

    def __init__(self, module, optimizer="sgd", hparams=None, device="cpu",
                 n_ps=None, colocate_ps=False):
        self.device = torch.device(device)
        named = list(module.named_parameters())
        # models with a grad arena (Inception convs) get their arena
        # params laid out FIRST in the flat buffers, so the fp32 arena
        # maps onto one contiguous flat_grad region and the per-step
        # gather is a single fused cast-copy
        self._arena = None
        arena_names = []
        if hasattr(module, "wire_grad_arena"):
            # wired on every device so the flat layout (arena params
            # first, 256-aligned) is identical on CPU and GPU; the
            # kernels only write the arena on GPU (conv dw_out is
            # is_cuda-gated), so on CPU the bulk copy moves zeros that
            # the autograd foreach pass then overwrites
            arena_names, self._arena = module.wire_grad_arena(self.device)
        pdict = dict(named)
        order = arena_names + [n for n, _ in named
                               if n not in set(arena_names)]
        params = [(name, pdict[name].detach().float().cpu())
                  for name in order]
        self.t = SyncReplicaTrainer(params, optimizer=optimizer,
                                    hparams=hparams, device=device,
                                    grad_dtype=torch.bfloat16, n_ps=n_ps,
                                    colocate_ps=colocate_ps)
        if self._arena is not None:
            # hard invariant behind the one bulk arena->flat_grad copy:
            # every arena param must sit at the SAME offset in both
            # buffers (store aligns starts to 256; wire_grad_arena must
            # match)
            offs = getattr(module, "_arena_offsets", {})
            for name in arena_names:
                if offs.get(name) != self.t.store.offsets[name][0]:
                    raise RuntimeError(
                        "grad-arena offset mismatch for %s: arena %s vs "
                        "store %s — wire_grad_arena layout must match "
                        "PStore alignment" % (
                            name, offs.get(name),
                            self.t.store.offsets[name][0]))
        self.roles = self.t.roles
        self.module = module
        module.to(device=self.device, dtype=torch.bfloat16)
        self._params = []
        self._gviews = []
        for name, p in module.named_parameters():
            p.data = self.t.store.view(name, bf16=True)
            self._params.append(p)
            self._gviews.append(self.t.grad_view(name))
        # grads stay None between steps: autograd then ASSIGNS (no
        # per-tensor accumulate kernels); step() batch-copies them into
        # the flat reduce buffer with a handful of foreach kernels
        self.t.flat_grad.zero_()

    def zero_grad(self):
        for p in self._params:
            p.grad = None

    def step(self, grad_scale=None):
        """Gather grads into the flat buffer (foreach copy), reduce
        (sharded) -> fused apply on PS -> broadcast shadows. Module
        weights update in place via the shadow views."""
        if self._arena is not None:
            # grad-arena params (conv weights) occupy flat_grad[0:Na]
            # in arena order: ONE fused fp32->bf16 cast-copy gathers
            # all of them (their .grad stays None — the bwd-weight
            # kernels accumulated into the arena directly)
            na = self._arena.numel()
            self.t.flat_grad[:na].copy_(self._arena)
        views, grads = [], []
        for p, gv in zip(self._params, self._gviews):
            if p.grad is not None:
                views.append(gv)
                grads.append(p.grad)
        if grads:
            torch._foreach_copy_(views, grads)
            for p in self._params:
                p.grad = None
        return self.t.step(grad_scale=grad_scale)

    @property
    def store(self):
        return self.t.store

    def save(self, path):
        self.t.save(path)

    def load(self, path):
        self.t.load(path)
