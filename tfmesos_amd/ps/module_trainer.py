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

import threading

import torch
import torch.distributed as dist

from tfmesos_amd.ps.replica import SyncReplicaTrainer


class _BucketOverlap(object):
    """DDP-style comm/compute overlap for the PS push (VERDICT round-1
    next #2): the flat gradient buffer is partitioned into buckets
    ordered by EXPECTED backward readiness (reverse module order, each
    layer's conv-arena slice next to its BN params), and each bucket's
    per-shard ``dist.reduce`` is issued ASYNC as soon as all its grads
    are produced — from inside backward, so under RCCL the collective
    kernels are stream-ordered right after the producing kernels and
    overlap with the rest of backward on the xGMI links.

    Cross-rank safety: buckets fire strictly in their fixed order (a
    ready bucket waits for earlier ones), so every rank issues the
    same collectives in the same order — the NCCL ordering contract.
    Pure-PS ranks (no backward) zero their contribution and issue all
    buckets in the same order at step time.

    Readiness signals: non-arena params via
    ``register_post_accumulate_grad_hook`` (fires on the autograd
    thread at the right stream position); conv grad-arena params via
    the conv backward's ``dw_cb`` (ops/__init__.py _Conv2dFn.backward —
    their ``.grad`` stays None by design)."""

    def __init__(self, mt, module, arena_names, bucket_elems=4 << 20):
        self.t = mt.t
        self.arena = mt._arena
        arena_set = set(arena_names)
        t = self.t

        # readiness key: named_modules registration order ~= forward
        # order; reverse of it ~= backward readiness. Root-level params
        # (classifier head weights like fc_w) are ready FIRST.
        key = {}
        mods = list(module.named_modules())
        for i, (mname, m) in enumerate(mods):
            for pname, _ in m.named_parameters(recurse=False):
                full = (mname + "." + pname) if mname else pname
                key[full] = i if mname else len(mods) + 1
        names = list(t.store.names)
        order = sorted(names, key=lambda n: -key.get(n, 0))

        # aligned flat range per name (start .. next param's start)
        total = t.store.flat.numel()
        starts = sorted((t.store.offsets[n][0], n) for n in names)
        ranges = {}
        for idx, (s, n) in enumerate(starts):
            e = starts[idx + 1][0] if idx + 1 < len(starts) else total
            ranges[n] = (s, e)

        self.buckets = []
        cur, cur_elems = [], 0
        for n in order:
            cur.append(n)
            cur_elems += t.store.offsets[n][1]
            if cur_elems >= bucket_elems:
                self.buckets.append(self._mk_bucket(cur, ranges, arena_set))
                cur, cur_elems = [], 0
        if cur:
            self.buckets.append(self._mk_bucket(cur, ranges, arena_set))
        self.b_of = {}
        for bi, b in enumerate(self.buckets):
            for n in b["names"]:
                self.b_of[n] = bi

        self._params = dict(module.named_parameters())
        self._lock = threading.Lock()
        self._works = []
        self._reset()
        self._hooks = [
            p.register_post_accumulate_grad_hook(self._mk_hook(n))
            for n, p in self._params.items() if n in self.b_of
        ]
        # conv arena params notify from the conv backward instead
        # (their .grad stays None on GPU)
        arena_bufs = {}
        for mname, m in module.named_modules():
            if hasattr(m, "_dw_cb") and hasattr(m, "weight"):
                full = (mname + ".weight") if mname else "weight"
                if full in arena_set:
                    m._dw_cb = self._mk_cb(full)
                    arena_bufs[full] = m

    def _mk_bucket(self, names, ranges, arena_set):
        rs = sorted(ranges[n] for n in names)
        merged = []
        for lo, hi in rs:
            if merged and merged[-1][1] == lo:
                merged[-1][1] = hi
            else:
                merged.append([lo, hi])
        return {"names": list(names),
                "ranges": [tuple(r) for r in merged],
                "arena": [n for n in names if n in arena_set]}

    def _mk_hook(self, name):
        def hook(_param):
            self._mark(name, hooked=True)
        return hook

    def _mk_cb(self, name):
        def cb():
            self._mark(name, hooked=False)
        return cb

    def _reset(self):
        self._state = [{"done": set(), "hooked": []} for _ in self.buckets]
        self._next = 0

    def _mark(self, name, hooked):
        with self._lock:
            bi = self.b_of[name]
            st = self._state[bi]
            if name in st["done"]:
                return
            st["done"].add(name)
            if hooked:
                st["hooked"].append(name)
            self._advance()

    def _advance(self):
        while self._next < len(self.buckets):
            b = self.buckets[self._next]
            st = self._state[self._next]
            if len(st["done"]) < len(b["names"]):
                return
            self._issue(self._next)
            self._next += 1

    def _issue(self, bi, contribute=True):
        t = self.t
        b, st = self.buckets[bi], self._state[bi]
        if contribute:
            na = self.arena.numel() if self.arena is not None else 0
            hooked = set(st["hooked"])
            if na and any(n not in hooked for n in b["arena"]):
                # GPU arena params: one cast-copy per bucket range,
                # stream-ordered after this bucket's dW kernels
                for lo, hi in b["ranges"]:
                    ahi = min(hi, na)
                    if ahi > lo:
                        t.flat_grad[lo:ahi].copy_(self.arena[lo:ahi])
            views, grads = [], []
            for n in st["hooked"]:
                p = self._params[n]
                if p.grad is not None:
                    views.append(t.grad_view(n))
                    grads.append(p.grad)
            if grads:
                torch._foreach_copy_(views, grads)
                for n in st["hooked"]:
                    self._params[n].grad = None
            # params that produced NO grad this step (and no arena
            # write): zero their ranges so stale sums don't reduce
            missing = [n for n in b["names"] if n not in st["done"]
                       and n not in hooked]
            for n in missing:
                if n not in b["arena"] or not na:
                    s, c = t.store.offsets[n]
                    t.flat_grad[s:s + c].zero_()
        for lo, hi in b["ranges"]:
            for i, (slo, shi) in enumerate(t.shards):
                rlo, rhi = max(lo, slo), min(hi, shi)
                if rhi > rlo:
                    w = dist.reduce(t.flat_grad[rlo:rhi],
                                    dst=t.roles.ps_ranks[i],
                                    op=dist.ReduceOp.SUM, async_op=True)
                    self._works.append(w)

    def flush_and_wait(self):
        """Called from step(): issue any remaining buckets (all of
        them, in order, for pure-PS ranks), wait for every reduce."""
        with self._lock:
            if self._next == 0 and not any(
                    st["done"] for st in self._state) \
                    and not self.t.roles.is_worker:
                # pure-PS rank: contribute zeros (dist.reduce is
                # in-place; the buffer holds last step's sums)
                self.t.flat_grad.zero_()
                for bi in range(len(self.buckets)):
                    self._issue(bi, contribute=False)
                self._next = len(self.buckets)
            else:
                while self._next < len(self.buckets):
                    self._issue(self._next)
                    self._next += 1
            works = self._works
            self._works = []
            self._reset()
        for w in works:
            w.wait()


class ModuleReplicaTrainer(object):

    def __init__(self, module, optimizer="sgd", hparams=None, device="cpu",
                 n_ps=None, colocate_ps=False, overlap=True):
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

        # backward/comm overlap (multi-rank PS mode only): bucketed
        # async reduces issued from inside backward. gloo only handles
        # CPU tensors, so overlap requires matching placement.
        self._overlap = None
        backend_ok = self.t.world > 1 and self.t.mode == "ps" and (
            dist.get_backend() == "nccl" or self.device.type == "cpu")
        if overlap and backend_ok:
            self._overlap = _BucketOverlap(self, module, arena_names)

    def zero_grad(self):
        for p in self._params:
            p.grad = None

    def step(self, grad_scale=None):
        """Gather grads into the flat buffer (foreach copy), reduce
        (sharded) -> fused apply on PS -> broadcast shadows. Module
        weights update in place via the shadow views."""
        if self._overlap is not None:
            # bucketed async reduces were issued during backward;
            # finish them and go straight to apply + broadcast
            self._overlap.flush_and_wait()
            return self.t.step(grad_scale=grad_scale, skip_reduce=True)
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
