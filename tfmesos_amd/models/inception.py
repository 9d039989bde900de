"""Inception-v3 on the hand-written MFMA conv kernels (BASELINE.json
config "Inception-v3 distributed train 1-ps/8-worker bf16").

The reference cites inception as its canonical heavyweight workload
(``README.rst:72``: distributed inception over tfmesos). Architecture
follows the standard Inception-v3 layout (stem 299x299 -> 5 InceptionA/
B/C/D/E stages -> 8x8 avgpool -> fc 1000); convolutions run on
``tfmesos_amd.ops.conv2d`` (implicit-GEMM MFMA, csrc/conv.hip), the
classifier GEMM on the MFMA GEMM kernel, batch-norm as plain fp32
tensor math (train-mode batch stats) — no MIOpen, no cuDNN shims.
"""

import torch
import torch.nn as nn

from tfmesos_amd import ops


class Conv2d(nn.Module):
    """Conv on the hand-written implicit-GEMM kernel (bias folded into
    the following BN, as in Inception-v3). Weights live in the kernels'
    native tap-major [K, R, S, C] layout — no per-call permute."""

    def __init__(self, cin, cout, kernel_size, stride=1, padding=0):
        super().__init__()
        if isinstance(kernel_size, int):
            kernel_size = (kernel_size, kernel_size)
        self.stride = stride
        self.padding = padding
        fan_in = cin * kernel_size[0] * kernel_size[1]
        self.weight = nn.Parameter(
            torch.randn(cout, kernel_size[0], kernel_size[1], cin)
            * (2.0 / fan_in) ** 0.5)
        self._dw_buf = None    # fp32 view into the model grad arena (GPU)
        self._dw_cb = None     # comm-overlap notify (ps/module_trainer.py)

    def forward(self, x):
        cin_w = self.weight.shape[3]
        if x.shape[1] < cin_w:
            # channel-padded weights (the 3-channel stem input): pad x
            # with zero channels — identical math, and C=8 engages the
            # stager's b128 channel-run path (C=3 forced scalar gathers)
            x = torch.nn.functional.pad(
                x, (0, 0, 0, 0, 0, cin_w - x.shape[1]))
        return ops.conv2d(x, self.weight, None, stride=self.stride,
                          padding=self.padding, weight_format="krsc",
                          dw_out=self._dw_buf if x.is_cuda else None,
                          dw_cb=self._dw_cb if x.is_cuda else None)


class BatchNorm2d(nn.Module):
    """Train-mode batch-norm on the fused HIP kernels (csrc/bn.hip);
    optional fused relu."""

    def __init__(self, num_features, eps=1e-3, relu=False):
        super().__init__()
        self.eps = eps
        self.relu = relu
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))

    def forward(self, x, out=None):
        return ops.batch_norm_act(x, self.weight, self.bias, eps=self.eps,
                                  relu=self.relu, out=out)


class BasicConv2d(nn.Module):
    """conv -> fused BN+relu (one autograd node, 3 kernels fwd)."""

    def __init__(self, cin, cout, **kw):
        super().__init__()
        self.conv = Conv2d(cin, cout, **kw)
        self.bn = BatchNorm2d(cout, relu=True)

    def forward(self, x, out=None):
        return self.bn(self.conv(x), out=out)


def _avg_pool(x, k, stride=1, padding=1):
    if k == 3 and stride == 1 and padding == 1 and x.is_cuda:
        return ops.avg_pool3x3(x)    # hand-written stencil (csrc/pool.hip)
    return torch.nn.functional.avg_pool2d(x, k, stride=stride,
                                          padding=padding)


def _max_pool(x, k, stride, out=None):
    if k == 3 and stride == 2 and x.is_cuda:
        # hand-written (csrc/pool.hip); out = concat-slice view (B/D)
        return ops.max_pool3x3s2(x, out=out)
    y = torch.nn.functional.max_pool2d(x, k, stride=stride)
    if out is not None:
        out.copy_(y)
        return out
    return y


class _JoinViews(torch.autograd.Function):
    """Zero-copy block concat: the branches already wrote their outputs
    into channel slices of one pre-allocated channels-last buffer (the
    BN apply's strided store), so forward just returns the buffer and
    backward hands each branch its channel-narrow gradient VIEW (the
    BN/conv backward kernels read strided dy in place). Replaces the
    aten cat copy per Inception block."""

    @staticmethod
    def forward(ctx, buf, *views):
        ctx.chans = [v.shape[1] for v in views]
        return buf.detach()

    @staticmethod
    def backward(ctx, dout):
        grads, off = [], 0
        for c in ctx.chans:
            grads.append(dout.narrow(1, off, c))
            off += c
        return (None, *grads)


def _fused_cat(x, specs, hw=None):
    """specs: [(channels, builder)] — builder(out_view_or_None) returns
    the branch output. GPU: each branch's terminal BN (or maxpool)
    writes into its slice of one buffer; CPU: plain torch.cat.
    hw: output spatial dims when they differ from x's (the stride-2
    reduction blocks B/D)."""
    if not x.is_cuda:
        return torch.cat([b(None) for _, b in specs], 1)
    N = x.shape[0]
    H, W = hw if hw is not None else (x.shape[2], x.shape[3])
    ctot = sum(c for c, _ in specs)
    buf = torch.empty((N, ctot, H, W), device=x.device, dtype=x.dtype,
                      memory_format=torch.channels_last)
    views, off = [], 0
    for c, b in specs:
        views.append(b(buf.narrow(1, off, c)))
        off += c
    return _JoinViews.apply(buf, *views)


def _group_cat(x, specs, hw=None, pool=None):
    """Grouped-BN block concat. specs: [(C, BasicConv2d, conv_thunk)] —
    conv_thunk() returns the branch's PRE-BN conv output; ONE grouped
    BN (ops.bn_group_apply) normalizes every branch into the concat
    buffer (6 kernels per BLOCK instead of per branch — the per-kernel
    execution floor dominates at these layer sizes). pool: optional
    (C, builder) no-BN branch (the B/D maxpool) writing its own slice.
    CPU: per-branch BN + torch.cat."""
    if not x.is_cuda:
        parts = [m.bn(thunk()) for _, m, thunk in specs]
        if pool is not None:
            parts.append(pool[1](None))
        return torch.cat(parts, 1)
    N = x.shape[0]
    H, W = hw if hw is not None else (x.shape[2], x.shape[3])
    cbn = sum(c for c, _, _ in specs)
    ctot = cbn + (pool[0] if pool is not None else 0)
    buf = torch.empty((N, ctot, H, W), device=x.device, dtype=x.dtype,
                      memory_format=torch.channels_last)
    xs = [thunk() for _, _, thunk in specs]
    y = ops.bn_group_apply(
        buf.narrow(1, 0, cbn), xs,
        [m.bn.weight for _, m, _ in specs],
        [m.bn.bias for _, m, _ in specs],
        eps=specs[0][1].bn.eps, relu=True)
    views = [y]
    if pool is not None:
        views.append(pool[1](buf.narrow(1, cbn, pool[0])))
    return _JoinViews.apply(buf, *views)


def _bn_multi(mods, xs):
    """Grouped BN for parallel inner-stage branches (one launch triple
    for all of them; per-channel math identical to per-branch BN)."""
    if not xs[0].is_cuda:
        return [m.bn(x) for m, x in zip(mods, xs)]
    return list(ops.bn_group_multi(
        xs, [m.bn.weight for m in mods], [m.bn.bias for m in mods],
        eps=mods[0].bn.eps, relu=True))


class InceptionA(nn.Module):
    def __init__(self, cin, pool_features):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 64, kernel_size=1)
        self.b5x5_1 = BasicConv2d(cin, 48, kernel_size=1)
        self.b5x5_2 = BasicConv2d(48, 64, kernel_size=5, padding=2)
        self.b3x3_1 = BasicConv2d(cin, 64, kernel_size=1)
        self.b3x3_2 = BasicConv2d(64, 96, kernel_size=3, padding=1)
        self.b3x3_3 = BasicConv2d(96, 96, kernel_size=3, padding=1)
        self.bpool = BasicConv2d(cin, pool_features, kernel_size=1)
        self.pf = pool_features

    def forward(self, x):
        # explicit fan-out: ONE n-way add sums the branch dx's in
        # backward (autograd would chain n-1 pairwise adds)
        xa, xb, xc, xd = ops.fan_out(x, 4)
        if x.is_cuda:
            y5, y3 = _bn_multi([self.b5x5_1, self.b3x3_1],
                               [self.b5x5_1.conv(xb), self.b3x3_1.conv(xc)])
        else:
            y5, y3 = self.b5x5_1(xb), self.b3x3_1(xc)
        return _group_cat(x, [
            (64, self.b1x1, lambda: self.b1x1.conv(xa)),
            (64, self.b5x5_2, lambda: self.b5x5_2.conv(y5)),
            (96, self.b3x3_3,
             lambda: self.b3x3_3.conv(self.b3x3_2(y3))),
            (self.pf, self.bpool, lambda: self.bpool.conv(_avg_pool(xd, 3))),
        ])


class InceptionB(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b3x3 = BasicConv2d(cin, 384, kernel_size=3, stride=2)
        self.b3x3dbl_1 = BasicConv2d(cin, 64, kernel_size=1)
        self.b3x3dbl_2 = BasicConv2d(64, 96, kernel_size=3, padding=1)
        self.b3x3dbl_3 = BasicConv2d(96, 96, kernel_size=3, stride=2)

    def forward(self, x):
        cin, h, w = x.shape[1], x.shape[2], x.shape[3]
        xa, xb, xc = ops.fan_out(x, 3)
        return _group_cat(x, [
            (384, self.b3x3, lambda: self.b3x3.conv(xa)),
            (96, self.b3x3dbl_3, lambda: self.b3x3dbl_3.conv(
                self.b3x3dbl_2(self.b3x3dbl_1(xb)))),
        ], hw=((h - 3) // 2 + 1, (w - 3) // 2 + 1),
            pool=(cin, lambda o: _max_pool(xc, 3, 2, out=o)))


class InceptionC(nn.Module):
    def __init__(self, cin, c7):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 192, kernel_size=1)
        self.b7_1 = BasicConv2d(cin, c7, kernel_size=1)
        self.b7_2 = BasicConv2d(c7, c7, kernel_size=(1, 7), padding=(0, 3))
        self.b7_3 = BasicConv2d(c7, 192, kernel_size=(7, 1), padding=(3, 0))
        self.b7d_1 = BasicConv2d(cin, c7, kernel_size=1)
        self.b7d_2 = BasicConv2d(c7, c7, kernel_size=(7, 1), padding=(3, 0))
        self.b7d_3 = BasicConv2d(c7, c7, kernel_size=(1, 7), padding=(0, 3))
        self.b7d_4 = BasicConv2d(c7, c7, kernel_size=(7, 1), padding=(3, 0))
        self.b7d_5 = BasicConv2d(c7, 192, kernel_size=(1, 7), padding=(0, 3))
        self.bpool = BasicConv2d(cin, 192, kernel_size=1)

    def forward(self, x):
        xa, xb, xc, xd = ops.fan_out(x, 4)
        if x.is_cuda:
            s1a, s1b = _bn_multi([self.b7_1, self.b7d_1],
                                 [self.b7_1.conv(xb), self.b7d_1.conv(xc)])
            s2a, s2b = _bn_multi(
                [self.b7_2, self.b7d_2],
                [self.b7_2.conv(s1a), self.b7d_2.conv(s1b)])
        else:
            s2a = self.b7_2(self.b7_1(xb))
            s2b = self.b7d_2(self.b7d_1(xc))
        return _group_cat(x, [
            (192, self.b1x1, lambda: self.b1x1.conv(xa)),
            (192, self.b7_3, lambda: self.b7_3.conv(s2a)),
            (192, self.b7d_5, lambda: self.b7d_5.conv(self.b7d_4(
                self.b7d_3(s2b)))),
            (192, self.bpool, lambda: self.bpool.conv(_avg_pool(xd, 3))),
        ])


class InceptionD(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b3_1 = BasicConv2d(cin, 192, kernel_size=1)
        self.b3_2 = BasicConv2d(192, 320, kernel_size=3, stride=2)
        self.b7_1 = BasicConv2d(cin, 192, kernel_size=1)
        self.b7_2 = BasicConv2d(192, 192, kernel_size=(1, 7), padding=(0, 3))
        self.b7_3 = BasicConv2d(192, 192, kernel_size=(7, 1), padding=(3, 0))
        self.b7_4 = BasicConv2d(192, 192, kernel_size=3, stride=2)

    def forward(self, x):
        cin, h, w = x.shape[1], x.shape[2], x.shape[3]
        xa, xb, xc = ops.fan_out(x, 3)
        if x.is_cuda:
            y3, y7 = _bn_multi([self.b3_1, self.b7_1],
                               [self.b3_1.conv(xa), self.b7_1.conv(xb)])
        else:
            y3, y7 = self.b3_1(xa), self.b7_1(xb)
        return _group_cat(x, [
            (320, self.b3_2, lambda: self.b3_2.conv(y3)),
            (192, self.b7_4, lambda: self.b7_4.conv(
                self.b7_3(self.b7_2(y7)))),
        ], hw=((h - 3) // 2 + 1, (w - 3) // 2 + 1),
            pool=(cin, lambda o: _max_pool(xc, 3, 2, out=o)))


class InceptionE(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b1x1 = BasicConv2d(cin, 320, kernel_size=1)
        self.b3_1 = BasicConv2d(cin, 384, kernel_size=1)
        self.b3_2a = BasicConv2d(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b3_2b = BasicConv2d(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.b3d_1 = BasicConv2d(cin, 448, kernel_size=1)
        self.b3d_2 = BasicConv2d(448, 384, kernel_size=3, padding=1)
        self.b3d_3a = BasicConv2d(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b3d_3b = BasicConv2d(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.bpool = BasicConv2d(cin, 192, kernel_size=1)

    def forward(self, x):
        xa, xb, xc, xd = ops.fan_out(x, 4)
        if x.is_cuda:
            b3, y3d = _bn_multi([self.b3_1, self.b3d_1],
                                [self.b3_1.conv(xb), self.b3d_1.conv(xc)])
            b3d = self.b3d_2(y3d)
        else:
            b3 = self.b3_1(xb)
            b3d = self.b3d_2(self.b3d_1(xc))
        b3a, b3b = ops.fan_out(b3, 2)
        b3da, b3db = ops.fan_out(b3d, 2)
        # the nested cats flatten: sub-branch slices are adjacent, so
        # one buffer (and ONE grouped BN) serves the whole block
        return _group_cat(x, [
            (320, self.b1x1, lambda: self.b1x1.conv(xa)),
            (384, self.b3_2a, lambda: self.b3_2a.conv(b3a)),
            (384, self.b3_2b, lambda: self.b3_2b.conv(b3b)),
            (384, self.b3d_3a, lambda: self.b3d_3a.conv(b3da)),
            (384, self.b3d_3b, lambda: self.b3d_3b.conv(b3db)),
            (192, self.bpool, lambda: self.bpool.conv(_avg_pool(xd, 3))),
        ])


class InceptionV3(nn.Module):
    """Standard Inception-v3 (no aux head; train-mode BN)."""

    def __init__(self, num_classes=1000, seed=0):
        super().__init__()
        torch.manual_seed(seed)
        # conv1 takes the input ZERO-PADDED from 3 to 8 channels (see
        # forward): C=3 forced the conv stager's scalar-gather fallback
        # and the single conv1 dispatch measured 155 us (~8 TF/s); at
        # C=8 the b128 channel-run path engages (~5x). The 5 pad input
        # channels of the weight see identically-zero activations, so
        # their gradients are zero and the math is unchanged.
        self.stem = nn.ModuleList([
            BasicConv2d(8, 32, kernel_size=3, stride=2),
            BasicConv2d(32, 32, kernel_size=3),
            BasicConv2d(32, 64, kernel_size=3, padding=1),
        ])
        self.stem2 = nn.ModuleList([
            BasicConv2d(64, 80, kernel_size=1),
            BasicConv2d(80, 192, kernel_size=3),
        ])
        self.mixed = nn.ModuleList([
            InceptionA(192, 32), InceptionA(256, 64), InceptionA(288, 64),
            InceptionB(288),
            InceptionC(768, 128), InceptionC(768, 160),
            InceptionC(768, 160), InceptionC(768, 192),
            InceptionD(768),
            InceptionE(1280), InceptionE(2048),
        ])
        self.fc_w = nn.Parameter(torch.randn(2048, num_classes) * 0.01)
        self.fc_b = nn.Parameter(torch.zeros(num_classes))
        self._arena = None

    def wire_grad_arena(self, device):
        """One fp32 grad arena for every conv weight: each layer's
        bwd-weight kernel atomically accumulates into its slice, so the
        whole backward needs ONE bulk zero (at forward start) instead
        of ~94 per-layer fills. Returns (names, arena) with names in
        arena slice order: ModuleReplicaTrainer lays these params FIRST
        in its flat buffers so the whole arena lands in the bf16 reduce
        buffer with ONE fused cast-copy (a mixed-dtype _foreach_copy_
        de-batches into per-tensor hipMemcpys — measured +0.4 ms/step).
        Constraint: one forward per backward (no cross-step grad
        accumulation) — which is the replica-trainer step pattern.

        Slice starts are 256-element aligned to MATCH PStore.init_params'
        flat layout (ps/store.py:_align): the trainer's one bulk
        ``flat_grad[:na].copy_(arena)`` is only correct if every arena
        param sits at the same offset in both buffers (the round-1 dense
        packing silently shifted 93/94 conv-weight gradients)."""
        from tfmesos_amd.ps.store import _align
        convs = [(n + ".weight", m) for n, m in self.named_modules()
                 if isinstance(m, Conv2d)]
        total = sum(_align(m.weight.numel()) for _, m in convs)
        self._arena = torch.zeros(total, dtype=torch.float32,
                                  device=device)
        off, names = 0, []
        self._arena_offsets = {}
        for name, m in convs:
            n = m.weight.numel()
            m._dw_buf = self._arena[off:off + n].view(m.weight.shape)
            m.weight._tfa_raw_grad = m._dw_buf
            names.append(name)
            self._arena_offsets[name] = off
            off += _align(n)
        return names, self._arena

    def _ensure_arena(self, device):
        if self._arena is not None and self._arena.device == device:
            self._arena.zero_()
            return
        self.wire_grad_arena(device)

    def forward(self, x):
        if x.is_cuda:
            self._ensure_arena(x.device)
        for m in self.stem:
            x = m(x)
        x = _max_pool(x, 3, 2)
        for m in self.stem2:
            x = m(x)
        x = _max_pool(x, 3, 2)
        for m in self.mixed:
            x = m(x)
        x = torch.nn.functional.adaptive_avg_pool2d(x.float(), 1)
        x = x.flatten(1).to(self.fc_w.dtype).contiguous()  # [B, 2048]
        return ops.linear(x, self.fc_w, self.fc_b)


def synthetic_images(batch, size=299, classes=1000, device="cpu",
                     dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.rand(batch, 3, size, size, generator=g)
    y = torch.randint(0, classes, (batch,), generator=g)
    return x.to(device=device, dtype=dtype), y.to(device)
