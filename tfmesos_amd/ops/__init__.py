"""Compute ops: hand-written HIP/CDNA4 kernels with CPU torch reference.

Policy (per the MI355X-native design): on a GPU device these ops REQUIRE
the in-tree HIP extension ``tfmesos_amd._C`` (built from
``tfmesos_amd/ops/csrc/*.hip`` for gfx950) and raise if it is missing —
no silent PyTorch fallback on GPU. On CPU (tests, CI without GPUs) they
run a plain PyTorch fp32 reference implementation of the same op; the
numerics tests compare HIP kernel output against these references.

Op inventory mirrors what the reference delegated to TensorFlow's PS
runtime: fused optimizer apply (SGD/momentum, Adagrad,
Adam), bf16 GEMM with fused bias+ReLU epilogue, fused softmax
cross-entropy, embedding gather / scatter-add.
"""

import torch

_EXT = None
_EXT_ERR = None


def _ext():
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from tfmesos_amd import _C  # built in-tree by setup.py/__graft_entry__
            _EXT = _C
        except ImportError as e:
            _EXT_ERR = e
    if _EXT is None:
        raise RuntimeError(
            "tfmesos_amd HIP extension not built (%s). On a GPU box the "
            "native kernels are mandatory — run `python __graft_entry__.py "
            "build` or `python setup.py build_ext --inplace`." % _EXT_ERR)
    return _EXT


def ext_available():
    try:
        _ext()
        return True
    except RuntimeError:
        return False


# --------------------------------------------------------------- optimizers

@torch.no_grad()
def fused_sgd(param, grad, lr, momentum=0.0, weight_decay=0.0, momentum_buf=None,
              bf16_out=None, grad_scale=1.0, neg_decay=0.0):
    """In-place SGD(+momentum, +wd) on an fp32 master param.

    grad may be bf16 or fp32 and is multiplied by grad_scale (worker-mean
    for sync replicas). If bf16_out is given, also writes the updated
    param as bf16 (the broadcast copy) in the same pass. neg_decay folds
    a soft nonnegativity penalty (g += neg_decay * min(p, 0) — the NMF
    objective) into the same pass.
    """
    if param.is_cuda:
        _ext().fused_sgd(param, grad, momentum_buf if momentum_buf is not None
                         else torch.empty(0, device=param.device),
                         bf16_out if bf16_out is not None
                         else torch.empty(0, dtype=torch.bfloat16, device=param.device),
                         float(lr), float(momentum), float(weight_decay),
                         float(grad_scale), float(neg_decay))
        return
    g = grad.float() * grad_scale
    if neg_decay:
        g = g + neg_decay * torch.clamp(param, max=0.0)
    if weight_decay:
        g = g + weight_decay * param
    if momentum and momentum_buf is not None:
        momentum_buf.mul_(momentum).add_(g)
        g = momentum_buf
    param.add_(g, alpha=-lr)
    if bf16_out is not None:
        bf16_out.copy_(param.to(torch.bfloat16))


@torch.no_grad()
def fused_adam(param, grad, exp_avg, exp_avg_sq, step, lr, beta1=0.9,
               beta2=0.999, eps=1e-8, weight_decay=0.0, bf16_out=None,
               grad_scale=1.0):
    """In-place Adam on fp32 master param (bias-corrected, as
    tf.train.AdamOptimizer used by mnist_replica.py:147)."""
    if param.is_cuda:
        _ext().fused_adam(param, grad, exp_avg, exp_avg_sq,
                          bf16_out if bf16_out is not None
                          else torch.empty(0, dtype=torch.bfloat16, device=param.device),
                          int(step), float(lr), float(beta1), float(beta2),
                          float(eps), float(weight_decay), float(grad_scale))
        return
    g = grad.float() * grad_scale
    if weight_decay:
        g = g + weight_decay * param
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param.addcdiv_(exp_avg / bc1, denom, value=-lr)
    if bf16_out is not None:
        bf16_out.copy_(param.to(torch.bfloat16))


@torch.no_grad()
def fused_adagrad(param, grad, accum, lr, eps=1e-10, weight_decay=0.0,
                  bf16_out=None, grad_scale=1.0):
    if param.is_cuda:
        _ext().fused_adagrad(param, grad, accum,
                             bf16_out if bf16_out is not None
                             else torch.empty(0, dtype=torch.bfloat16, device=param.device),
                             float(lr), float(eps), float(weight_decay),
                             float(grad_scale))
        return
    g = grad.float() * grad_scale
    if weight_decay:
        g = g + weight_decay * param
    accum.addcmul_(g, g, value=1.0)
    param.addcdiv_(g, accum.sqrt().add_(eps), value=-lr)
    if bf16_out is not None:
        bf16_out.copy_(param.to(torch.bfloat16))


# --------------------------------------------------------------------- gemm

def gemm_bias_act(a, b, bias=None, act="none", trans_a=False, trans_b=False,
                  out=None, aux=None, colsum_out=None):
    """C = act(op(A) @ op(B) + bias), bf16 in, fp32 accumulate.

    Output dtype: bf16 by default; pass ``out`` (bf16 or fp32) to write
    in place — backward GEMMs write fp32 straight into the flat gradient
    buffer, skipping a cast+copy. Fused epilogues (one kernel each):

    * ``act="relu_bwd"`` with ``aux``: mask the product by ``aux > 0``
      (the saved forward activation) — the dX GEMM absorbs relu_bwd.
    * ``colsum_out``: also write ``sum_k op(B)[k, n]`` (fp32 [N]) — the
      dW GEMM absorbs the bias gradient's column sum.

    GPU: hand-written MFMA kernel (csrc/gemm.hip) with split-K for deep
    skinny shapes. CPU: torch reference in fp32.
    """
    act_code = {"none": 0, "relu": 1, "relu_bwd": 2, "sub": 3}[act]
    if a.is_cuda:
        empty = torch.empty(0, device=a.device)
        ebias = bias if bias is not None else empty
        eaux = aux if aux is not None else empty
        ecs = colsum_out if colsum_out is not None else empty
        if out is None:
            out = torch.empty(
                (a.shape[1] if trans_a else a.shape[0],
                 b.shape[0] if trans_b else b.shape[1]),
                device=a.device, dtype=a.dtype)
        return _ext().gemm_bias_act_out(a, b, ebias, act_code,
                                        bool(trans_a), bool(trans_b), out,
                                        eaux, ecs)
    x = a.float().t() if trans_a else a.float()
    y = b.float().t() if trans_b else b.float()
    c = x @ y
    if bias is not None:
        c = c + bias.float()
    if act == "relu":
        c = torch.relu(c)
    elif act == "relu_bwd":
        c = c * (aux.float() > 0)
    elif act == "sub":
        c = c - aux.float()
    if colsum_out is not None:
        colsum_out.copy_(y.sum(0).to(colsum_out.dtype))
    if out is not None:
        out.copy_(c.to(out.dtype))
        return out
    return c.to(a.dtype)


@torch.no_grad()
def gemm_sgd(a, b, param, lr, shadow=None, grad_scale=1.0, neg_decay=0.0,
             trans_a=False, trans_b=False):
    """Fused GEMM -> SGD: p -= lr*(grad_scale*(op(A)@op(B)) +
    neg_decay*min(p,0)); optional bf16 shadow refresh. The gradient
    never materializes on GPU (the split-K reduce IS the apply)."""
    if a.is_cuda:
        e = torch.empty(0, dtype=torch.bfloat16, device=a.device)
        _ext().gemm_sgd(a, b, bool(trans_a), bool(trans_b),
                        param.view(-1), shadow.view(-1) if shadow is not None
                        else e, float(lr), float(grad_scale),
                        float(neg_decay))
        return
    x = a.float().t() if trans_a else a.float()
    y = b.float().t() if trans_b else b.float()
    g = (x @ y).reshape(param.shape)
    fused_sgd(param, g, lr, grad_scale=grad_scale, neg_decay=neg_decay,
              bf16_out=shadow)


@torch.no_grad()
def gemm_sgd_pair(spec1, spec2, lr, grad_scale=1.0, neg_decay=0.0):
    """Two fused GEMM->SGD updates with SIMULTANEOUS semantics: both
    gradient GEMMs read pre-update operands (their stripe phases run
    before either apply). spec: (a, b, param, shadow, trans_a,
    trans_b)."""
    (a1, b1, p1, s1, ta1, tb1) = spec1
    (a2, b2, p2, s2, ta2, tb2) = spec2
    if a1.is_cuda:
        e = torch.empty(0, dtype=torch.bfloat16, device=a1.device)
        _ext().gemm_sgd_pair(
            a1, b1, bool(ta1), bool(tb1), p1.view(-1),
            s1.view(-1) if s1 is not None else e,
            a2, b2, bool(ta2), bool(tb2), p2.view(-1),
            s2.view(-1) if s2 is not None else e,
            float(lr), float(grad_scale), float(neg_decay))
        return
    def grad(a, b, ta, tb, p):
        x = a.float().t() if ta else a.float()
        y = b.float().t() if tb else b.float()
        return (x @ y).reshape(p.shape)
    g1 = grad(a1, b1, ta1, tb1, p1)
    g2 = grad(a2, b2, ta2, tb2, p2)
    fused_sgd(p1, g1, lr, grad_scale=grad_scale, neg_decay=neg_decay,
              bf16_out=s1)
    fused_sgd(p2, g2, lr, grad_scale=grad_scale, neg_decay=neg_decay,
              bf16_out=s2)


class _FanOutFn(torch.autograd.Function):
    """Explicit fan-out of a tensor to n consumers: backward sums the
    n incoming gradients with ONE n-way add kernel instead of
    autograd's (n-1) pairwise adds (35 of them per Inception step)."""

    @staticmethod
    def forward(ctx, x, n):
        ctx.n = n
        return tuple(x.view_as(x) for _ in range(n))

    @staticmethod
    def backward(ctx, *dys):
        ds = [d for d in dys if d is not None]
        if not ds:
            return None, None
        if len(ds) == 1:
            return ds[0], None
        if ds[0].is_cuda and ds[0].dtype == torch.bfloat16:
            cl = [d.contiguous(memory_format=torch.channels_last)
                  if d.dim() == 4 else d.contiguous() for d in ds]
            return _ext().add_n(cl), None
        out = ds[0].clone()
        for d in ds[1:]:
            out += d
        return out, None


def fan_out(x, n):
    """Returns n differentiable aliases of x (see _FanOutFn)."""
    return _FanOutFn.apply(x, n)


def colsum(x, out=None):
    """out[n] = sum_m x[m,n] in fp32 (bias gradients)."""
    if x.is_cuda:
        return _ext().colsum(x, out if out is not None else
                             torch.empty(0, dtype=torch.float32,
                                         device=x.device))
    s = x.float().sum(0)
    if out is not None:
        out.copy_(s)
        return out
    return s


# ------------------------------------------------------------- softmax-xent

def softmax_xent_fwd(logits, labels):
    """Returns (mean_loss fp32 scalar, probs bf16 [B,C]).

    Fused rowwise softmax + cross-entropy (the reference workload's loss,
    mnist_replica.py:143-145).
    """
    if logits.is_cuda:
        return _ext().softmax_xent_fwd(logits, labels)
    lg = logits.float()
    probs = torch.softmax(lg, dim=1)
    loss = torch.nn.functional.nll_loss(torch.log_softmax(lg, 1), labels)
    return loss, probs.to(logits.dtype)


def softmax_xent_fused(logits, labels, scale=None):
    """Returns (mean_loss fp32 scalar, dlogits bf16 [B,C]) in one fused
    kernel (fwd softmax + loss + bwd (p - onehot)*scale); scale defaults
    to 1/B. The mnist step's loss path is this single launch."""
    B = logits.shape[0]
    s = float(scale if scale is not None else 1.0 / B)
    if logits.is_cuda:
        return _ext().softmax_xent_fused(logits, labels, s)
    lg = logits.float()
    probs = torch.softmax(lg, dim=1)
    loss = torch.nn.functional.nll_loss(torch.log_softmax(lg, 1), labels)
    d = probs.clone()
    d[torch.arange(B), labels] -= 1.0
    return loss, (d * s).to(logits.dtype)


def softmax_xent_bwd(probs, labels, scale=None):
    """dlogits = (probs - onehot) * scale (scale defaults to 1/B)."""
    if probs.is_cuda:
        return _ext().softmax_xent_bwd(
            probs, labels, float(scale if scale is not None else 1.0 / probs.shape[0]))
    B = probs.shape[0]
    s = scale if scale is not None else 1.0 / B
    d = probs.float()
    d[torch.arange(B), labels] -= 1.0
    return (d * s).to(probs.dtype)


# ---------------------------------------------------------------- embedding

def embedding_gather(table, ids):
    """rows = table[ids]; table [V,D] (bf16 or fp32), ids int64 [N]."""
    if table.is_cuda:
        return _ext().embedding_gather(table, ids)
    return table.index_select(0, ids)


def embedding_scatter_add(table, ids, rows):
    """table[ids] += rows (duplicate ids accumulate)."""
    if table.is_cuda:
        _ext().embedding_scatter_add(table, ids, rows)
        return
    table.index_add_(0, ids, rows.to(table.dtype))


# --------------------------------------------------------------- relu bwd

def relu_bwd(grad_out, act):
    """dx = grad_out * (act > 0)."""
    if grad_out.is_cuda:
        return _ext().relu_bwd(grad_out, act)
    return (grad_out.float() * (act.float() > 0)).to(grad_out.dtype)


# --------------------------------------------------------------------- conv

class _Conv2dFn(torch.autograd.Function):
    """Autograd wrapper over the implicit-GEMM HIP conv kernels (NCHW,
    bf16 activations/weights, fp32 weight grads). The Inception path
    (BASELINE.json conv config) builds on this; CPU falls back to the
    torch fp32 reference so model code runs in CI."""

    @staticmethod
    def forward(ctx, x, w, bias, stride, padding, krsc, dw_out=None,
                dw_cb=None):
        ctx.stride = stride
        ctx.padding = padding
        ctx.has_bias = bias is not None
        ctx.krsc = krsc
        ctx.dw_out = dw_out
        ctx.dw_cb = dw_cb
        ctx.pw = False
        if x.is_cuda:
            # channels-last memory: gathers become contiguous channel
            # runs (csrc/conv.hip); weights are tap-major [K,R,S,C] —
            # either natively (krsc params skip the per-call permute)
            # or permuted here from KCRS
            xm = x.contiguous(memory_format=torch.channels_last)
            wm = w.contiguous() if krsc else \
                w.permute(0, 2, 3, 1).contiguous()
            ctx.save_for_backward(xm, w)
            ctx.pw = (wm.shape[1] == 1 and wm.shape[2] == 1
                      and stride == (1, 1) and padding == (0, 0))
            if ctx.pw:
                # pointwise conv IS a GEMM over pixels; the tuned GEMM
                # kernel (csrc/gemm.hip) beats the implicit-GEMM conv
                # path on the large-P Inception shapes (tools/
                # bench_1x1.py: 7.3 vs 9.2 us fwd at 35^2) but loses on
                # 8^2 (its tile grid is too small without split-tap),
                # hence the P gate
                N, C, H, W = xm.shape
                P = N * H * W
                if P >= 4096:
                    x2 = xm.permute(0, 2, 3, 1).reshape(P, C)
                    eb = bias.float() if bias is not None else None
                    y2 = gemm_bias_act(x2, wm.view(wm.shape[0], C),
                                       bias=eb, trans_b=True)
                    return y2.view(N, H, W, -1).permute(0, 3, 1, 2)
            eb = bias.float() if bias is not None else \
                torch.empty(0, device=x.device)
            return _ext().conv2d_fwd(xm, wm, eb, stride[0], stride[1],
                                     padding[0], padding[1], False)
        ctx.save_for_backward(x, w)
        wf = w.permute(0, 3, 1, 2).float() if krsc else w.float()
        y = torch.nn.functional.conv2d(
            x.float(), wf, bias.float() if bias is not None else None,
            stride=stride, padding=padding)
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        krsc = ctx.krsc
        if x.is_cuda:
            # channel-narrow concat-grad views are read strided by the
            # conv kernels (ConvShape.LDY) — no contiguous() copy
            if not _is_cl_narrow(dy):
                dy = dy.contiguous(memory_format=torch.channels_last)
            R, S = (w.shape[1], w.shape[2]) if krsc \
                else (w.shape[2], w.shape[3])
            dx = None
            if ctx.needs_input_grad[0]:
                # kernel reads W in its native [K,R,S,C] layout (the
                # fwd tensor) — krsc params pass straight through
                wk = w if krsc else w.permute(0, 2, 3, 1).contiguous()
                if ctx.pw:
                    # pointwise: dX = dY @ W on the tuned GEMM kernel
                    # (faster than the conv bwd-data path on every
                    # Inception 1x1 shape — tools/bench_1x1.py); the
                    # GEMM path needs a dense 2-D view
                    dy = dy.contiguous(memory_format=torch.channels_last)
                    N, K, Ho, Wo = dy.shape
                    dy2 = dy.permute(0, 2, 3, 1).reshape(N * Ho * Wo, K)
                    dx2 = gemm_bias_act(dy2, wk.reshape(K, -1))
                    dx = dx2.view(N, Ho, Wo, -1).permute(0, 3, 1, 2)
                else:
                    dx = _ext().conv2d_bwd_data(
                        dy, wk, x.shape[2], x.shape[3],
                        ctx.stride[0], ctx.stride[1],
                        ctx.padding[0], ctx.padding[1])
            if ctx.dw_out is not None:
                # grad-arena path (krsc only): atomically accumulate
                # into the model's pre-zeroed fp32 buffer; the trainer
                # gathers it with one batched bf16 copy — no per-layer
                # fill / cast kernels, and autograd sees no w grad.
                _ext().conv2d_bwd_weight_out(
                    dy, x, R, S, ctx.stride[0], ctx.stride[1],
                    ctx.padding[0], ctx.padding[1], ctx.dw_out)
                dw = None
                if ctx.dw_cb is not None:
                    # comm-overlap hook: the dW kernel for this layer is
                    # now ENQUEUED on the compute stream, so a reduce
                    # issued here is stream-ordered after it — the
                    # bucket manager uses this to overlap the PS push
                    # with the rest of backward (ps/module_trainer.py)
                    ctx.dw_cb()
            else:
                dwm = _ext().conv2d_bwd_weight(
                    dy, x, R, S, ctx.stride[0], ctx.stride[1],
                    ctx.padding[0], ctx.padding[1])
                # kernel emits tap-major [K,R,S,C]: native for krsc
                dw = dwm.to(w.dtype) if krsc \
                    else dwm.permute(0, 3, 1, 2).to(w.dtype)
        else:
            dy = dy.contiguous()
            wk = w.permute(0, 3, 1, 2) if krsc else w
            dyf, xf, wf = dy.float(), x.float(), wk.float()
            dx = None
            if ctx.needs_input_grad[0]:
                dx = torch.nn.grad.conv2d_input(
                    x.shape, wf, dyf, stride=ctx.stride,
                    padding=ctx.padding).to(x.dtype)
            dw = torch.nn.grad.conv2d_weight(
                xf, wk.shape, dyf, stride=ctx.stride, padding=ctx.padding)
            if krsc:
                dw = dw.permute(0, 2, 3, 1)
            dw = dw.contiguous().to(w.dtype)
        db = dy.float().sum(dim=(0, 2, 3)) if ctx.has_bias else None
        return dx, dw, db, None, None, None, None, None


def conv2d(x, w, bias=None, stride=1, padding=0, weight_format="kcrs",
           dw_out=None, dw_cb=None):
    """2-D convolution (NCHW activations): hand-written implicit-GEMM
    MFMA kernels on GPU (csrc/conv.hip), torch fp32 reference on CPU.
    Differentiable. weight_format "kcrs" (torch layout) or "krsc"
    (tap-major — the kernels' native layout; parameters stored this way
    skip a permute+copy per call in fwd AND in the weight-grad path).

    dw_out (GPU + krsc only): a pre-zeroed fp32 [K,R,S,C] buffer the
    weight grad is atomically accumulated into instead of being returned
    through autograd (w.grad stays None) — the model grad-arena path
    that batches ~90 per-layer fill/cast kernels into one bulk zero and
    one foreach gather copy per step."""
    if isinstance(stride, int):
        stride = (stride, stride)
    if isinstance(padding, int):
        padding = (padding, padding)
    if dw_out is not None:
        assert weight_format == "krsc" and x.is_cuda, \
            "dw_out requires krsc weights on GPU"
    return _Conv2dFn.apply(x, w, bias, tuple(stride), tuple(padding),
                           weight_format == "krsc", dw_out, dw_cb)


class _SoftmaxXentFn(torch.autograd.Function):
    """Differentiable fused softmax cross-entropy (mean over batch) on
    the HIP kernels; used by autograd models (Inception classifier)."""

    @staticmethod
    def forward(ctx, logits, labels):
        if logits.is_cuda:
            loss, probs = _ext().softmax_xent_fwd(logits, labels)
        else:
            lg = logits.float()
            probs = torch.softmax(lg, 1).to(logits.dtype)
            loss = torch.nn.functional.cross_entropy(lg, labels)
        ctx.save_for_backward(probs, labels)
        return loss

    @staticmethod
    def backward(ctx, gl):
        probs, labels = ctx.saved_tensors
        B = probs.shape[0]
        d = softmax_xent_bwd(probs, labels, scale=1.0 / B)
        if not torch.equal(gl, torch.ones_like(gl)):
            d = d * gl.to(d.dtype)
        return d, None


def softmax_xent_loss(logits, labels):
    """Mean softmax cross-entropy, differentiable w.r.t. logits."""
    return _SoftmaxXentFn.apply(logits, labels)


class _LinearFn(torch.autograd.Function):
    """Differentiable y = x @ w + b on the MFMA GEMM kernels (bf16).
    Backward reuses the fused epilogues: dx = dy w^T, dw = x^T dy with
    the bias-grad colsum fused into the dw GEMM."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        if x.is_cuda:
            return gemm_bias_act(x, w, b)
        y = x.float() @ w.float()
        if b is not None:
            y = y + b.float()
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy = dy.contiguous()
        if x.is_cuda:
            db = torch.empty(w.shape[1], dtype=torch.float32,
                             device=x.device) if ctx.has_bias else None
            dw = torch.empty(w.shape, dtype=torch.float32, device=x.device)
            gemm_bias_act(x, dy, trans_a=True, out=dw, colsum_out=db)
            dx = gemm_bias_act(dy, w, trans_b=True)
            dw = dw.to(w.dtype)
        else:
            dyf = dy.float()
            dx = (dyf @ w.float().t()).to(x.dtype)
            dw = (x.float().t() @ dyf).to(w.dtype)
            db = dyf.sum(0) if ctx.has_bias else None
        return dx, dw, db


def linear(x, w, b=None):
    """Differentiable linear layer on the hand-written MFMA GEMM."""
    return _LinearFn.apply(x, w, b)


def _is_cl_narrow(t):
    """Channels-last tensor OR a channel-narrow view of one (torch.cat's
    backward hands out such views; the HIP BN/pool backward kernels read
    them in place instead of copying)."""
    if t.dim() != 4:
        return False
    s = t.stride()
    return (s[1] == 1 and s[3] >= t.shape[1] and s[2] == t.shape[3] * s[3]
            and s[0] == t.shape[2] * s[2])


class _BatchNormActFn(torch.autograd.Function):
    """Fused train-mode batch-norm (+ optional relu) on the HIP kernels
    (csrc/bn.hip); fp32 torch reference on CPU."""

    @staticmethod
    def forward(ctx, x, weight, bias, eps, relu, out=None):
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
            eout = out if out is not None \
                else torch.empty(0, device=x.device, dtype=torch.bfloat16)
            y, mean, invstd = _ext().bn_fwd(
                x, weight.to(torch.bfloat16), bias.to(torch.bfloat16),
                eps, relu, eout)
            if out is not None:
                # the apply wrote into the caller's concat-buffer view;
                # detach severs the view metadata so autograd attaches
                # this Function's backward instead of rejecting the
                # aliased output (see _JoinViews in models/inception.py)
                ctx.save_for_backward(x, None, weight, bias, mean, invstd)
                ctx.relu = relu
                return y.detach()
            # y is NOT saved for backward: the relu mask is recomputed
            # from sign(g*xhat + b) inside the bwd kernels
        else:
            xf = x.float()
            mean = xf.mean(dim=(0, 2, 3))
            var = xf.var(dim=(0, 2, 3), unbiased=False)
            invstd = torch.rsqrt(var + eps)
            y = (xf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
            y = y * weight.float().view(1, -1, 1, 1) \
                + bias.float().view(1, -1, 1, 1)
            if relu:
                y = torch.relu(y)
            y = y.to(x.dtype)
            if out is not None:
                with torch.no_grad():
                    out.copy_(y)
                ctx.save_for_backward(x, out, weight, bias, mean, invstd)
                ctx.relu = relu
                return out.detach()
        ctx.save_for_backward(x, y, weight, bias, mean, invstd)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, weight, bias, mean, invstd = ctx.saved_tensors
        if x.is_cuda:
            if not _is_cl_narrow(dy):
                dy = dy.contiguous(memory_format=torch.channels_last)
            dx, dgamma, dbeta = _ext().bn_bwd(
                x, dy, weight.to(torch.bfloat16), bias.to(torch.bfloat16),
                mean, invstd, ctx.relu)
        else:
            dyf = dy.contiguous().float()
            if ctx.relu:
                dyf = dyf * (y.float() > 0)
            xf = x.float()
            nhw = x.numel() / x.shape[1]
            xhat = (xf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
            s1 = dyf.sum(dim=(0, 2, 3))
            s2 = (dyf * xhat).sum(dim=(0, 2, 3))
            dgamma, dbeta = s2, s1
            dx = (weight.float() * invstd).view(1, -1, 1, 1) * (
                dyf - (s1 / nhw).view(1, -1, 1, 1)
                - xhat * (s2 / nhw).view(1, -1, 1, 1))
            dx = dx.to(x.dtype)
        return (dx, dgamma.to(weight.dtype), dbeta.to(weight.dtype), None,
                None, None)


class _GroupBNFn(torch.autograd.Function):
    """Grouped train-mode BN(+relu) over the terminal branches of an
    Inception block: ONE stats/finalize/apply launch triple for ALL
    branches (per-channel math — results identical to per-branch BNs;
    at these layer sizes the per-kernel execution floor made launch
    count the dominant BN cost). Writes into a channel-narrow view of
    the block's concat buffer and returns it (same .detach() contract
    as _BatchNormActFn's out= path; see _JoinViews)."""

    @staticmethod
    def forward(ctx, out_view, eps, relu, n, *args):
        xs = [a.contiguous(memory_format=torch.channels_last)
              for a in args[:n]]
        gs = [g.to(torch.bfloat16) for g in args[n:2 * n]]
        bs = [b.to(torch.bfloat16) for b in args[2 * n:3 * n]]
        mean, invstd = _ext().bn_group_fwd(xs, gs, bs, out_view,
                                           eps, relu)
        ctx.save_for_backward(mean, invstd, *xs, *gs, *bs)
        ctx.n = n
        ctx.relu = relu
        return out_view.detach()

    @staticmethod
    def backward(ctx, dy):
        n = ctx.n
        saved = ctx.saved_tensors
        mean, invstd = saved[0], saved[1]
        xs = list(saved[2:2 + n])
        gs = list(saved[2 + n:2 + 2 * n])
        bs = list(saved[2 + 2 * n:2 + 3 * n])
        if not _is_cl_narrow(dy):
            dy = dy.contiguous(memory_format=torch.channels_last)
        outs = _ext().bn_group_bwd(xs, dy, gs, bs, mean, invstd, ctx.relu)
        dxs = outs[:n]
        dgamma, dbeta = outs[n], outs[n + 1]
        dgs, dbs = [], []
        off = 0
        for g in gs:
            c = g.numel()
            dgs.append(dgamma[off:off + c])
            dbs.append(dbeta[off:off + c])
            off += c
        return (None, None, None, None, *dxs, *dgs, *dbs)


class _GroupBNMultiFn(torch.autograd.Function):
    """Grouped BN over PARALLEL inner-stage branches with separate
    dense outputs (e.g. the two 1x1 stems of an Inception block): one
    stats/finalize/apply launch triple instead of one per branch."""

    @staticmethod
    def forward(ctx, eps, relu, n, *args):
        xs = [a.contiguous(memory_format=torch.channels_last)
              for a in args[:n]]
        gs = [g.to(torch.bfloat16) for g in args[n:2 * n]]
        bs = [b.to(torch.bfloat16) for b in args[2 * n:3 * n]]
        outs = _ext().bn_group_fwd_multi(xs, gs, bs, eps, relu)
        ys, mean, invstd = outs[:n], outs[n], outs[n + 1]
        ctx.save_for_backward(mean, invstd, *xs, *gs, *bs)
        ctx.n = n
        ctx.relu = relu
        return tuple(ys)

    @staticmethod
    def backward(ctx, *dys):
        n = ctx.n
        saved = ctx.saved_tensors
        mean, invstd = saved[0], saved[1]
        xs = list(saved[2:2 + n])
        gs = list(saved[2 + n:2 + 2 * n])
        bs = list(saved[2 + 2 * n:2 + 3 * n])
        dys = [d if _is_cl_narrow(d)
               else d.contiguous(memory_format=torch.channels_last)
               for d in dys]
        outs = _ext().bn_group_bwd_multi(xs, dys, gs, bs, mean, invstd,
                                         ctx.relu)
        dxs = outs[:n]
        dgamma, dbeta = outs[n], outs[n + 1]
        dgs, dbs = [], []
        off = 0
        for g in gs:
            c = g.numel()
            dgs.append(dgamma[off:off + c])
            dbs.append(dbeta[off:off + c])
            off += c
        return (None, None, None, *dxs, *dgs, *dbs)


def bn_group_multi(xs, weights, biases, eps=1e-3, relu=True):
    """Grouped BN for parallel branches with separate outputs (GPU
    only). Returns the per-branch normalized tensors."""
    return _GroupBNMultiFn.apply(eps, relu, len(xs), *xs,
                                 *weights, *biases)


def bn_group_apply(out_view, xs, weights, biases, eps=1e-3, relu=True):
    """Grouped BN over parallel branches (GPU only): normalizes each
    xs[i] with (weights[i], biases[i]) and writes the results into
    consecutive channel slices of ``out_view`` (a channels-last or
    channel-narrow view whose C == sum of branch channels)."""
    return _GroupBNFn.apply(out_view, eps, relu, len(xs),
                            *xs, *weights, *biases)


def batch_norm_act(x, weight, bias, eps=1e-3, relu=False, out=None):
    """Differentiable fused train-mode BN (+relu); channels-last on GPU.

    out (optional): a channel-narrow channels-last view the normalized
    output is written into (strided store) — the Inception blocks pass
    slices of a pre-allocated concat buffer so the block concat costs
    zero copies (see models/inception.py _fused_cat)."""
    if not x.is_cuda:
        x = x.contiguous()
    return _BatchNormActFn.apply(x, weight, bias, eps, relu, out)


def mlp_head_fused(h, w, b, labels, scale=None, dw2=None, db2=None):
    """Fused classifier head (one kernel): logits = h@w+b, softmax,
    mean xent loss, dlogits = (p-onehot)*scale, dh = (dlogits@w^T)
    masked by h>0 (h is a relu output), and — when dw2/db2 (fp32 or
    bf16 grad views) are given on the MFMA path (B<=128, H<=128) —
    dW2 = h^T@dlogits and db2 = colsum(dlogits) in the SAME kernel.
    Returns (loss, dlogits, dh). GPU limits C<=16, H<=512, B<=512;
    CPU reference otherwise."""
    B = h.shape[0]
    s = float(scale if scale is not None else 1.0 / B)
    if h.is_cuda:
        e = torch.empty(0, device=h.device)
        return _ext().mlp_head_fused(h, w, b, labels, s,
                                     dw2 if dw2 is not None else e,
                                     db2 if db2 is not None else e)
    hf, wf = h.float(), w.float()
    logits = hf @ wf + b.float()
    probs = torch.softmax(logits, 1)
    loss = torch.nn.functional.cross_entropy(logits, labels)
    d = probs.clone()
    d[torch.arange(B), labels] -= 1.0
    d *= s
    dh = (d @ wf.t()) * (hf > 0)
    db = d.to(h.dtype)
    if dw2 is not None:
        dw2.copy_((hf.t() @ db.float()).reshape(dw2.shape).to(dw2.dtype))
        db2.copy_(db.float().sum(0).to(db2.dtype))
    return loss, db, dh.to(h.dtype)


def mlp_tail_sgd(x, dh, w1_m, w1_s, b1_m, b1_s, g_w2, w2_m, w2_s,
                 g_b2, b2_m, b2_s, lr):
    """mnist single-GPU fused tail (one kernel): dW1 = x^T @ dh on the
    small-tile GEMM with the plain-SGD apply of ALL FOUR params fused
    into the epilogue — W1/b1 straight from the fp32 accumulators
    (the gradient never materializes), W2/b2 by re-reading the small
    bf16 classifier grads the head kernel wrote. Masters are fp32
    flat-store views; shadows the matching bf16 views. Only valid for
    sgd with no momentum/weight-decay and grad_scale 1 (the world==1
    colocated bench config); the PS path applies via fused_sgd.
    GPU-only (raises elsewhere — keep the HIP path the one that runs).
    """
    return _ext().mlp_tail_sgd(x, dh, w1_m, w1_s, b1_m, b1_s,
                               g_w2, w2_m, w2_s, g_b2, b2_m, b2_s,
                               float(lr))


def mlp_fwd_head_fused(x, w1, b1, w2, b2, labels, scale=None,
                       dw2=None, db2=None):
    """Whole MLP fwd + classifier head in TWO kernels (GPU): split-K
    GEMM stripes for h = relu(x@w1+b1), then the fused MFMA head
    consumes the stripes directly — h never exists in global memory.
    Returns (loss, dh); dw2/db2 classifier grads are written by the
    head when given. Limits: B<=128, H<=128 (H%4==0), C<=16.
    CPU: composed fp32 reference."""
    B = x.shape[0]
    s = float(scale if scale is not None else 1.0 / B)
    if x.is_cuda:
        e = torch.empty(0, device=x.device)
        return _ext().mlp_fwd_head_fused(
            x, w1, b1, w2, b2, labels, s,
            dw2 if dw2 is not None else e,
            db2 if db2 is not None else e)
    h = torch.relu(x.float() @ w1.float() + b1.float()).to(x.dtype)
    loss, _, dh = mlp_head_fused(h, w2, b2, labels, scale=s,
                                 dw2=dw2, db2=db2)
    return loss, dh


class _AvgPool3x3Fn(torch.autograd.Function):
    """3x3 stride-1 pad-1 average pool (the Inception block pool) on a
    channels-last HIP stencil kernel; the stencil is symmetric so the
    backward is the same kernel applied to dy."""

    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            return _ext().avg_pool3x3(
                x.contiguous(memory_format=torch.channels_last))
        return torch.nn.functional.avg_pool2d(
            x.float(), 3, stride=1, padding=1).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        if dy.is_cuda:
            return _ext().avg_pool3x3(
                dy.contiguous(memory_format=torch.channels_last))
        return torch.nn.functional.avg_pool2d(
            dy.float(), 3, stride=1, padding=1).to(dy.dtype)


def avg_pool3x3(x):
    """Differentiable 3x3/s1/p1 average pool (count_include_pad)."""
    return _AvgPool3x3Fn.apply(x)


class _MaxPool3x3s2Fn(torch.autograd.Function):
    """3x3 stride-2 max pool (the Inception reduction pool) — forward
    saves the winning tap; backward gathers deterministically."""

    @staticmethod
    def forward(ctx, x, out=None):
        if x.is_cuda:
            x = x.contiguous(memory_format=torch.channels_last)
            e = out if out is not None \
                else torch.empty(0, device=x.device, dtype=x.dtype)
            y, idx = _ext().maxpool3x3s2_fwd(x, e)
            ctx.save_for_backward(idx)
            ctx.hw = (x.shape[2], x.shape[3])
            ctx.gpu = True
            return y
        ctx.gpu = False
        y, idx = torch.nn.functional.max_pool2d(
            x.float(), 3, stride=2, return_indices=True)
        ctx.save_for_backward(idx)
        ctx.xshape = x.shape
        y = y.to(x.dtype)
        if out is not None:
            out.copy_(y)
            return out
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        if ctx.gpu:
            if not _is_cl_narrow(dy):
                dy = dy.contiguous(memory_format=torch.channels_last)
            return _ext().maxpool3x3s2_bwd(dy, idx, *ctx.hw), None
        return torch.nn.functional.max_unpool2d(
            dy.float(), idx, 3, stride=2,
            output_size=ctx.xshape[2:]).to(dy.dtype), None


def max_pool3x3s2(x, out=None):
    """Differentiable 3x3/s2 max pool (no padding). out (GPU): a
    channel-narrow channels-last view — the kernel stores the block's
    concat slice directly (Inception B/D reduction blocks)."""
    return _MaxPool3x3s2Fn.apply(x, out)
