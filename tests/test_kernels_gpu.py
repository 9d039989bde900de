"""Numerics: every HIP kernel vs a plain PyTorch fp32 reference.

All tests are @gpu — they run on the MI355X box (pytest -m gpu) against
the in-tree gfx950 extension; inputs are random and asymmetric
(transpose-detecting, guide §5.4 rule 16).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def bf(x):
    return x.to(DEV, torch.bfloat16)


# ------------------------------------------------------------------ apply

def test_fused_sgd_matches_reference():
    from tfmesos_amd import ops
    torch.manual_seed(0)
    n = 5000
    p_ref = torch.randn(n)
    g = torch.randn(n)
    p = p_ref.to(DEV)
    mb = torch.zeros(n, device=DEV)
    out = torch.zeros(n, dtype=torch.bfloat16, device=DEV)
    ops.fused_sgd(p, g.to(DEV), lr=0.1, momentum=0.9, weight_decay=0.01,
                  momentum_buf=mb, bf16_out=out, grad_scale=0.5)
    # cpu reference
    mb_ref = torch.zeros(n)
    ops.fused_sgd(p_ref, g, lr=0.1, momentum=0.9, weight_decay=0.01,
                  momentum_buf=mb_ref, bf16_out=None, grad_scale=0.5)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-5)
    assert torch.allclose(mb.cpu(), mb_ref, atol=1e-5)
    assert torch.allclose(out.float().cpu(), p_ref, atol=0.02)


def test_fused_sgd_bf16_grad():
    from tfmesos_amd import ops
    torch.manual_seed(1)
    n = 1000
    p_ref = torch.randn(n)
    g = torch.randn(n).to(torch.bfloat16)
    p = p_ref.to(DEV)
    ops.fused_sgd(p, g.to(DEV), lr=0.1)
    ops.fused_sgd(p_ref, g, lr=0.1)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-5)


def test_fused_adam_matches_reference():
    from tfmesos_amd import ops
    torch.manual_seed(2)
    n = 4096
    p_ref = torch.randn(n)
    p = p_ref.to(DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    m_ref = torch.zeros(n)
    v_ref = torch.zeros(n)
    for step in range(1, 4):
        g = torch.randn(n)
        ops.fused_adam(p, g.to(DEV), m, v, step, lr=0.01)
        ops.fused_adam(p_ref, g, m_ref, v_ref, step, lr=0.01)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-5)
    assert torch.allclose(v.cpu(), v_ref, atol=1e-5)


def test_fused_adagrad_matches_reference():
    from tfmesos_amd import ops
    torch.manual_seed(3)
    n = 3000
    p_ref = torch.randn(n)
    p = p_ref.to(DEV)
    acc = torch.full((n,), 0.1, device=DEV)
    acc_ref = torch.full((n,), 0.1)
    for _ in range(3):
        g = torch.randn(n)
        ops.fused_adagrad(p, g.to(DEV), acc, lr=0.05)
        ops.fused_adagrad(p_ref, g, acc_ref, lr=0.05)
    assert torch.allclose(p.cpu(), p_ref, atol=1e-5)


# ------------------------------------------------------------------- gemm

@pytest.mark.parametrize("ta", [False, True])
@pytest.mark.parametrize("tb", [False, True])
@pytest.mark.parametrize("mnk", [(100, 100, 784), (100, 10, 100),
                                 (33, 65, 17), (128, 128, 128),
                                 (1000, 1000, 200), (784, 100, 100)])
def test_gemm_transposes(ta, tb, mnk):
    from tfmesos_amd import ops
    M, N, K = mnk
    torch.manual_seed(hash((ta, tb, mnk)) % 2**31)
    A = torch.randn(K, M) if ta else torch.randn(M, K)
    B = torch.randn(N, K) if tb else torch.randn(K, N)
    ref = (A.t() if ta else A).float() @ (B.t() if tb else B).float()
    out = ops.gemm_bias_act(bf(A), bf(B), trans_a=ta, trans_b=tb)
    err = (out.float().cpu() - ref).abs().max()
    tol = 0.02 * ref.abs().max() + 0.05
    assert err < tol, (err, tol)


def test_gemm_bias_relu():
    from tfmesos_amd import ops
    torch.manual_seed(10)
    A, B = torch.randn(100, 784), torch.randn(784, 100)
    bias = torch.randn(100)
    ref = torch.relu(A.float() @ B.float() + bias)
    out = ops.gemm_bias_act(bf(A), bf(B), bias.to(DEV), act="relu")
    err = (out.float().cpu() - ref).abs().max()
    assert err < 0.02 * ref.abs().max() + 0.05


def test_gemm_out_fp32():
    from tfmesos_amd import ops
    torch.manual_seed(11)
    A, B = torch.randn(64, 32), torch.randn(32, 48)
    ref = A.float() @ B.float()
    out = torch.zeros(64, 48, device=DEV)
    ops.gemm_bias_act(bf(A), bf(B), out=out)
    assert (out.cpu() - ref).abs().max() < 0.15


def test_gemm_splitk_repeatable():
    """(100,100,784) nn hits the split-K path; per-slice workspace
    stripes summed in fixed order make it bitwise deterministic."""
    from tfmesos_amd import ops
    torch.manual_seed(12)
    A, B = bf(torch.randn(100, 784)), bf(torch.randn(784, 100))
    ref = A.float() @ B.float()
    outs = [ops.gemm_bias_act(A, B) for _ in range(5)]
    for o in outs:
        assert (o.float() - ref).abs().max() < 0.02 * ref.abs().max() + 0.05
        assert torch.equal(o, outs[0])


def test_gemm_splitk_bias_relu_deep():
    from tfmesos_amd import ops
    torch.manual_seed(13)
    A, B = torch.randn(64, 2000), torch.randn(2000, 64)
    bias = torch.randn(64)
    ref = torch.relu(A.float() @ B.float() + bias)
    out = ops.gemm_bias_act(bf(A), bf(B), bias.to(DEV), act="relu")
    assert (out.float().cpu() - ref).abs().max() < 0.02 * ref.abs().max() + 0.1


def test_gemm_relu_bwd_epilogue():
    """dX = (dY @ W^T) * (h > 0) fused into the GEMM."""
    from tfmesos_amd import ops
    torch.manual_seed(14)
    dy, w, h = torch.randn(100, 10), torch.randn(100, 10), torch.randn(100, 100)
    dyb, wb, hb = bf(dy), bf(w), bf(h)
    ref = (dyb.float() @ wb.float().t()) * (hb.float() > 0)
    out = ops.gemm_bias_act(dyb, wb, trans_b=True, act="relu_bwd", aux=hb)
    assert (out.float() - ref).abs().max() < 0.02 * ref.abs().max() + 0.05


def test_gemm_colsum_epilogue():
    """dW = X^T dY with the bias gradient colsum(dY) fused in."""
    from tfmesos_amd import ops
    torch.manual_seed(15)
    x, dy = torch.randn(100, 784), torch.randn(100, 100)
    xb, dyb = bf(x), bf(dy)
    ref_w = xb.float().t() @ dyb.float()
    ref_b = dyb.float().sum(0)
    out = torch.zeros(784, 100, device=DEV)
    cs = torch.zeros(100, device=DEV)
    ops.gemm_bias_act(xb, dyb, trans_a=True, out=out, colsum_out=cs)
    assert (out.cpu() - ref_w.cpu()).abs().max() < 0.3
    assert (cs.cpu() - ref_b.cpu()).abs().max() < 0.05


def test_gemm_odd_leading_dim():
    """ld not divisible by the vector widths falls back to scalar staging."""
    from tfmesos_amd import ops
    torch.manual_seed(16)
    A, B = torch.randn(33, 17), torch.randn(17, 21)
    ref = A.float() @ B.float()
    out = ops.gemm_bias_act(bf(A), bf(B))
    assert (out.float().cpu() - ref).abs().max() < 0.15


# ----------------------------------------------------------- softmax-xent

def test_softmax_xent_fwd_bwd():
    from tfmesos_amd import ops
    torch.manual_seed(20)
    B, C = 100, 10
    logits = torch.randn(B, C) * 3
    labels = torch.randint(0, C, (B,))
    loss, probs = ops.softmax_xent_fwd(bf(logits), labels.to(DEV))
    ref_probs = torch.softmax(logits.float(), 1)
    ref_loss = torch.nn.functional.cross_entropy(logits.float(), labels)
    assert abs(float(loss) - float(ref_loss)) < 0.02
    assert (probs.float().cpu() - ref_probs).abs().max() < 0.01
    d = ops.softmax_xent_bwd(probs, labels.to(DEV))
    ref_d = ref_probs.clone()
    ref_d[torch.arange(B), labels] -= 1
    ref_d /= B
    assert (d.float().cpu() - ref_d).abs().max() < 1e-3


def test_softmax_xent_fused_one_kernel():
    from tfmesos_amd import ops
    torch.manual_seed(22)
    B, C = 100, 10
    logits = torch.randn(B, C) * 3
    labels = torch.randint(0, C, (B,))
    loss, d = ops.softmax_xent_fused(bf(logits), labels.to(DEV))
    ref_probs = torch.softmax(bf(logits).float().cpu(), 1)
    ref_loss = torch.nn.functional.cross_entropy(logits.float(), labels)
    ref_d = ref_probs.clone()
    ref_d[torch.arange(B), labels] -= 1
    ref_d /= B
    assert abs(float(loss) - float(ref_loss)) < 0.02
    assert (d.float().cpu() - ref_d).abs().max() < 1e-3


def test_softmax_xent_fused_large_falls_back():
    """Shapes beyond the one-workgroup kernel compose fwd+bwd correctly."""
    from tfmesos_amd import ops
    torch.manual_seed(23)
    B, C = 512, 100
    logits = torch.randn(B, C) * 2
    labels = torch.randint(0, C, (B,))
    loss, d = ops.softmax_xent_fused(bf(logits), labels.to(DEV))
    ref_loss = torch.nn.functional.cross_entropy(logits.float(), labels)
    assert abs(float(loss) - float(ref_loss)) < 0.05
    assert d.shape == (B, C)


def test_softmax_xent_wide():
    from tfmesos_amd import ops
    torch.manual_seed(21)
    B, C = 32, 1000  # wider than one wave pass
    logits = torch.randn(B, C) * 2
    labels = torch.randint(0, C, (B,))
    loss, probs = ops.softmax_xent_fwd(bf(logits), labels.to(DEV))
    ref_loss = torch.nn.functional.cross_entropy(logits.float(), labels)
    assert abs(float(loss) - float(ref_loss)) < 0.05


# -------------------------------------------------------------- embedding

def test_embedding_gather():
    from tfmesos_amd import ops
    torch.manual_seed(30)
    V, D = 1000, 200
    table = torch.randn(V, D)
    ids = torch.randint(0, V, (77,))
    out = ops.embedding_gather(bf(table), ids.to(DEV))
    ref = table[ids]
    assert (out.float().cpu() - ref).abs().max() < 0.02


def test_embedding_scatter_add_duplicates():
    from tfmesos_amd import ops
    torch.manual_seed(31)
    V, D = 50, 64
    table = torch.zeros(V, D, device=DEV)
    ids = torch.tensor([3, 7, 3, 3, 0])
    rows = torch.randn(5, D)
    ops.embedding_scatter_add(table, ids.to(DEV), rows.to(DEV))
    ref = torch.zeros(V, D)
    ref.index_add_(0, ids, rows)
    assert torch.allclose(table.cpu(), ref, atol=1e-4)


# ------------------------------------------------------------ elementwise

def test_relu_bwd():
    from tfmesos_amd import ops
    torch.manual_seed(40)
    dy = torch.randn(100, 100)
    act = torch.randn(100, 100)
    out = ops.relu_bwd(bf(dy), bf(act))
    ref = dy.to(torch.bfloat16).float() * (act.to(torch.bfloat16).float() > 0)
    assert (out.float().cpu() - ref).abs().max() < 1e-3


def test_colsum():
    from tfmesos_amd import ops
    torch.manual_seed(41)
    x = torch.randn(100, 110)
    out = ops.colsum(bf(x))
    ref = x.to(torch.bfloat16).float().sum(0)
    assert (out.cpu() - ref).abs().max() < 0.05


# ------------------------------------------------------------------- e2e

def test_mlp_step_gpu_vs_cpu():
    """Full fused-kernel train step on GPU tracks the fp32 CPU reference."""
    from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
    from tfmesos_amd.ps.store import PStore
    model = MnistMLP()
    params = model.init_params()

    def run(device, dtype):
        store = PStore(device=device)
        store.init_params([(n, t.clone()) for n, t in params],
                          optimizer="sgd", lr=0.01)
        fg = torch.zeros_like(store.flat)

        def gv(name):
            s, c = store.offsets[name]
            return fg[s:s + c].view(store.shapes[name])

        x, y = synthetic_batch(100, device=device, dtype=dtype)
        losses = []
        pv = (lambda n: store.view(n, bf16=True)) if dtype == torch.bfloat16 \
            else (lambda n: store.view(n))
        for _ in range(10):
            losses.append(float(model.fwd_bwd(pv, x, y, gv)))
            store.apply_flat(fg)
        return losses

    cpu_losses = run("cpu", torch.float32)
    gpu_losses = run(DEV, torch.bfloat16)
    for c, g in zip(cpu_losses, gpu_losses):
        assert abs(c - g) < 0.05 + 0.05 * abs(c), (cpu_losses, gpu_losses)
    assert gpu_losses[-1] < gpu_losses[0]


def test_native_ext_is_mandatory_on_gpu(monkeypatch):
    """On GPU, ops must fail loudly if the extension is missing."""
    import tfmesos_amd.ops as ops
    monkeypatch.setattr(ops, "_EXT", None)
    monkeypatch.setattr(ops, "_EXT_ERR", ImportError("simulated"))
    x = torch.randn(4, 4, device=DEV, dtype=torch.bfloat16)
    with pytest.raises(RuntimeError, match="extension not built"):
        ops.gemm_bias_act(x, x)


# ------------------------------------------------------------------- conv

@pytest.mark.parametrize("shape", [
    # (N, C, H, W, K, R, S, stride, pad)
    (2, 3, 16, 16, 8, 3, 3, 1, 1),
    (2, 8, 15, 15, 16, 3, 3, 2, 0),      # stride 2, odd size
    (2, 16, 8, 8, 32, 1, 1, 1, 0),       # pointwise
    (2, 16, 48, 48, 24, 1, 1, 1, 0),     # pointwise, P>=4096 (GEMM route)
    (1, 4, 12, 12, 6, 1, 7, 1, 3),       # asymmetric 1x7 (Inception)
    (1, 4, 12, 12, 6, 7, 1, 1, 3),       # asymmetric 7x1 — pad clamps
    (2, 3, 31, 31, 8, 3, 3, 2, 0),       # Inception stem-ish
])
def test_conv2d_fwd_matches_torch(shape):
    from tfmesos_amd import ops
    N, C, H, W, K, R, S, st, pd = shape
    pads = (pd if R > 1 else 0, pd if S > 1 else 0)
    torch.manual_seed(hash(shape) % 2**31)
    x = torch.randn(N, C, H, W)
    w = torch.randn(K, C, R, S) * 0.2
    b = torch.randn(K)
    xb, wb = bf(x), bf(w)
    y = ops.conv2d(xb, wb, b.to(DEV), stride=st, padding=pads)
    ref = torch.nn.functional.conv2d(xb.float().cpu(), wb.float().cpu(), b,
                                     stride=st, padding=pads)
    err = (y.float().cpu() - ref).abs().max()
    tol = 0.03 * ref.abs().max() + 0.06
    assert err < tol, (err, tol)


def test_conv2d_bwd_matches_torch():
    from tfmesos_amd import ops
    torch.manual_seed(77)
    N, C, H, W, K, R, S, st, pd = 2, 8, 14, 14, 16, 3, 3, 2, 1
    x = bf(torch.randn(N, C, H, W)).requires_grad_(True)
    w = bf(torch.randn(K, C, R, S) * 0.2).requires_grad_(True)
    y = ops.conv2d(x, w, None, stride=st, padding=pd)
    g = torch.randn_like(y)
    y.backward(g)

    xf = x.detach().float().cpu().requires_grad_(True)
    wf = w.detach().float().cpu().requires_grad_(True)
    ref = torch.nn.functional.conv2d(xf, wf, stride=st, padding=pd)
    ref.backward(g.float().cpu())

    dxe = (x.grad.float().cpu() - xf.grad).abs().max()
    dwe = (w.grad.float().cpu() - wf.grad).abs().max()
    assert dxe < 0.03 * xf.grad.abs().max() + 0.06, dxe
    assert dwe < 0.03 * wf.grad.abs().max() + 0.3, dwe


def test_conv2d_train_step_decreases_loss():
    from tfmesos_amd import ops
    torch.manual_seed(78)
    x = bf(torch.randn(4, 3, 16, 16))
    tgt = torch.randint(0, 4, (4,), device=DEV)
    w1 = bf(torch.randn(8, 3, 3, 3) * 0.2).requires_grad_(True)
    wf = bf(torch.randn(4, 8 * 8 * 8) * 0.05).requires_grad_(True)

    def loss_fn():
        h = torch.relu(ops.conv2d(x, w1, None, stride=2, padding=1))
        logits = h.reshape(4, -1).float() @ wf.float().t()
        return torch.nn.functional.cross_entropy(logits, tgt)

    l0 = None
    for _ in range(12):
        loss = loss_fn()
        if l0 is None:
            l0 = float(loss)
        loss.backward()
        with torch.no_grad():
            for p in (w1, wf):
                p -= 0.1 * p.grad
                p.grad = None
    assert float(loss_fn()) < l0


# --------------------------------------------------------------------- bn

@pytest.mark.parametrize("relu", [False, True])
@pytest.mark.parametrize("shape", [(4, 16, 14, 14), (2, 64, 35, 35),
                                   (8, 32, 7, 9)])
def test_batch_norm_act_fwd_bwd(shape, relu):
    from tfmesos_amd import ops
    torch.manual_seed(hash((shape, relu)) % 2**31)
    N, C, H, W = shape
    x = bf(torch.randn(N, C, H, W) * 2 + 0.5).requires_grad_(True)
    g = torch.rand(C, device=DEV) + 0.5
    b = torch.randn(C, device=DEV) * 0.2
    g = g.requires_grad_(True)
    b = b.requires_grad_(True)
    y = ops.batch_norm_act(x, g, b, eps=1e-3, relu=relu)
    dy = torch.randn_like(y)
    y.backward(dy)

    # fp32 reference for the forward
    xf = x.detach().float().cpu()
    gf = g.detach().float().cpu()
    bff = b.detach().float().cpu()
    ref = torch.nn.functional.batch_norm(
        xf, None, None, gf, bff, training=True, eps=1e-3)
    if relu:
        ref = torch.relu(ref)
    assert (y.float().cpu() - ref).abs().max() < 0.05

    # backward reference computed with THE SAME relu mask the kernel's
    # bit-exact-with-forward recompute produces (at exact boundaries the
    # bf16 subgradient choice legitimately differs from fp32): validates
    # the reduction/affine math itself
    mask = (y.float().cpu() > 0) if relu else torch.ones_like(ref)
    dyf = dy.float().cpu() * mask
    nhw = xf.numel() / xf.shape[1]
    mean = xf.mean(dim=(0, 2, 3), keepdim=True)
    var = xf.var(dim=(0, 2, 3), unbiased=False, keepdim=True)
    xhat = (xf - mean) * torch.rsqrt(var + 1e-3)
    s1 = dyf.sum(dim=(0, 2, 3))
    s2 = (dyf * xhat).sum(dim=(0, 2, 3))
    dx_ref = (gf * torch.rsqrt(var + 1e-3).flatten()).view(1, -1, 1, 1) * (
        dyf - (s1 / nhw).view(1, -1, 1, 1)
        - xhat * (s2 / nhw).view(1, -1, 1, 1))
    assert (x.grad.float().cpu() - dx_ref).abs().max() < 0.05
    assert (g.grad.float().cpu() - s2).abs().max() < \
        0.02 * s2.abs().max() + 0.5
    assert (b.grad.float().cpu() - s1).abs().max() < \
        0.02 * s1.abs().max() + 0.5


def test_mlp_head_fused_matches_composed():
    """One-kernel classifier head == gemm + softmax_xent + masked dh."""
    from tfmesos_amd import ops
    torch.manual_seed(50)
    B, H, C = 100, 100, 10
    h = torch.relu(bf(torch.randn(B, H)))
    w = bf(torch.randn(H, C) * 0.1)
    b = bf(torch.randn(C) * 0.1)
    y = torch.randint(0, C, (B,), device=DEV)
    dw2 = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2 = torch.zeros(C, device=DEV, dtype=torch.float32)
    loss, dl, dh = ops.mlp_head_fused(h, w, b, y, dw2=dw2, db2=db2)
    logits = ops.gemm_bias_act(h, w, b)
    loss2, dl2 = ops.softmax_xent_fused(logits, y)
    dh2 = ops.gemm_bias_act(dl2, w, trans_b=True, act="relu_bwd", aux=h)
    dw2r = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2r = torch.zeros(C, device=DEV, dtype=torch.float32)
    ops.gemm_bias_act(h, dl2, trans_a=True, out=dw2r, colsum_out=db2r)
    assert abs(float(loss) - float(loss2)) < 1e-2
    assert (dl.float() - dl2.float()).abs().max() < 1e-3
    assert (dh.float() - dh2.float()).abs().max() < 2e-3
    assert (dw2 - dw2r).abs().max() < 2e-3 * dw2r.abs().max() + 1e-4
    assert (db2 - db2r).abs().max() < 2e-3 * db2r.abs().max() + 1e-4


def test_avg_pool3x3_fwd_bwd():
    from tfmesos_amd import ops
    torch.manual_seed(60)
    for shape in [(2, 16, 9, 11), (1, 7, 8, 8)]:   # vec + scalar-C paths
        x = bf(torch.randn(*shape)).requires_grad_(True)
        y = ops.avg_pool3x3(x)
        ref = torch.nn.functional.avg_pool2d(
            x.detach().float().cpu(), 3, stride=1, padding=1)
        assert (y.float().cpu() - ref).abs().max() < 0.02, shape
        dy = bf(torch.randn(*shape))
        y.backward(dy)
        xf = x.detach().float().cpu().requires_grad_(True)
        r2 = torch.nn.functional.avg_pool2d(xf, 3, stride=1, padding=1)
        r2.backward(dy.float().cpu())
        assert (x.grad.float().cpu() - xf.grad).abs().max() < 0.02, shape


def test_gemm_colsum_epilogue_bf16_out():
    """bf16 grad buffers (half the xGMI reduce bytes): dW and the fused
    colsum both land in bf16."""
    from tfmesos_amd import ops
    torch.manual_seed(61)
    x, dy = bf(torch.randn(100, 784)), bf(torch.randn(100, 100))
    out = torch.zeros(784, 100, device=DEV, dtype=torch.bfloat16)
    cs = torch.zeros(100, device=DEV, dtype=torch.bfloat16)
    ops.gemm_bias_act(x, dy, trans_a=True, out=out, colsum_out=cs)
    ref_w = x.float().t() @ dy.float()
    ref_b = dy.float().sum(0)
    assert (out.float() - ref_w).abs().max() < 0.02 * ref_w.abs().max() + 0.3
    assert (cs.float() - ref_b).abs().max() < 0.02 * ref_b.abs().max() + 0.1


def test_maxpool3x3s2_fwd_bwd():
    from tfmesos_amd import ops
    torch.manual_seed(62)
    for shape in [(2, 16, 15, 15), (1, 7, 9, 11)]:
        x = bf(torch.randn(*shape)).requires_grad_(True)
        y = ops.max_pool3x3s2(x)
        ref = torch.nn.functional.max_pool2d(
            x.detach().float().cpu(), 3, stride=2)
        assert torch.equal(y.float().cpu(), ref.to(torch.bfloat16).float()
                           .to(torch.float32)) or \
            (y.float().cpu() - ref).abs().max() < 1e-6, shape
        dy = bf(torch.randn(*y.shape))
        y.backward(dy)
        xf = x.detach().float().cpu().requires_grad_(True)
        r2 = torch.nn.functional.max_pool2d(xf, 3, stride=2)
        r2.backward(dy.float().cpu())
        dxe = (x.grad.float().cpu() - xf.grad).abs().max()
        assert dxe < 0.01 * xf.grad.abs().max() + 0.01, (shape, dxe)


def test_bn_bwd_accepts_channel_narrow_dy():
    """torch.cat backward hands each branch a channel-narrow VIEW of the
    block gradient; the BN backward reads it in place (no copy)."""
    from tfmesos_amd import ops
    torch.manual_seed(70)
    N, C, H, W, CT = 2, 32, 9, 9, 96
    x = bf(torch.randn(N, C, H, W)).requires_grad_(True)
    g = (torch.rand(C, device=DEV) + 0.5).requires_grad_(True)
    b = (torch.randn(C, device=DEV) * 0.2).requires_grad_(True)
    y = ops.batch_norm_act(x, g, b, relu=True)
    big = bf(torch.randn(N, CT, H, W)).contiguous(
        memory_format=torch.channels_last)
    dy_view = big.narrow(1, 32, C)          # strided channel slice
    assert not dy_view.is_contiguous(memory_format=torch.channels_last)
    y.backward(dy_view)
    gx1 = x.grad.clone()
    x.grad = None

    y2 = ops.batch_norm_act(x, g, b, relu=True)
    y2.backward(dy_view.contiguous(memory_format=torch.channels_last))
    assert torch.equal(gx1, x.grad)


def test_bn_strided_out_matches_dense():
    """bn_fwd with a channel-narrow out view (the Inception fused-cat
    path) writes exactly what the dense apply produces."""
    from tfmesos_amd import ops
    torch.manual_seed(11)
    x = bf(torch.randn(4, 48, 9, 9)).contiguous(
        memory_format=torch.channels_last)
    g = bf(torch.randn(48).abs() + 0.4)
    b = bf(torch.randn(48))
    dense = ops.batch_norm_act(x, g, b, relu=True)
    buf = torch.zeros(4, 128, 9, 9, device=DEV, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    out = ops.batch_norm_act(x, g, b, relu=True, out=buf.narrow(1, 40, 48))
    torch.cuda.synchronize()
    assert torch.equal(out, dense)
    assert torch.equal(buf.narrow(1, 40, 48), dense)
    # untouched slices stay zero
    assert buf.narrow(1, 0, 40).abs().sum().item() == 0.0


def test_fused_cat_block_matches_cat():
    """InceptionA with the zero-copy fused concat is bit-identical
    (forward AND gradients) to the same block running plain torch.cat."""
    import tfmesos_amd.models.inception as inc

    def run(block, x, fuse):
        orig = inc._fused_cat
        if not fuse:
            inc._fused_cat = \
                lambda xx, specs: torch.cat([b(None) for _, b in specs], 1)
        try:
            xx = x.detach().clone().requires_grad_(True)
            y = block(xx)
            loss = y.float().square().mean()
            for p in block.parameters():
                p.grad = None
            loss.backward()
            return (y.detach().clone(), xx.grad.detach().clone(),
                    [p.grad.detach().clone() if p.grad is not None else
                     getattr(p, "_tfa_raw_grad", torch.zeros(1)).clone()
                     for p in block.parameters()])
        finally:
            inc._fused_cat = orig

    torch.manual_seed(5)
    block = inc.InceptionA(192, 32).to(DEV, torch.bfloat16)
    x = bf(torch.randn(2, 192, 17, 17)).contiguous(
        memory_format=torch.channels_last)
    y1, dx1, g1 = run(block, x, fuse=True)
    y2, dx2, g2 = run(block, x, fuse=False)
    assert torch.equal(y1, y2)
    assert torch.equal(dx1, dx2)
    # conv weight grads accumulate with z-sliced fp32 atomics, so the
    # reduction order (and the final bf16 rounding) varies between
    # runs — compare within a few ulps instead of bitwise
    for a, b in zip(g1, g2):
        assert torch.allclose(a.float(), b.float(), rtol=5e-2, atol=1e-6)


def test_mlp_fwd_head_fused_matches_composed():
    """Two-kernel fwd+head (split-K stripes consumed by the MFMA head)
    == the composed pipeline, including the in-kernel dW2/db2 grads."""
    from tfmesos_amd import ops
    torch.manual_seed(51)
    B, K, H, C = 100, 784, 100, 10
    x = bf(torch.rand(B, K))
    w1 = bf(torch.randn(K, H) * 0.03)
    b1 = bf(torch.randn(H) * 0.1)
    w2 = bf(torch.randn(H, C) * 0.1)
    b2 = bf(torch.randn(C) * 0.1)
    y = torch.randint(0, C, (B,), device=DEV)
    dw2 = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2 = torch.zeros(C, device=DEV, dtype=torch.float32)
    loss, dh = ops.mlp_fwd_head_fused(x, w1, b1, w2, b2, y,
                                      dw2=dw2, db2=db2)

    h = ops.gemm_bias_act(x, w1, b1, act="relu")
    logits = ops.gemm_bias_act(h, w2, b2)
    loss2, dl2 = ops.softmax_xent_fused(logits, y)
    dh2 = ops.gemm_bias_act(dl2, w2, trans_b=True, act="relu_bwd", aux=h)
    dw2r = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2r = torch.zeros(C, device=DEV, dtype=torch.float32)
    ops.gemm_bias_act(h, dl2, trans_a=True, out=dw2r, colsum_out=db2r)

    assert abs(float(loss) - float(loss2)) < 1e-2, (float(loss),
                                                    float(loss2))
    assert (dh.float() - dh2.float()).abs().max() < 3e-3
    assert (dw2 - dw2r).abs().max() < 3e-3 * dw2r.abs().max() + 1e-4
    assert (db2 - db2r).abs().max() < 3e-3 * db2r.abs().max() + 1e-4


def test_mlp_head_fused_odd_shapes():
    """MFMA head path with non-multiple-of-32 B/H and small C (padding
    rows/cols must stay zeroed through every phase)."""
    from tfmesos_amd import ops
    torch.manual_seed(52)
    B, H, C = 37, 60, 7
    h = torch.relu(bf(torch.randn(B, H)))
    w = bf(torch.randn(H, C) * 0.1)
    b = bf(torch.randn(C) * 0.1)
    y = torch.randint(0, C, (B,), device=DEV)
    dw2 = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2 = torch.zeros(C, device=DEV, dtype=torch.float32)
    loss, dl, dh = ops.mlp_head_fused(h, w, b, y, dw2=dw2, db2=db2)

    logits = ops.gemm_bias_act(h, w, b)
    loss2, dl2 = ops.softmax_xent_fused(logits, y)
    dh2 = ops.gemm_bias_act(dl2, w, trans_b=True, act="relu_bwd", aux=h)
    dw2r = torch.zeros(H, C, device=DEV, dtype=torch.float32)
    db2r = torch.zeros(C, device=DEV, dtype=torch.float32)
    ops.gemm_bias_act(h, dl2, trans_a=True, out=dw2r, colsum_out=db2r)
    assert abs(float(loss) - float(loss2)) < 1e-2
    assert (dl.float() - dl2.float()).abs().max() < 1e-3
    assert (dh.float() - dh2.float()).abs().max() < 2e-3
    assert (dw2 - dw2r).abs().max() < 3e-3 * dw2r.abs().max() + 1e-4
    assert (db2 - db2r).abs().max() < 3e-3 * db2r.abs().max() + 1e-4


@pytest.mark.gpu
def test_bn_group_matches_per_branch():
    """Grouped BN (one launch triple for all block branches) must match
    per-branch batch_norm_act exactly in both directions."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(5)
    dev = "cuda:0"
    N, H, W = 4, 9, 9
    Cs = [64, 96, 32]
    ctot = sum(Cs)
    xs = [torch.randn(N, c, H, W, device=dev, dtype=torch.bfloat16)
          .contiguous(memory_format=torch.channels_last)
          .requires_grad_(True) for c in Cs]
    ws = [torch.rand(c, device=dev, dtype=torch.bfloat16)
          .requires_grad_(True) for c in Cs]
    bs = [torch.randn(c, device=dev, dtype=torch.bfloat16)
          .requires_grad_(True) for c in Cs]

    buf = torch.empty(N, ctot, H, W, device=dev, dtype=torch.bfloat16,
                      memory_format=torch.channels_last)
    y = ops.bn_group_apply(buf.narrow(1, 0, ctot), xs, ws, bs,
                           eps=1e-3, relu=True)
    dy = torch.randn_like(y)
    y.backward(dy)

    off = 0
    for i, c in enumerate(Cs):
        x2 = xs[i].detach().clone().requires_grad_(True)
        w2 = ws[i].detach().clone().requires_grad_(True)
        b2 = bs[i].detach().clone().requires_grad_(True)
        y2 = ops.batch_norm_act(x2, w2, b2, eps=1e-3, relu=True)
        y2.backward(dy.narrow(1, off, c).contiguous(
            memory_format=torch.channels_last))
        assert torch.equal(y.narrow(1, off, c), y2), "fwd branch %d" % i
        assert torch.allclose(xs[i].grad.float(), x2.grad.float(),
                              atol=1e-3), "dx branch %d" % i
        assert torch.allclose(ws[i].grad.float(), w2.grad.float(),
                              rtol=1e-2, atol=1e-2), "dgamma branch %d" % i
        assert torch.allclose(bs[i].grad.float(), b2.grad.float(),
                              rtol=1e-2, atol=1e-2), "dbeta branch %d" % i
        off += c


@pytest.mark.gpu
def test_bn_group_multi_matches_per_branch():
    """Grouped BN with per-branch DENSE outputs (parallel inner-stage
    branches) must match per-branch batch_norm_act exactly."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(6)
    dev = "cuda:0"
    N, H, W = 3, 7, 7
    Cs = [48, 64]
    xs = [torch.randn(N, c, H, W, device=dev, dtype=torch.bfloat16)
          .contiguous(memory_format=torch.channels_last)
          .requires_grad_(True) for c in Cs]
    ws = [torch.rand(c, device=dev, dtype=torch.bfloat16)
          .requires_grad_(True) for c in Cs]
    bs = [torch.randn(c, device=dev, dtype=torch.bfloat16)
          .requires_grad_(True) for c in Cs]
    ys = ops.bn_group_multi(xs, ws, bs, eps=1e-3, relu=True)
    dys = [torch.randn_like(y) for y in ys]
    torch.autograd.backward(ys, dys)
    for i, c in enumerate(Cs):
        x2 = xs[i].detach().clone().requires_grad_(True)
        w2 = ws[i].detach().clone().requires_grad_(True)
        b2 = bs[i].detach().clone().requires_grad_(True)
        y2 = ops.batch_norm_act(x2, w2, b2, eps=1e-3, relu=True)
        y2.backward(dys[i])
        assert torch.equal(ys[i], y2), "fwd branch %d" % i
        assert torch.allclose(xs[i].grad.float(), x2.grad.float(),
                              atol=1e-3), "dx branch %d" % i
        assert torch.allclose(ws[i].grad.float(), w2.grad.float(),
                              rtol=1e-2, atol=1e-2), i
        assert torch.allclose(bs[i].grad.float(), b2.grad.float(),
                              rtol=1e-2, atol=1e-2), i


@pytest.mark.gpu
def test_gemm_fused_sub_epilogue():
    """act='sub' (out = A@B - aux, the NMF residual) vs fp32 torch."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(11)
    a = torch.randn(257, 300, device="cuda:0", dtype=torch.bfloat16)
    b = torch.randn(300, 190, device="cuda:0", dtype=torch.bfloat16)
    x = torch.randn(257, 190, device="cuda:0", dtype=torch.bfloat16)
    got = ops.gemm_bias_act(a, b, act="sub", aux=x)
    ref = a.float() @ b.float() - x.float()
    err = (got.float() - ref).abs().max() / ref.abs().max()
    assert float(err) < 0.02, float(err)


@pytest.mark.gpu
def test_fused_sgd_neg_decay():
    """neg_decay (g += c*min(p,0)) vs the explicit clamp+add form."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(12)
    p1 = torch.randn(5000, device="cuda:0")
    p2 = p1.clone()
    g = torch.randn(5000, device="cuda:0")
    ops.fused_sgd(p1, g, lr=0.1, grad_scale=0.5, neg_decay=0.3)
    g2 = g * 0.5 + 0.3 * torch.clamp(p2, max=0.0)
    p2 -= 0.1 * g2
    assert torch.allclose(p1, p2, atol=1e-6)


@pytest.mark.gpu
def test_gemm_sgd_pair_simultaneous_semantics():
    """Fused GEMM->SGD pair vs explicit fp32 reference: both gradient
    GEMMs must read PRE-update factors (simultaneous update)."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(13)
    dev = "cuda:0"
    n, r, k = 300, 64, 300
    W = torch.randn(n, r, device=dev) * 0.1
    H = torch.randn(r, k, device=dev) * 0.1
    Wb = W.to(torch.bfloat16)
    Hb = H.to(torch.bfloat16)
    E = torch.randn(n, k, device=dev, dtype=torch.bfloat16) * 0.1
    W0, H0 = W.clone(), H.clone()
    lr, scale, c = 0.05, 0.25, 0.1
    ops.gemm_sgd_pair((E, Hb, W, Wb, False, True),
                      (Wb, E, H, Hb, True, False),
                      lr=lr, grad_scale=scale, neg_decay=c)
    # reference: both grads from PRE-update values
    Wb0 = W0.to(torch.bfloat16).float()
    gW = E.float() @ Hb.float().t() * scale + c * torch.clamp(W0, max=0.0)
    gH = Wb0.t() @ E.float() * scale + c * torch.clamp(H0, max=0.0)
    refW = W0 - lr * gW
    refH = H0 - lr * gH
    assert torch.allclose(W, refW, atol=2e-2, rtol=1e-2), \
        float((W - refW).abs().max())
    assert torch.allclose(H, refH, atol=2e-2, rtol=1e-2), \
        float((H - refH).abs().max())
    assert torch.allclose(Wb.float(), W.to(torch.bfloat16).float())
    assert torch.allclose(Hb.float(), H.to(torch.bfloat16).float())


def test_conv2d_bwd_channel_narrow_dy_view():
    """Concat backward hands each branch a channel-NARROW view of the
    joint grad buffer (ldy > K). The transpose-read bwd kernels take
    the row stride (cs.LDY) straight from the view — no contiguous
    copy — so a strided dy must produce the same dX/dW as its
    contiguous clone."""
    from tfmesos_amd import ops
    torch.manual_seed(99)
    N, C, H, W, K, R, S = 2, 16, 13, 13, 24, 3, 3
    x = bf(torch.randn(N, C, H, W)).requires_grad_(True)
    w = bf(torch.randn(K, C, R, S) * 0.2).requires_grad_(True)
    y = ops.conv2d(x, w, None, stride=1, padding=1)

    big = bf(torch.randn(N, K + 40, H, W))      # concat-like buffer
    dy_view = big.narrow(1, 8, K)               # strided channel slice
    y.backward(dy_view)
    dx_v, dw_v = x.grad.clone(), w.grad.clone()

    x.grad = None
    w.grad = None
    y2 = ops.conv2d(x, w, None, stride=1, padding=1)
    y2.backward(dy_view.contiguous(memory_format=torch.channels_last))
    assert torch.equal(dx_v, x.grad), "dX differs for narrow-dy view"
    assert torch.equal(dw_v, w.grad), "dW differs for narrow-dy view"
