"""MLP fwd/bwd: our explicit backward must match torch autograd (CPU)."""

import torch

from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
from tfmesos_amd.ps.store import PStore


def test_fwd_bwd_matches_autograd():
    model = MnistMLP(hidden_units=16, image_pixels=20, classes=5)
    params = dict(model.init_params())
    x, y = synthetic_batch(8, 20, 5)

    # autograd reference
    pt = {k: v.clone().requires_grad_(True) for k, v in params.items()}
    h = torch.relu(x @ pt["hid_w"] + pt["hid_b"])
    logits = h @ pt["sm_w"] + pt["sm_b"]
    loss_ref = torch.nn.functional.cross_entropy(logits, y)
    loss_ref.backward()

    # our explicit path
    ps = PStore()
    ps.init_params(list(params.items()))
    grads = {k: torch.zeros(*ps.shapes[k]) for k in ps.names}
    loss = model.fwd_bwd(lambda n: ps.view(n), x, y, lambda n: grads[n])

    assert torch.allclose(loss, loss_ref, atol=1e-5)
    for k in params:
        assert torch.allclose(grads[k], pt[k].grad, atol=1e-4), k


def test_training_reduces_loss():
    model = MnistMLP()
    ps = PStore()
    ps.init_params(model.init_params(), optimizer="sgd", lr=0.1)
    x, y = synthetic_batch(100)
    grads = {k: torch.zeros(*ps.shapes[k]) for k in ps.names}
    loss0 = model.loss_only(lambda n: ps.view(n), x, y)
    for _ in range(200):
        model.fwd_bwd(lambda n: ps.view(n), x, y, lambda n: grads[n])
        ps.push_apply(grads)
    loss1 = model.loss_only(lambda n: ps.view(n), x, y)
    assert loss1 < 0.2, (loss0, loss1)


def test_cpu_head_ops_match_autograd():
    """CPU references of the fused head ops (mlp_head_fused with dw2/
    db2, mlp_fwd_head_fused) match plain autograd — these are the
    oracles the GPU numerics tests compare the HIP kernels against."""
    import torch

    from tfmesos_amd import ops

    torch.manual_seed(7)
    B, K, H, C = 50, 64, 32, 10
    x = torch.rand(B, K)
    w1 = torch.randn(K, H) * 0.1
    b1 = torch.randn(H) * 0.1
    w2 = torch.randn(H, C) * 0.1
    b2 = torch.randn(C) * 0.1
    y = torch.randint(0, C, (B,))

    xr = x.clone()
    w1r = w1.clone().requires_grad_(True)
    b1r = b1.clone().requires_grad_(True)
    w2r = w2.clone().requires_grad_(True)
    b2r = b2.clone().requires_grad_(True)
    z = xr @ w1r + b1r
    z.retain_grad()
    h = torch.relu(z)
    logits = h @ w2r + b2r
    loss_ref = torch.nn.functional.cross_entropy(logits, y)
    loss_ref.backward()

    dw2 = torch.zeros(H, C)
    db2 = torch.zeros(C)
    loss1, dl1, dh1 = ops.mlp_head_fused(h.detach(), w2, b2, y,
                                         dw2=dw2, db2=db2)
    assert torch.allclose(loss1, loss_ref, atol=1e-5)
    assert torch.allclose(dh1, z.grad, atol=1e-4)
    assert torch.allclose(dw2, w2r.grad, atol=1e-4)
    assert torch.allclose(db2, b2r.grad, atol=1e-4)

    dw2b = torch.zeros(H, C)
    db2b = torch.zeros(C)
    loss2, dh2 = ops.mlp_fwd_head_fused(x, w1, b1, w2, b2, y,
                                        dw2=dw2b, db2=db2b)
    assert torch.allclose(loss2, loss_ref, atol=1e-5)
    assert torch.allclose(dh2, z.grad, atol=1e-4)
    assert torch.allclose(dw2b, w2r.grad, atol=1e-4)
    assert torch.allclose(db2b, b2r.grad, atol=1e-4)
