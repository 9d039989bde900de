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
