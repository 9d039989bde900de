import math

import pytest
import torch

from tfmesos_amd.ps.store import PStore


def _params():
    torch.manual_seed(0)
    return [("w", torch.randn(4, 3)), ("b", torch.zeros(3))]


def test_init_and_views():
    ps = PStore()
    ps.init_params(_params(), optimizer="sgd", lr=0.1)
    assert ps.view("w").shape == (4, 3)
    assert ps.view("b", bf16=True).dtype == torch.bfloat16
    pulled = ps.pull()
    assert set(pulled) == {"w", "b"}


def test_sgd_apply_matches_torch():
    torch.manual_seed(1)
    w0 = torch.randn(4, 3)
    ps = PStore()
    ps.init_params([("w", w0.clone())], optimizer="sgd", lr=0.5)
    g = torch.randn(4, 3)
    ps.push_apply({"w": g})
    assert torch.allclose(ps.view("w"), w0 - 0.5 * g, atol=1e-6)
    assert ps.global_step == 1
    # bf16 shadow refreshed
    assert torch.allclose(ps.view("w", bf16=True).float(), ps.view("w"),
                          atol=0.01)


def test_momentum_sgd():
    w0 = torch.ones(4)
    ps = PStore()
    ps.init_params([("w", w0.clone())], optimizer="sgd", lr=0.1, momentum=0.9)
    g = torch.ones(4)
    ps.push_apply({"w": g})
    ps.push_apply({"w": g})
    # v1 = g; w1 = w0 - .1*g ; v2 = .9*g + g = 1.9g ; w2 = w1 - .19
    assert torch.allclose(ps.view("w"), torch.full((4,), 1 - 0.1 - 0.19),
                          atol=1e-6)


def test_adam_apply_matches_torch_optim():
    torch.manual_seed(2)
    w0 = torch.randn(10)
    ref = w0.clone().requires_grad_(True)
    opt = torch.optim.Adam([ref], lr=0.01)
    ps = PStore()
    ps.init_params([("w", w0.clone())], optimizer="adam", lr=0.01)
    for i in range(3):
        g = torch.randn(10)
        ref.grad = g.clone()
        opt.step()
        ps.push_apply({"w": g})
    assert torch.allclose(ps.view("w"), ref.detach(), atol=1e-5)


def test_adagrad_apply_matches_torch_optim():
    torch.manual_seed(3)
    w0 = torch.randn(10)
    ref = w0.clone().requires_grad_(True)
    opt = torch.optim.Adagrad([ref], lr=0.05, initial_accumulator_value=0.1,
                              eps=1e-10)
    ps = PStore()
    ps.init_params([("w", w0.clone())], optimizer="adagrad", lr=0.05,
                   initial_accumulator=0.1)
    for i in range(3):
        g = torch.randn(10)
        ref.grad = g.clone()
        opt.step()
        ps.push_apply({"w": g})
    assert torch.allclose(ps.view("w"), ref.detach(), atol=1e-5)


def test_checkpoint_roundtrip(tmp_path):
    ps = PStore()
    ps.init_params(_params(), optimizer="adam", lr=0.01)
    ps.push_apply({"w": torch.randn(4, 3), "b": torch.randn(3)})
    p = str(tmp_path / "ck.pt")
    ps.save(p)
    ps2 = PStore()
    ps2.load(p)
    assert ps2.global_step == 1
    assert torch.equal(ps2.view("w"), ps.view("w"))
    assert torch.equal(ps2.state["exp_avg"], ps.state["exp_avg"])


def test_flat_layout_alignment():
    ps = PStore()
    ps.init_params(_params())
    for name in ps.names:
        start, _ = ps.offsets[name]
        assert start % 256 == 0
