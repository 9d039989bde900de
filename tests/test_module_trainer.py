"""ModuleReplicaTrainer (autograd models on the PS flat store) and the
Inception-v3 model — CPU tests; GPU counterparts in test_inception_gpu."""

import pytest
import torch
import torch.nn as nn

from tfmesos_amd.models.inception import BasicConv2d, InceptionV3
from tfmesos_amd.ps.module_trainer import ModuleReplicaTrainer


class TinyCNN(nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(0)
        self.c1 = BasicConv2d(3, 8, kernel_size=3, stride=2, padding=1)
        self.c2 = BasicConv2d(8, 16, kernel_size=3, stride=2, padding=1)
        self.fc = nn.Linear(16 * 4 * 4, 5)

    def forward(self, x):
        return self.fc(self.c2(self.c1(x)).flatten(1)).float()


def test_module_trainer_zero_copy_views():
    m = TinyCNN()
    tr = ModuleReplicaTrainer(m, hparams={"lr": 0.05})
    # parameters alias the bf16 shadow; step() batch-copies autograd
    # grads into flat grad views aligned with the store layout
    w = dict(m.named_parameters())["c1.conv.weight"]
    assert w.data.data_ptr() >= tr.store.flat_bf16.data_ptr()
    names = [n for n, _ in m.named_parameters()]
    gv = tr._gviews[names.index("c1.conv.weight")]
    assert gv.data_ptr() >= tr.t.flat_grad.data_ptr()
    assert gv.shape == w.shape
    assert w.dtype == torch.bfloat16


def test_module_trainer_trains():
    m = TinyCNN()
    tr = ModuleReplicaTrainer(m, hparams={"lr": 0.05})
    x = torch.rand(8, 3, 16, 16, dtype=torch.bfloat16)
    y = torch.randint(0, 5, (8,))
    losses = []
    for _ in range(15):
        tr.zero_grad()
        loss = torch.nn.functional.cross_entropy(m(x), y)
        loss.backward()
        tr.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.8, losses


def test_module_trainer_checkpoint_roundtrip(tmp_path):
    m = TinyCNN()
    tr = ModuleReplicaTrainer(m, hparams={"lr": 0.05})
    x = torch.rand(4, 3, 16, 16, dtype=torch.bfloat16)
    y = torch.randint(0, 5, (4,))
    for _ in range(3):
        tr.zero_grad()
        torch.nn.functional.cross_entropy(m(x), y).backward()
        tr.step()
    path = str(tmp_path / "ck.pt")
    tr.save(path)
    flat = tr.store.flat.clone()

    m2 = TinyCNN()
    tr2 = ModuleReplicaTrainer(m2, hparams={"lr": 0.05})
    tr2.load(path)
    assert torch.equal(tr2.store.flat, flat)
    assert tr2.store.global_step == 3


def test_inception_v3_parameter_inventory():
    m = InceptionV3(num_classes=10)
    n = sum(p.numel() for p in m.parameters())
    # standard Inception-v3 trunk is ~21.8M params (+ fc head)
    assert 20_000_000 < n < 28_000_000, n
    names = [k for k, _ in m.named_parameters()]
    assert "fc_w" in names and any("b7d_5" in k for k in names)


@pytest.mark.timeout(600)
def test_inception_v3_cpu_forward():
    m = InceptionV3(num_classes=10)
    x = torch.rand(1, 3, 299, 299)
    y = m(x)
    assert y.shape == (1, 10)
    assert torch.isfinite(y).all()
