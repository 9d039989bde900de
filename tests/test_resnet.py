"""ResNet-18 on the generic kernel stack: the framework is not
hardwired to the three BASELINE workloads — any conv/BN/residual
network composes from the same ops and trains on the same PS machinery.
"""

import pytest
import torch

from tfmesos_amd.models.resnet import ResNet18
from tfmesos_amd.ops import softmax_xent_loss
from tfmesos_amd.ps.module_trainer import ModuleReplicaTrainer


def test_resnet18_cpu_train_step_decreases_loss():
    torch.manual_seed(0)
    model = ResNet18(num_classes=10, seed=3)
    trainer = ModuleReplicaTrainer(model, optimizer="sgd",
                                   hparams={"lr": 0.05}, device="cpu")
    x = torch.randn(4, 3, 64, 64)
    y = torch.randint(0, 10, (4,))
    losses = []
    for _ in range(3):
        trainer.zero_grad()
        loss = softmax_xent_loss(model(x), y)
        loss.backward()
        trainer.step()
        losses.append(float(loss.detach()))
    assert all(l == l for l in losses), losses          # finite
    assert losses[-1] < losses[0], losses               # learning


@pytest.mark.gpu
@pytest.mark.timeout(300)
def test_resnet18_gpu_train_step():
    dev = torch.device("cuda", 0)
    model = ResNet18(num_classes=100, seed=3)
    trainer = ModuleReplicaTrainer(model, optimizer="sgd",
                                   hparams={"lr": 0.05}, device=dev)
    x = torch.randn(8, 3, 224, 224, device=dev, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (8,), device=dev)
    losses = []
    for _ in range(4):
        trainer.zero_grad()
        loss = softmax_xent_loss(model(x).contiguous(), y)
        loss.backward()
        trainer.step()
        losses.append(float(loss.detach()))
    assert all(l == l for l in losses), losses
    assert losses[-1] < losses[0] + 0.2, losses
