"""ResNet-18 on the same MI355X kernel stack as Inception-v3.

Not one of the BASELINE workloads — it exists to show the framework
generalizes past them: the implicit-GEMM conv kernels (any R/S/stride,
channels-last), the fused train-mode BN(+relu), the pointwise-GEMM
routing and `ModuleReplicaTrainer`'s flat-store wiring are all generic,
so a standard residual network is just composition (reference analogue:
any TF program could be handed to tfmesos' launcher — README.rst:47-66).

Channels-last bf16 end to end; CPU runs the ops' fp32 torch references.
"""

import torch
import torch.nn as nn

from tfmesos_amd.models.inception import BasicConv2d, BatchNorm2d, Conv2d


class BasicBlock(nn.Module):
    """conv3x3-BN-relu -> conv3x3-BN -> (+ identity / 1x1-s2 projection)
    -> relu. The second BN runs without the fused relu (the activation
    follows the residual add)."""

    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.c1 = BasicConv2d(cin, cout, kernel_size=3, stride=stride,
                              padding=1)
        self.c2 = Conv2d(cout, cout, kernel_size=3, padding=1)
        self.bn2 = BatchNorm2d(cout, relu=False)
        if stride != 1 or cin != cout:
            self.proj = Conv2d(cin, cout, kernel_size=1, stride=stride)
            self.bnp = BatchNorm2d(cout, relu=False)
        else:
            self.proj = None

    def forward(self, x):
        idn = x if self.proj is None else self.bnp(self.proj(x))
        y = self.bn2(self.c2(self.c1(x)))
        return torch.relu(y + idn)


class ResNet18(nn.Module):

    def __init__(self, num_classes=1000, seed=0):
        super().__init__()
        torch.manual_seed(seed)
        self.stem = BasicConv2d(3, 64, kernel_size=7, stride=2, padding=3)
        widths = (64, 128, 256, 512)
        blocks = []
        cin = 64
        for i, w in enumerate(widths):
            blocks.append(BasicBlock(cin, w, stride=1 if i == 0 else 2))
            blocks.append(BasicBlock(w, w))
            cin = w
        self.blocks = nn.ModuleList(blocks)
        self.fc_w = nn.Parameter(torch.randn(512, num_classes) * 0.02)
        self.fc_b = nn.Parameter(torch.zeros(num_classes))

    def forward(self, x):
        from tfmesos_amd import ops
        x = self.stem(x)
        # 3x3 s2 max pool, pad 1: the hand-written pool kernel is the
        # pad-0 Inception geometry, so this one runs the torch op
        x = torch.nn.functional.max_pool2d(x, 3, stride=2, padding=1)
        for b in self.blocks:
            x = b(x)
        x = x.mean((2, 3))                      # global average pool
        return ops.linear(x.contiguous(), self.fc_w, self.fc_b)
