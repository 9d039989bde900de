"""Helper: one rank of a 2-process ModuleReplicaTrainer run (gloo CPU),
colocated PS. Rank 0 saves final masters."""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tfmesos_amd.models.inception import BasicConv2d  # noqa: E402
from tfmesos_amd.ps.module_trainer import ModuleReplicaTrainer  # noqa: E402


class TinyCNN(torch.nn.Module):
    def __init__(self):
        super().__init__()
        torch.manual_seed(0)
        self.c1 = BasicConv2d(3, 8, kernel_size=3, stride=2, padding=1)
        self.fc = torch.nn.Linear(8 * 8 * 8, 5)

    def forward(self, x):
        return self.fc(self.c1(x).flatten(1)).float()


def main():
    steps, out = int(sys.argv[1]), sys.argv[2]
    variant = sys.argv[3] if len(sys.argv) > 3 else "colocate"
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        dist.init_process_group("gloo", rank=rank, world_size=world)
    m = TinyCNN()
    tr = ModuleReplicaTrainer(m, optimizer="sgd", hparams={"lr": 0.05},
                              colocate_ps=variant.startswith("colocate"),
                              overlap="no-overlap" not in variant)
    torch.manual_seed(42)          # same batch on every rank
    x = torch.rand(8, 3, 16, 16, dtype=torch.bfloat16)
    y = torch.randint(0, 5, (8,))
    for _ in range(steps):
        tr.zero_grad()
        if tr.roles.is_worker:
            loss = torch.nn.functional.cross_entropy(m(x), y)
            loss.backward()
        tr.step()
    tr.t.sync_masters()
    if rank == 0:
        torch.save(tr.store.flat.clone(), out)
    if world > 1:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
