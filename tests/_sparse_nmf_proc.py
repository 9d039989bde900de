"""Helper: one rank of a 3-process sparse NMF run; the worker prints
first/last minibatch losses."""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tfmesos_amd.models.nmf import SparseNMF  # noqa: E402


def main():
    steps = int(sys.argv[1])
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo", rank=rank, world_size=world)
    wl = SparseNMF(rank, world, n=300, factor_rank=32, batch=64, lr=0.2)
    for _ in range(steps):
        wl.one_step()
    wl.finalize()
    if not wl.is_ps and wl.losses:
        k = max(1, len(wl.losses) // 4)
        first = sum(wl.losses[:k]) / k
        last = sum(wl.losses[-k:]) / k
        print("LOSSES %g %g" % (first, last))
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
