"""Sparse-embedding PS (row pull / scatter-add push over point-to-point
dist channels) — multi-process gloo tests on CPU."""

import os
import subprocess
import sys

import pytest
import torch

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)


def test_embedding_table_push_pull_semantics():
    from tfmesos_amd.ps.sparse import EmbeddingTable
    t = EmbeddingTable("w", rows=20, dim=8, lr=0.5, seed=1)
    ids = torch.tensor([3, 7, 3])
    before = t.master.clone()
    rows = t.pull(ids)
    assert torch.allclose(rows.float(), before[ids].to(torch.bfloat16).float())
    g = torch.randn(3, 8)
    t.push(ids, g)
    # duplicates accumulate (TF sparse-apply semantics)
    want = before.clone()
    want.index_add_(0, ids, -0.5 * g)
    assert torch.allclose(t.master, want, atol=1e-2)
    # shadow refreshed for touched rows
    assert torch.allclose(t.pull(ids).float(),
                          want[ids].to(torch.bfloat16).float(), atol=1e-2)


def _spawn_bench_nmf(world, steps=8):
    from tfmesos_amd.utils import free_port
    port = free_port()
    procs = []
    for rank in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "PYTHONPATH": REPO,
        })
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "bench.py"),
             "--workload", "nmf", "--steps", str(steps), "--warmup", "2",
             "--nmf-n", "300", "--nmf-rank", "32", "--nmf-batch", "64"],
            env=env, stdout=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        assert p.returncode == 0, out
        outs.append(out)
    return outs


@pytest.mark.timeout(240)
def test_sparse_nmf_two_ps_one_worker():
    outs = _spawn_bench_nmf(3)
    assert any("steps/s" in o for o in outs)


@pytest.mark.timeout(240)
def test_sparse_nmf_single_ps():
    outs = _spawn_bench_nmf(2)
    assert any("steps/s" in o for o in outs)


@pytest.mark.timeout(240)
def test_sparse_nmf_loss_decreases():
    """Run the worker loop in-process (world 3 via subprocesses for ps,
    this process as worker) is complex; instead assert via a dedicated
    subprocess run that records losses."""
    from tfmesos_amd.utils import free_port
    port = free_port()
    script = os.path.join(HERE, "_sparse_nmf_proc.py")
    procs = []
    for rank in range(3):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": "3",
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "PYTHONPATH": REPO,
        })
        procs.append(subprocess.Popen(
            [sys.executable, script, "20"], env=env, stdout=subprocess.PIPE,
            text=True))
    for p in procs:
        out, _ = p.communicate(timeout=180)
        assert p.returncode == 0, out
        if "LOSSES" in out:
            first, last = [float(x) for x in
                           out.split("LOSSES")[1].split()[:2]]
            assert last < first, out
