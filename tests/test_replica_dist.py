"""Multi-process distributed PS tests over gloo on CPU (world_size 2/3).

Sync mode with one worker must produce EXACTLY the same masters as the
single-process reference (same seed, same batch, fp32) — the strongest
check that reduce/apply/broadcast plumbing is correct by construction.
"""

import os
import subprocess
import sys

import pytest
import torch

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)


def _spawn_world(world, mode, steps, prefix, n_ps=1, colocate=False):
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
        argv = [sys.executable, os.path.join(HERE, "_replica_proc.py"),
                mode, str(steps), prefix, str(n_ps)]
        if colocate:
            argv.append(colocate if isinstance(colocate, str) else "colocate")
        procs.append(subprocess.Popen(argv, env=env))
    for p in procs:
        assert p.wait(timeout=180) == 0


def _single_process_reference(steps):
    from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
    from tfmesos_amd.ps.store import PStore
    model = MnistMLP()
    store = PStore()
    store.init_params(model.init_params(), optimizer="sgd", lr=0.1)
    fg = torch.zeros_like(store.flat)

    def gv(n):
        s, c = store.offsets[n]
        return fg[s:s + c].view(store.shapes[n])

    x, y = synthetic_batch(50, seed=42)
    # workers see the bf16 broadcast shadow, so the reference computes
    # with bf16 param views too (masters stay fp32)
    for _ in range(steps):
        model.fwd_bwd(lambda n: store.view(n, bf16=True), x, y, gv)
        store.apply_flat(fg)
    return {n: store.view(n).clone() for n in store.names}


@pytest.mark.timeout(240)
def test_sync_world2_matches_single_process(tmp_path):
    prefix = str(tmp_path / "w2")
    _spawn_world(2, "sync", 5, prefix)
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-6), n


@pytest.mark.timeout(240)
def test_sync_world3_two_workers(tmp_path):
    """2 workers with identical batches: mean grad == single grad, so
    masters must again match the single-process reference."""
    prefix = str(tmp_path / "w3")
    _spawn_world(3, "sync", 5, prefix)
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-5), n


@pytest.mark.timeout(240)
def test_async_world2_trains(tmp_path):
    prefix = str(tmp_path / "wa")
    _spawn_world(2, "async", 5, prefix)
    step = torch.load(prefix + ".step", weights_only=True)
    assert step == 5
    got = torch.load(prefix + ".pt", weights_only=True)
    # async with 1 worker == sync with 1 worker == single-process ref
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-6), n


@pytest.mark.timeout(240)
def test_async_open_ended_uneven_workers(tmp_path):
    """Open-ended async (the reference's contract, README.rst:68-72):
    the server takes NO step count; workers run UNEVEN step counts and
    send stop sentinels. Global step must equal the total worker
    steps."""
    prefix = str(tmp_path / "wopen")
    _spawn_world(3, "async-open", 4, prefix)   # workers run 4 and 5 steps
    step = torch.load(prefix + ".step", weights_only=True)
    assert step == 4 + 5


@pytest.mark.timeout(240)
def test_empty_shards_small_model(tmp_path):
    """n_ps=3 over a tiny model (total <= 2*256 after alignment): one
    shard is EMPTY — collectives and applies must skip it instead of
    issuing zero-numel ops."""
    from tfmesos_amd.ps.replica import _shard_ranges
    shards = _shard_ranges(512, 3)
    assert any(hi == lo for lo, hi in shards), shards
    prefix = str(tmp_path / "wempty")
    _spawn_world(4, "sync-tiny", 5, prefix, n_ps=3)
    got = torch.load(prefix + ".pt", weights_only=True)
    assert all(torch.isfinite(v).all() for v in got.values())


@pytest.mark.timeout(240)
def test_sync_two_ps_shards_match_single_process(tmp_path):
    """2 PS shards + 1 worker: sharded reduce/apply/broadcast must equal
    the single-process reference exactly (multi-PS sharding, SURVEY.md
    §2c strategy 5)."""
    prefix = str(tmp_path / "w3ps2")
    _spawn_world(3, "sync", 5, prefix, n_ps=2)
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-6), n


@pytest.mark.timeout(240)
def test_async_two_ps_shards(tmp_path):
    prefix = str(tmp_path / "waps2")
    _spawn_world(3, "async", 5, prefix, n_ps=2)
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-6), n


@pytest.mark.timeout(240)
def test_sync_colocated_ps_matches_single_process(tmp_path):
    """colocate_ps: every rank is a worker AND rank 0 applies — the
    bench's 1-ps/N-worker-on-N-GPUs mapping. Identical batches => mean
    grad == single grad => masters match the reference exactly."""
    prefix = str(tmp_path / "wc")
    _spawn_world(2, "sync", 5, prefix, colocate=True)
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-5), n


@pytest.mark.timeout(240)
def test_sync_allreduce_mode_matches_single_process(tmp_path):
    """allreduce mode (one collective + replicated deterministic apply)
    must be bit-identical to the PS path and the single-process ref."""
    prefix = str(tmp_path / "war")
    _spawn_world(2, "sync", 5, prefix, colocate="allreduce")
    got = torch.load(prefix + ".pt", weights_only=True)
    want = _single_process_reference(5)
    for n in want:
        assert torch.allclose(got[n], want[n], atol=1e-5), n


@pytest.mark.timeout(240)
def test_sync_adam_world2_matches_single_process(tmp_path):
    """Adam optimizer state lives on the PS; the 2-rank run must match
    single-process exactly (same seed/batch)."""
    import torch as T

    from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
    from tfmesos_amd.ps.store import PStore

    prefix = str(tmp_path / "wadam")
    _spawn_world(2, "sync-adam", 4, prefix)
    got = T.load(prefix + ".pt", weights_only=True)

    model = MnistMLP()
    store = PStore()
    store.init_params(model.init_params(), optimizer="adam", lr=0.01)
    fg = T.zeros_like(store.flat)

    def gv(n):
        s, c = store.offsets[n]
        return fg[s:s + c].view(store.shapes[n])

    x, y = synthetic_batch(50, seed=42)
    for _ in range(4):
        model.fwd_bwd(lambda n: store.view(n, bf16=True), x, y, gv)
        store.apply_flat(fg)
    for n in store.names:
        assert T.allclose(got[n], store.view(n), atol=1e-6), n
