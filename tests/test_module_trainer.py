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


def test_arena_layout_matches_store_offsets():
    """The grad arena's slice offsets must equal the PStore flat
    offsets (both 256-element aligned): the trainer's ONE bulk
    flat_grad[:na].copy_(arena) is only correct under that invariant
    (dense arena packing shifted 93/94 conv
    gradients)."""
    m = InceptionV3(num_classes=10)
    tr = ModuleReplicaTrainer(m, hparams={"lr": 0.01})
    assert tr._arena is not None
    assert m._arena_offsets, "arena must be wired on every device"
    for name, off in m._arena_offsets.items():
        assert off == tr.store.offsets[name][0], name
    # numerically: mark each conv's arena slice, run the bulk-copy step
    # path, and check every grad_view sees its own mark
    from tfmesos_amd.models.inception import Conv2d
    convs = [(n + ".weight", mod) for n, mod in m.named_modules()
             if isinstance(mod, Conv2d)]
    for i, (_, mod) in enumerate(convs):
        mod._dw_buf.fill_(float(i + 1))
    tr.t.flat_grad[:tr._arena.numel()].copy_(tr._arena)
    for i, (name, mod) in enumerate(convs):
        gv = tr.t.grad_view(name)
        assert torch.all(gv == float(i + 1)), name


def test_store_load_preserves_views(tmp_path):
    """Full-checkpoint load must copy INTO the existing flat buffers:
    trainers hold live views (bf16 broadcast source, nn.Module .data
    aliases) that a rebind would orphan."""
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
    want = tr.store.flat.clone()
    # train further, then restore — module params must SEE the restore
    for _ in range(2):
        tr.zero_grad()
        torch.nn.functional.cross_entropy(m(x), y).backward()
        tr.step()
    assert not torch.equal(tr.store.flat, want)
    flat_before = tr.store.flat
    bf16_before = tr.store.flat_bf16
    tr.load(path)
    assert tr.store.flat is flat_before, "load must not rebind flat"
    assert tr.store.flat_bf16 is bf16_before
    assert torch.equal(tr.store.flat, want)
    w = dict(m.named_parameters())["c1.conv.weight"]
    assert torch.equal(
        w.data.float().flatten(),
        tr.store.view("c1.conv.weight").to(torch.bfloat16).float().flatten())


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


@pytest.mark.timeout(240)
def test_module_trainer_world2_matches_single_process(tmp_path):
    """2-rank gloo run of the nn.Module adapter (bf16 grads reduced,
    colocated PS) vs the single-process run: identical batches => the
    mean grad equals the single grad => identical masters."""
    import os
    import subprocess
    import sys

    from tfmesos_amd.utils import free_port

    here = os.path.dirname(os.path.abspath(__file__))
    repo = os.path.dirname(here)
    out2 = str(tmp_path / "w2.pt")
    port = free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   PYTHONPATH=repo)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(here, "_module_trainer_proc.py"),
             "4", out2], env=env))
    for p in procs:
        assert p.wait(timeout=180) == 0

    # single-process reference with the same seeds
    env = dict(os.environ, PYTHONPATH=repo)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out1 = str(tmp_path / "w1.pt")
    r = subprocess.run(
        [sys.executable, os.path.join(here, "_module_trainer_proc.py"),
         "4", out1], env=env, timeout=180)
    assert r.returncode == 0
    got2 = torch.load(out2, weights_only=True)
    got1 = torch.load(out1, weights_only=True)
    assert torch.allclose(got1, got2, atol=1e-5)


def _run_world2(tmp_path, variant, tag):
    import os
    import subprocess
    import sys

    from tfmesos_amd.utils import free_port

    here = os.path.dirname(os.path.abspath(__file__))
    repo = os.path.dirname(here)
    out = str(tmp_path / ("%s.pt" % tag))
    port = free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                   PYTHONPATH=repo)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(here, "_module_trainer_proc.py"),
             "4", out, variant], env=env))
    for p in procs:
        assert p.wait(timeout=180) == 0
    return torch.load(out, weights_only=True)


@pytest.mark.timeout(240)
def test_overlap_matches_no_overlap_world2(tmp_path):
    """The bucketed backward-overlap reduce path must produce masters
    identical to the plain (blocking, post-backward) reduce path."""
    a = _run_world2(tmp_path, "colocate", "ovl")
    b = _run_world2(tmp_path, "colocate-no-overlap", "novl")
    assert torch.allclose(a, b, atol=1e-6)


@pytest.mark.timeout(240)
def test_overlap_pure_ps_rank_world2(tmp_path):
    """Non-colocated: rank 0 is a PURE PS (no backward, no hooks) — it
    must issue the same bucket reduces in the same order with a zero
    contribution. Masters must match the colocated 1-worker run is not
    expected (different n_workers), but must match its own no-overlap
    variant exactly."""
    a = _run_world2(tmp_path, "ps", "psovl")
    b = _run_world2(tmp_path, "ps-no-overlap", "psnovl")
    assert torch.allclose(a, b, atol=1e-6)


def test_fused_apply_gate_is_world1_sgd_gpu_only():
    """The mnist fused-apply fast path must never engage on CPU, under
    momentum, or for non-sgd optimizers (bench.py relies on this gate
    to keep the PS path authoritative everywhere else)."""
    import torch

    from tfmesos_amd.models.mlp import MnistMLP, synthetic_batch
    from tfmesos_amd.ps.replica import SyncReplicaTrainer

    m = MnistMLP(hidden_units=100)
    x, _ = synthetic_batch(100, device="cpu")
    t = SyncReplicaTrainer(m.init_params(), optimizer="sgd",
                           hparams={"lr": 0.01}, device="cpu")
    assert not m.supports_fused_apply(t, x)          # CPU tensors

    t2 = SyncReplicaTrainer(m.init_params(), optimizer="sgd",
                            hparams={"lr": 0.01, "momentum": 0.9},
                            device="cpu")
    assert not m.supports_fused_apply(t2, x)         # momentum

    t3 = SyncReplicaTrainer(m.init_params(), optimizer="adam",
                            hparams={"lr": 0.01}, device="cpu")
    assert not m.supports_fused_apply(t3, x)         # optimizer
