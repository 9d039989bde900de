"""End-to-end launcher test ON the MI355X box: cluster() -> local
GPU-slot scheduler -> agent with HIP_VISIBLE_DEVICES pinning -> remote
execution of HIP-kernel ops on the granted device."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.mark.timeout(300)
def test_cluster_launches_gpu_worker_and_runs_mfma():
    from tfmesos_amd import cluster, rt

    jobs = [dict(name="ps", num=1),
            dict(name="worker", num=1, gpus=1)]

    def gpu_probe(ctx):
        import os

        import torch as T

        from tfmesos_amd import ops
        assert T.cuda.is_available(), "worker was granted a GPU"
        a = T.randn(64, 64, device="cuda:0", dtype=T.bfloat16)
        b = T.randn(64, 64, device="cuda:0", dtype=T.bfloat16)
        out = ops.gemm_bias_act(a, b)          # MFMA kernel, no fallback
        ref = a.float() @ b.float()
        err = float((out.float() - ref).abs().max())
        rel = err / max(1e-6, float(ref.abs().max()))
        return {"device": ctx.device,
                "visible": os.environ.get("HIP_VISIBLE_DEVICES"),
                "err": err, "rel": rel}

    with cluster(jobs, quiet=True) as c:
        sess = rt.Session(c.targets["/job:worker/task:0"],
                          targets=c.targets, secret=c.secret)
        r = sess.call(gpu_probe, device="/job:worker/task:0")
        sess.close()
    assert r["device"] == "cuda:0"
    assert r["visible"] is not None
    # relative bound: bf16 MFMA with fp32 accumulate on a K=64 product
    # stays well inside 2% of the output magnitude (the old abs<1.0
    # bound would have passed a badly broken kernel)
    assert r["rel"] < 0.02, r


@pytest.mark.timeout(300)
def test_tfa_run_replica_training_on_gpu(tmp_path):
    """CLI-launched between-graph sync training where the worker holds
    the GPU (ps on CPU), exercising the full cmd-mode contract."""
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "script", "tfa_run"),
         "-w", "1", "-s", "1", "-Gw", "1", "--worker-logs", "*", "--",
         sys.executable, os.path.join(repo, "examples", "mnist",
                                      "mnist_replica.py"),
         "--sync_replicas", "--train_steps", "30"],
        capture_output=True, text=True, timeout=240,
        env=dict(os.environ, PYTHONPATH=repo), cwd=repo)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "validation xent" in r.stdout
