"""End-to-end loopback test: real scheduler + real agent subprocesses on
CPU — the reference's documented acceptance test shape ("prints 42",
README.rst:50-65)."""

import subprocess
import sys
import os

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tfmesos_amd import cluster, Job  # noqa: E402
from tfmesos_amd import rt  # noqa: E402
from tfmesos_amd.backend import LocalBackend  # noqa: E402


@pytest.mark.timeout(120)
def test_plus_e2e_cpu():
    jobs = [Job(name="ps", num=2, mem=64), Job(name="worker", num=2, mem=64)]
    backend = LocalBackend(gpus=0, mem=4096.0)
    with cluster(jobs, backend=backend, quiet=True, start_timeout=90) as c:
        assert len(c.targets) == 4
        a = rt.constant(24.0, device="/job:ps/task:0")
        b = rt.constant(18.0, device="/job:ps/task:1")
        op = rt.add(a, b, device="/job:worker/task:1")
        with rt.Session(c.targets["/job:worker/task:0"], targets=c.targets,
                        secret=c.secret) as sess:
            assert int(sess.run(op)) == 42


@pytest.mark.timeout(120)
def test_remote_run_and_store():
    jobs = [Job(name="w", num=1, mem=64)]
    backend = LocalBackend(gpus=0, mem=4096.0)
    with cluster(jobs, backend=backend, quiet=True, start_timeout=90) as c:
        with rt.Session(c.targets["/job:w/task:0"], targets=c.targets,
                        secret=c.secret) as sess:
            import torch

            def fn(ctx, x):
                return x * 2 + 1

            out = sess.call(fn, torch.ones(3), device="/job:w/task:0")
            assert torch.equal(out, torch.tensor([3.0, 3.0, 3.0]))


@pytest.mark.timeout(120)
def test_cmd_mode_env_contract(tmp_path):
    """Replica (cmd) mode: placeholders + TFA_* env vars + exit code."""
    out = tmp_path / "env.txt"
    cmd = (f"{sys.executable} -c \"import os,sys;"
           f"open(r'{out}','a').write("
           "os.environ['TFA_JOB_NAME']+':'+os.environ['TFA_TASK_INDEX']+':'"
           "+os.environ['TFA_PS_HOSTS'].count(',').__str__()+':'"
           "+'{job_name}'+':'+'{task_index}'+chr(10))\"")
    jobs = [Job(name="ps", num=2, mem=64, cmd=cmd),
            Job(name="worker", num=1, mem=64, cmd=cmd)]
    backend = LocalBackend(gpus=0, mem=4096.0)
    with cluster(jobs, backend=backend, quiet=True, start_timeout=90) as c:
        c.join(timeout=60)
    lines = sorted(out.read_text().strip().splitlines())
    assert lines == ["ps:0:1:ps:0", "ps:1:1:ps:1", "worker:0:1:worker:0"]


@pytest.mark.timeout(240)
def test_cluster_env_kwarg_reaches_tasks(tmp_path):
    """cluster(env=...) is exported into every task's environment
    (reference tfmesos/scheduler.py:186 env kwarg)."""
    out = tmp_path / "envdump"
    jobs = [dict(name="worker", num=1,
                 cmd="echo $MY_CUSTOM_VAR > %s" % out)]
    from tfmesos_amd import cluster
    with cluster(jobs, quiet=True, env={"MY_CUSTOM_VAR": "hello42"}) as c:
        c.join(timeout=60)
    assert out.read_text().strip() == "hello42"
