"""End-to-end example tests on CPU (the reference used its examples as
the acceptance suite — README.rst:50-65; here they run under pytest)."""

import os
import subprocess
import sys

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
EX = os.path.join(REPO, "examples")


def _run(argv, timeout=240):
    env = dict(os.environ, PYTHONPATH=REPO)
    return subprocess.run([sys.executable] + argv, env=env, cwd=REPO,
                          capture_output=True, timeout=timeout, text=True)


@pytest.mark.timeout(300)
def test_mnist_ingraph_example():
    r = _run([os.path.join(EX, "mnist", "mnist.py"), "--steps", "4"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "trained 8 global steps" in r.stdout


@pytest.mark.timeout(300)
def test_matrix_factorization_example():
    r = _run([os.path.join(EX, "matrix_factorization.py"),
              "--steps", "12", "--size", "300", "--rank", "40"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "final loss" in r.stdout


@pytest.mark.timeout(300)
def test_tfa_run_launches_mnist_replica_sync():
    """CLI -> scheduler -> agents (cmd mode, placeholder substitution,
    env contract) -> sync replica training -> log forwarding."""
    r = _run([os.path.join(REPO, "script", "tfa_run"),
              "-w", "2", "-s", "1", "--worker-logs", "*", "--",
              sys.executable, os.path.join(EX, "mnist", "mnist_replica.py"),
              "--ps_hosts", "{ps_hosts}", "--worker_hosts", "{worker_hosts}",
              "--job_name", "{job_name}", "--worker_index", "{task_index}",
              "--sync_replicas", "--train_steps", "6"])
    assert r.returncode == 0, r.stdout + r.stderr
    assert "[worker:0]" in r.stdout          # forwarded logs, prefixed
    assert "validation xent" in r.stdout


@pytest.mark.timeout(300)
def test_tfa_run_async_mode():
    r = _run([os.path.join(REPO, "script", "tfa_run"),
              "-w", "2", "-s", "1", "--",
              sys.executable, os.path.join(EX, "mnist", "mnist_replica.py"),
              "--train_steps", "5"])
    assert r.returncode == 0, r.stdout + r.stderr


def test_cli_parser_tfrun_flag_parity():
    from tfmesos_amd.cli import build_parser, parse_worker_logs
    p = build_parser()
    a = p.parse_args(["-w", "3", "-s", "2", "-m", "zk://x", "-n", "job",
                      "-C", "DOCKER", "-f", "-Cw", "2", "-Gw", "1",
                      "-Mw", "2048", "-Cs", "1", "-Gs", "0", "-Ms", "512",
                      "-v", "-V", "/a:/b", "-r", "role", "--worker-logs",
                      "1,2", "--", "echo", "hi"])
    assert a.nworker == 3 and a.nserver == 2
    assert a.worker_gpus == 1 and a.worker_mem == 2048
    assert a.volume == ["/a:/b"]
    assert parse_worker_logs("*", 3) == [0, 1, 2]
    assert parse_worker_logs("1,2", 3) == [1, 2]
    assert parse_worker_logs("0", 3) == [0]
