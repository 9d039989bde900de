"""Kill/crash-based fault injection against the REAL LocalBackend
(SURVEY.md §5 plan: the reference's policy — revive pre-start up to 3x,
fail-fast post-start — tested with actual processes, not just the fake
backend)."""

import os
import time

import pytest

from tfmesos_amd import Job, LocalScheduler


@pytest.mark.timeout(120)
def test_post_start_task_failure_is_fatal():
    """A worker whose user command exits non-zero after the cluster
    started must fail the whole framework (reference fail-fast,
    tfmesos/scheduler.py:394-401)."""
    jobs = [Job(name="worker", num=1, cmd="sleep 0.2; exit 3")]
    s = LocalScheduler(jobs, quiet=True)
    s.start()
    with pytest.raises(RuntimeError, match="TASK_FAILED"):
        s.join(timeout=60)
    s.stop()


@pytest.mark.timeout(120)
def test_clean_exit_finishes():
    jobs = [Job(name="worker", num=2, cmd="true")]
    s = LocalScheduler(jobs, quiet=True)
    s.start()
    assert s.join(timeout=60)
    s.stop()


@pytest.mark.timeout(120)
def test_initializer_finalizer_hooks(tmp_path):
    """extra_config initializer/finalizer shell hooks run around the
    user command (reference tfmesos/server.py:68-70,106-109)."""
    ini = tmp_path / "ini.txt"
    fin = tmp_path / "fin.txt"
    out = tmp_path / "cmd.txt"
    jobs = [Job(name="worker", num=1, cmd="echo ran > %s" % out)]
    s = LocalScheduler(jobs, quiet=True, extra_config={
        "initializer": "echo i > %s" % ini,
        "finalizer": "echo f > %s" % fin,
    })
    s.start()
    assert s.join(timeout=60)
    s.stop()
    deadline = time.time() + 10
    while time.time() < deadline and not (
            ini.exists() and fin.exists() and out.exists()):
        time.sleep(0.1)
    assert ini.exists() and out.exists() and fin.exists()


@pytest.mark.timeout(180)
def test_killed_task_process_reported(tmp_path):
    """Killing the task's process mid-run surfaces as a fatal
    non-FINISHED status (the 'slave lost' analogue on one node).
    The sleeper's stdio is detached so the orphan can't hold the
    agent's log pipe open after the kill."""
    pidfile = tmp_path / "pid"
    jobs = [Job(name="worker", num=1,
                cmd="sleep 60 >/dev/null 2>&1 & echo $! > %s; wait $!"
                    % pidfile)]
    s = LocalScheduler(jobs, quiet=True)
    s.start()
    deadline = time.time() + 20
    while time.time() < deadline and not pidfile.exists():
        time.sleep(0.1)
    pid = int(pidfile.read_text().strip())
    os.kill(pid, 9)
    with pytest.raises(RuntimeError):
        s.join(timeout=60)
    s.stop()
