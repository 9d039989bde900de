"""Resource backends: who owns the machine's CPU/GPU/mem slots.

The reference delegated resource offers / task launch / status updates to
an external Mesos master+agent (driven via pymesos,
``tfmesos/scheduler.py:223,277,384``). Here the same offer-match/launch/
status surface is implemented in-process:

* ``LocalBackend`` — inventories the local node (8 MI355X slots, CPUs,
  RAM), synthesizes offers over the free pool, launches tasks as local
  subprocesses pinned with ``HIP_VISIBLE_DEVICES``, and reports process
  exits as status updates.
* ``FakeBackend`` (tests) — fully scripted offers/launches, no processes.

The listener (scheduler) receives ``resource_offers(backend, [Offer])``
and ``status_update(backend, Update)`` callbacks, mirroring the Mesos
callback shape so the scheduler logic stays testable without GPUs.
"""

import logging
import os
import subprocess
import sys
import threading

from tfmesos_amd import utils

logger = logging.getLogger(__name__)

# Terminal task states (reference tfmesos/scheduler.py:440-443)
TASK_STAGING = "TASK_STAGING"
TASK_RUNNING = "TASK_RUNNING"
TASK_FINISHED = "TASK_FINISHED"
TASK_FAILED = "TASK_FAILED"
TASK_KILLED = "TASK_KILLED"
TASK_ERROR = "TASK_ERROR"
TERMINAL_STATES = (TASK_FINISHED, TASK_FAILED, TASK_KILLED, TASK_ERROR)


class Offer(object):
    _next = [0]

    def __init__(self, hostname, cpus, mem, gpu_ids):
        Offer._next[0] += 1
        self.id = "offer-%d" % Offer._next[0]
        self.hostname = hostname
        self.cpus = cpus
        self.mem = mem
        self.gpu_ids = list(gpu_ids)

    def __repr__(self):
        return "<Offer %s cpus=%s mem=%s gpus=%s>" % (
            self.id, self.cpus, self.mem, self.gpu_ids)


class Update(object):
    def __init__(self, task_id, state, message=""):
        self.task_id = task_id
        self.state = state
        self.message = message


class LaunchSpec(object):
    """What the scheduler hands to backend.launch() per task."""

    def __init__(self, task_id, argv=None, shell_cmd=None, env=None,
                 gpu_ids=(), cpus=1.0, mem=1024.0, cwd=None):
        self.task_id = task_id
        self.argv = argv              # exec-style launch
        self.shell_cmd = shell_cmd    # or shell launch
        self.env = dict(env or {})
        self.gpu_ids = list(gpu_ids)
        self.cpus = cpus
        self.mem = mem
        self.cwd = cwd


class LocalBackend(object):
    """Single-node slot allocator + subprocess launcher."""

    def __init__(self, cpus=None, mem=None, gpus=None, hostname=None):
        self.total_cpus = cpus if cpus is not None else float(utils.cpu_count())
        self.total_mem = mem if mem is not None else float(utils.mem_mb())
        ngpu = gpus if gpus is not None else utils.gpu_count()
        self.free_gpu_ids = list(range(ngpu))
        self.free_cpus = self.total_cpus
        self.free_mem = self.total_mem
        self.hostname = hostname or utils.hostname()

        self.listener = None
        self._lock = threading.RLock()
        self._procs = {}        # task_id -> (Popen, gpu_ids, cpus, mem)
        self._stopping = False
        self._suppressed = False
        self._threads = []

    # -- lifecycle ---------------------------------------------------------

    def start(self, listener):
        self.listener = listener
        self._emit_offer()

    def stop(self):
        with self._lock:
            self._stopping = True
            procs = list(self._procs.items())
        for task_id, (proc, _, _, _) in procs:
            if proc.poll() is None:
                proc.terminate()
        for task_id, (proc, _, _, _) in procs:
            try:
                proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                proc.kill()
                proc.wait(timeout=10)
        for t in self._threads:
            t.join(timeout=5)

    # -- offers ------------------------------------------------------------

    def _emit_offer(self):
        if self.listener is None or self._stopping or self._suppressed:
            return
        with self._lock:
            offer = Offer(self.hostname, self.free_cpus, self.free_mem,
                          self.free_gpu_ids)
        self.listener.resource_offers(self, [offer])

    def revive(self):
        """Mesos reviveOffers analogue: re-emit the free pool."""
        self._suppressed = False
        self._emit_offer()

    def suppress(self):
        """Mesos suppressOffers analogue."""
        self._suppressed = True

    def decline(self, offer):
        pass  # the free pool is re-offered on revive()/resource release

    # -- launch ------------------------------------------------------------

    def launch(self, offer, specs):
        """Deduct resources and spawn one subprocess per LaunchSpec."""
        with self._lock:
            for spec in specs:
                for g in spec.gpu_ids:
                    self.free_gpu_ids.remove(g)
                self.free_cpus -= spec.cpus
                self.free_mem -= spec.mem
            for spec in specs:
                self._spawn(spec)

    def _spawn(self, spec):
        env = dict(os.environ)
        env.update(spec.env)
        # Device pinning: the task sees ONLY its granted MI355X slots.
        env["HIP_VISIBLE_DEVICES"] = ",".join(str(g) for g in spec.gpu_ids)
        env["CUDA_VISIBLE_DEVICES"] = env["HIP_VISIBLE_DEVICES"]
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        # Propagate the client's import path (reference
        # tfmesos/scheduler.py:162-177 ships sys.path as PYTHONPATH).
        env["PYTHONPATH"] = os.pathsep.join(
            [p for p in sys.path if p] +
            [p for p in env.get("PYTHONPATH", "").split(os.pathsep) if p])
        env["PYTHONUNBUFFERED"] = "1"
        if spec.argv is not None:
            proc = subprocess.Popen(spec.argv, env=env, cwd=spec.cwd)
        else:
            proc = subprocess.Popen(spec.shell_cmd, shell=True, env=env,
                                    cwd=spec.cwd)
        with self._lock:
            self._procs[spec.task_id] = (proc, spec.gpu_ids, spec.cpus, spec.mem)
        t = threading.Thread(target=self._reap, args=(spec.task_id, proc),
                             daemon=True)
        t.start()
        self._threads.append(t)
        if self.listener:
            self.listener.status_update(self, Update(spec.task_id, TASK_RUNNING))

    def _reap(self, task_id, proc):
        rc = proc.wait()
        with self._lock:
            entry = self._procs.pop(task_id, None)
            if entry is not None:
                _, gpu_ids, cpus, mem = entry
                self.free_gpu_ids.extend(gpu_ids)
                self.free_cpus += cpus
                self.free_mem += mem
            stopping = self._stopping
        if stopping or self.listener is None:
            return
        if rc == 0:
            self.listener.status_update(self, Update(task_id, TASK_FINISHED))
        elif rc in (-15, -9):  # terminated/killed by us
            self.listener.status_update(self, Update(task_id, TASK_KILLED,
                                                     "signal %d" % -rc))
        else:
            self.listener.status_update(
                self, Update(task_id, TASK_FAILED, "exit code %s" % rc))
        # freed resources -> new offer for any waiting tasks
        self._emit_offer()

    def kill(self, task_id):
        with self._lock:
            entry = self._procs.get(task_id)
        if entry and entry[0].poll() is None:
            entry[0].terminate()


class FakeBackend(object):
    """Scripted backend for scheduler unit tests (no processes)."""

    def __init__(self, cpus=8.0, mem=65536.0, gpus=8, hostname="testhost"):
        self.free_cpus = cpus
        self.free_mem = mem
        self.free_gpu_ids = list(range(gpus))
        self.hostname = hostname
        self.listener = None
        self.launched = []     # list of LaunchSpec
        self.killed = []
        self.suppressed = False
        self.auto_offer = False

    def start(self, listener):
        self.listener = listener
        if self.auto_offer:
            self.offer()

    def stop(self):
        pass

    def offer(self):
        o = Offer(self.hostname, self.free_cpus, self.free_mem,
                  self.free_gpu_ids)
        self.listener.resource_offers(self, [o])
        return o

    def revive(self):
        self.suppressed = False
        if self.auto_offer:
            self.offer()

    def suppress(self):
        self.suppressed = True

    def decline(self, offer):
        pass

    def launch(self, offer, specs):
        for spec in specs:
            for g in spec.gpu_ids:
                self.free_gpu_ids.remove(g)
            self.free_cpus -= spec.cpus
            self.free_mem -= spec.mem
            self.launched.append(spec)

    def kill(self, task_id):
        self.killed.append(task_id)

    def send_update(self, task_id, state, message=""):
        self.listener.status_update(self, Update(task_id, state, message))
