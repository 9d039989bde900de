"""Local GPU-slot scheduler: resource allocation + task lifecycle.

Replaces the reference's Mesos framework scheduler
(``TFMesosScheduler``, ``tfmesos/scheduler.py:180-481``) with an
in-process allocator over one MI355X node:

* same greedy first-fit offer matching (reference ``:252-266``),
* same rendezvous barrier — every task dials back and registers before
  the cluster is activated (reference ``:341-362``),
* same activation message shape (job_name/task_index/cluster_def/...,
  reference ``:296-308``) plus torch.distributed rendezvous info,
* same failure policy — pre-start failures revived with a fresh id up to
  MAX_FAILURE_COUNT=3, post-start failures fail-fast (reference
  ``:181,384-434``),
* same ``targets`` map and ``finished()`` semantics (reference
  ``:279-286,474-477``).

Unlike the reference (which mutates task state from the pymesos driver
thread and the select loop concurrently with no lock — a real race), all
shared state here is guarded by ``self._lock``.
"""

import logging
import os
import secrets as _secrets
import select
import socket
import sys
import threading
import time

from tfmesos_amd import wire
from tfmesos_amd.backend import (
    LocalBackend, LaunchSpec, TERMINAL_STATES,
    TASK_FINISHED, TASK_RUNNING)
from tfmesos_amd.spec import Job, Task
from tfmesos_amd.utils import setup_logger, free_port

logger = logging.getLogger(__name__)

MAX_FAILURE_COUNT = 3


class LocalScheduler(object):

    def __init__(self, task_spec, role=None, master=None, name=None,
                 quiet=False, volumes=None, env=None, extra_config=None,
                 forward_addresses=None, protocol="rccl", backend=None,
                 containerizer_type=None, force_pull_image=False,
                 start_timeout=600.0):
        """``task_spec``: list of Job. Accepts (and ignores where
        meaningless locally) the reference's kwargs: master/
        containerizer_type/force_pull_image existed for Mesos+Docker
        (reference ``tfmesos/scheduler.py:183-198``)."""
        self.jobs = task_spec
        self.role = role
        self.name = name or "tfmesos_amd"
        self.quiet = quiet
        self.volumes = dict(volumes or {})
        self.env = dict(env or {})
        self.extra_config = dict(extra_config or {})
        self.forward_addresses = forward_addresses or {}
        self.protocol = protocol
        self.start_timeout = start_timeout
        self.backend = backend or LocalBackend()
        self.secret = _secrets.token_bytes(32)

        self._lock = threading.RLock()
        self.started = False
        self.stopped = False
        self._error = None
        self.task_failure_count = {}
        self.job_finished = {}

        # data-plane backend for launched tasks: RCCL only when every
        # task owns a GPU; mixed CPU-ps/GPU-worker clusters use gloo
        # with CPU-staged collectives (ps/replica.py)
        all_gpu = all(j.gpus >= 1 for j in self.jobs if j.num > j.start)
        self.dist_backend = "nccl" if all_gpu else "gloo"

        self.tasks = []
        for job in self.jobs:
            for task_index in range(job.start, job.num):
                self.tasks.append(Task(
                    Task.fresh_id(), job.name, task_index,
                    cpus=job.cpus, mem=job.mem, gpus=job.gpus,
                    cmd=job.cmd, volumes=self.volumes, env=self.env))
        # global rank assignment for the torch.distributed data plane:
        # jobs in declaration order, tasks by index (stable across revives)
        self._rank_of = {}
        for rank, t in enumerate(self.tasks):
            self._rank_of[(t.job_name, t.task_index)] = rank
        self.dist_port = free_port()

        if not quiet:
            setup_logger(logger)

    # ------------------------------------------------------------- offers

    def resource_offers(self, backend, offers):
        """Greedy first-fit match of unoffered tasks into offers
        (reference ``tfmesos/scheduler.py:223-277``)."""
        with self._lock:
            if self.stopped or self._error:
                return
            for offer in offers:
                if all(t.offered for t in self.tasks):
                    backend.suppress()
                    backend.decline(offer)
                    return
                cpus, mem = offer.cpus, offer.mem
                gpu_ids = list(offer.gpu_ids)
                specs = []
                for t in self.tasks:
                    if t.offered:
                        continue
                    if t.cpus <= cpus and t.mem <= mem and t.gpus <= len(gpu_ids):
                        t.offered = True
                        t.gpu_ids = gpu_ids[: int(t.gpus)]
                        gpu_ids = gpu_ids[int(t.gpus):]
                        cpus -= t.cpus
                        mem -= t.mem
                        specs.append(self._launch_spec(t))
                if specs:
                    backend.launch(offer, specs)
                else:
                    backend.decline(offer)

    def _launch_spec(self, task):
        env = dict(task.env)
        env["TFA_SECRET"] = self.secret.hex()
        env["TFA_TASK_ID"] = task.task_id
        env["TFA_DIST_BACKEND"] = self.dist_backend
        spec = LaunchSpec(
            task.task_id,
            argv=[sys.executable, "-m", "tfmesos_amd.agent",
                  task.task_id, self.addr],
            env=env, gpu_ids=task.gpu_ids, cpus=task.cpus, mem=task.mem,
            cwd=os.getcwd())
        return spec

    # ------------------------------------------------------------- status

    def status_update(self, backend, update):
        with self._lock:
            if self.stopped:
                return
            if update.state not in TERMINAL_STATES:
                return
            task = None
            for t in self.tasks:
                if t.task_id == update.task_id:
                    task = t
                    break
            if task is None:
                return  # stale update for a revived task's old id
            if self.started:
                if update.state != TASK_FINISHED:
                    self._fatal("/job:%s/task:%s %s (%s)" % (
                        task.job_name, task.task_index, update.state,
                        update.message))
                    return
                key = self._job_of(task)
                self.job_finished[key] = self.job_finished.get(key, 0) + 1
                if task.connection is not None:
                    try:
                        task.connection.close()
                    except OSError:
                        pass
                    task.connection = None
            else:
                if update.state == TASK_RUNNING:
                    return
                # pre-start failure: revive with a fresh id, up to 3 tries
                # (reference tfmesos/scheduler.py:404-434)
                fkey = (task.job_name, task.task_index)
                self.task_failure_count[fkey] = \
                    self.task_failure_count.get(fkey, 0) + 1
                if self.task_failure_count[fkey] >= MAX_FAILURE_COUNT:
                    self._fatal("/job:%s/task:%s failed %d times: %s" % (
                        task.job_name, task.task_index,
                        self.task_failure_count[fkey], update.message))
                    return
                logger.warning("reviving /job:%s/task:%s after %s",
                               task.job_name, task.task_index, update.state)
                if task.connection is not None:
                    try:
                        task.connection.close()
                    except OSError:
                        pass
                task.task_id = Task.fresh_id()
                task.offered = False
                task.addr = None
                task.connection = None
                task.initialized = False
                task.gpu_ids = []
                backend.revive()

    def _job_of(self, task):
        for job in self.jobs:
            if job.name == task.job_name:
                return job
        raise KeyError(task.job_name)

    def _fatal(self, msg):
        logger.error("fatal: %s", msg)
        self._error = RuntimeError(msg)

    # -------------------------------------------------------------- start

    def _check_feasible(self):
        """Fail fast when the jobs_def can NEVER fit the node: every
        task must be alive simultaneously for the rendezvous barrier, so
        total demand must fit total inventory. The reference (and
        round 1) instead idled until the rendezvous timeout; raising at
        start() with the offending task is strictly more useful."""
        total_cpus = getattr(self.backend, "total_cpus",
                             getattr(self.backend, "free_cpus", None))
        total_mem = getattr(self.backend, "total_mem",
                            getattr(self.backend, "free_mem", None))
        gpu_ids = getattr(self.backend, "free_gpu_ids", None)
        if total_cpus is None or total_mem is None or gpu_ids is None:
            return  # unknown inventory: can't prove infeasibility
        cpus, mem, gpus = 0.0, 0.0, 0
        for t in self.tasks:
            cpus += t.cpus
            mem += t.mem
            gpus += int(t.gpus)
            if cpus > total_cpus or mem > total_mem or gpus > len(gpu_ids):
                raise RuntimeError(
                    "jobs_def does not fit this node: /job:%s/task:%s "
                    "pushes cumulative demand to cpus=%.1f/%.1f "
                    "mem=%.0f/%.0f gpus=%d/%d — it would wait for the "
                    "rendezvous timeout and never start" % (
                        t.job_name, t.task_index, cpus, total_cpus,
                        mem, total_mem, gpus, len(gpu_ids)))

    def start(self):
        self._check_feasible()
        lfd = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        lfd.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        lfd.bind(("127.0.0.1", 0))
        lfd.listen(64)
        self._lfd = lfd
        self.addr = "127.0.0.1:%d" % lfd.getsockname()[1]
        logger.info("scheduler listening at %s", self.addr)

        self.backend.start(self)

        deadline = time.time() + self.start_timeout
        pending = {}  # sock -> partial state
        while True:
            with self._lock:
                if self._error:
                    self.stop()
                    raise self._error
                if all(t.initialized for t in self.tasks):
                    break
            if time.time() > deadline:
                self.stop()
                raise RuntimeError("rendezvous timed out after %.0fs: %s" % (
                    self.start_timeout,
                    [t for t in self.tasks if not t.initialized]))
            rs, _, _ = select.select([lfd] + list(pending), [], [], 0.1)
            for fd in rs:
                if fd is lfd:
                    conn, _ = lfd.accept()
                    conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                    pending[conn] = True
                    continue
                pending.pop(fd, None)
                try:
                    msg = wire.recv_msg(fd, self.secret)
                    task_id, addr = msg["task_id"], msg["addr"]
                except (wire.WireError, KeyError) as e:
                    logger.warning("bad registration: %s", e)
                    fd.close()
                    continue
                with self._lock:
                    matched = False
                    for t in self.tasks:
                        if t.task_id == task_id:
                            t.addr = addr
                            t.connection = fd
                            t.initialized = True
                            matched = True
                            logger.info("registered /job:%s/task:%s at %s "
                                        "(%d/%d)", t.job_name, t.task_index,
                                        addr,
                                        sum(x.initialized for x in self.tasks),
                                        len(self.tasks))
                            break
                    if not matched:
                        logger.warning("unknown task id %s", task_id)
                        fd.close()

        self._activate_cluster()
        with self._lock:
            self.started = True
        logger.info("cluster started; targets: %s", self.targets)

    def _activate_cluster(self):
        """Send every task its config; await 'ok'
        (reference ``tfmesos/scheduler.py:288-318``)."""
        cluster_def = self.cluster_def
        for task in self.tasks:
            msg = {
                "cmd": "config",
                "job_name": task.job_name,
                "task_index": task.task_index,
                "cpus": task.cpus,
                "mem": task.mem,
                "gpus": task.gpus,
                "gpu_ids": task.gpu_ids,
                "user_cmd": task.cmd,
                "cwd": os.getcwd(),
                "cluster_def": cluster_def,
                "forward_addresses": self.forward_addresses,
                "extra_config": self.extra_config,
                "protocol": self.protocol,
                "rank": self._rank_of[(task.job_name, task.task_index)],
                "world_size": len(self.tasks),
                "dist_master": "127.0.0.1:%d" % self.dist_port,
            }
            wire.send_msg(task.connection, msg, self.secret)
            reply = wire.recv_msg(task.connection, self.secret)
            if reply != "ok":
                raise RuntimeError("task /job:%s/task:%s refused config: %r" %
                                   (task.job_name, task.task_index, reply))

    # ---------------------------------------------------------- accessors

    @property
    def cluster_def(self):
        cluster_def = {}
        with self._lock:
            for job in self.jobs:
                addrs = [
                    t.addr for t in sorted(
                        (t for t in self.tasks if t.job_name == job.name),
                        key=lambda t: t.task_index)
                ]
                cluster_def[job.name] = addrs
        return cluster_def

    @property
    def targets(self):
        """Device name -> endpoint (reference returned grpc:// URLs,
        ``tfmesos/scheduler.py:279-286``; ours are tfa:// RPC endpoints)."""
        targets = {}
        with self._lock:
            for task in self.tasks:
                target_name = "/job:%s/task:%s" % (task.job_name,
                                                   task.task_index)
                targets[target_name] = "tfa://" + (task.addr or "")
        return targets

    def finished(self):
        with self._lock:
            if self._error:
                raise self._error
            # ANY job with all its tasks finished => done (reference
            # tfmesos/scheduler.py:474-477)
            for job in self.jobs:
                ntask = job.num - job.start
                if ntask > 0 and self.job_finished.get(job, 0) >= ntask:
                    return True
        return False

    def join(self, poll=0.1, timeout=None):
        """Block until finished() or a fatal error."""
        deadline = None if timeout is None else time.time() + timeout
        while not self.finished():
            if deadline is not None and time.time() > deadline:
                raise TimeoutError("cluster did not finish in %.0fs" % timeout)
            time.sleep(poll)
        return True

    def stop(self):
        with self._lock:
            if self.stopped:
                return
            self.stopped = True
            tasks = list(self.tasks)
        for task in tasks:
            if task.connection is not None:
                try:
                    wire.send_msg(task.connection, {"cmd": "shutdown"},
                                  self.secret)
                except OSError:
                    pass
                try:
                    task.connection.close()
                except OSError:
                    pass
                task.connection = None
        self.backend.stop()
        try:
            self._lfd.close()
        except (AttributeError, OSError):
            pass
        logger.info("scheduler stopped")
