"""Job / Task specs.

Field-for-field parity with the reference spec classes
(``tfmesos/scheduler.py:21-31`` Job, ``:34-59`` Task) so user jobs_defs
port unchanged; runtime state differs because tasks here are local
processes, not Mesos containers.
"""

import uuid


class Job(object):
    """Declarative role spec: a named group of identical tasks.

    name:  role name (e.g. 'ps', 'worker')
    num:   index of the last task + 1
    start: index of the first task (partial ranges allowed)
    cpus/mem/gpus: per-task resource ask (mem in MB)
    cmd:   None  -> serve mode (task runs the built-in PS/worker runtime
                    and blocks; the client drives it remotely)
           str   -> replica mode (task runs this shell command with
                    {ps_hosts} {worker_hosts} {job_name} {task_index}
                    placeholders substituted)
    """

    def __init__(self, name, num, cpus=1.0, mem=1024.0, gpus=0, cmd=None, start=0):
        self.name = name
        self.num = num
        self.cpus = cpus
        self.gpus = gpus
        self.mem = mem
        self.cmd = cmd
        self.start = start

    def __repr__(self):
        return (
            "Job(name=%r, num=%r, cpus=%r, mem=%r, gpus=%r, cmd=%r, start=%r)"
            % (self.name, self.num, self.cpus, self.mem, self.gpus, self.cmd, self.start)
        )


class Task(object):
    """One schedulable task instance of a Job."""

    def __init__(self, task_id, job_name, task_index, cpus=1.0, mem=1024.0,
                 gpus=0, cmd=None, volumes=None, env=None):
        self.task_id = task_id
        self.job_name = job_name
        self.task_index = task_index
        self.cpus = cpus
        self.mem = mem
        self.gpus = gpus
        self.cmd = cmd
        self.volumes = dict(volumes or {})
        self.env = dict(env or {})

        # runtime state
        self.offered = False
        self.gpu_ids = []         # device indices granted at match time
        self.addr = None          # "host:port" advertised by the agent
        self.connection = None    # control socket to the agent
        self.initialized = False  # agent registered back
        self.process = None       # local backend: subprocess handle

    @staticmethod
    def fresh_id():
        return str(uuid.uuid4())

    def __repr__(self):
        return "<Task %s /job:%s/task:%s addr=%s>" % (
            self.task_id, self.job_name, self.task_index, self.addr)
