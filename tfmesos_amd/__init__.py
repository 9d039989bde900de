"""tfmesos_amd — an MI355X-native distributed-training launcher + data plane.

A from-scratch framework with the capabilities of douban/tfmesos
(reference: /root/reference, a Mesos-based TensorFlow cluster launcher):
the same ``cluster()`` / ``Job`` public API (reference
``tfmesos/__init__.py:7-22``, ``tfmesos/scheduler.py:21-31``), but

* the Mesos resource-offer scheduler is replaced by a **local GPU-slot
  scheduler** over the MI355X devices of one node, launching one pinned
  process per task (``HIP_VISIBLE_DEVICES``);
* the TensorFlow gRPC parameter-server data plane is replaced by a
  PyTorch-ROCm PS runtime whose dense gradient push / parameter pull ride
  **RCCL collectives over xGMI** and whose hot ops (fused optimizer apply,
  bf16 MFMA GEMM, fused softmax-xent, embedding push/pull) are
  **hand-written HIP/CDNA4 kernels** (``tfmesos_amd/ops/csrc``);
* the pickle-over-TCP wire protocol (an RCE wart in the reference,
  ``tfmesos/utils.py:6-15``) is replaced by HMAC-authenticated msgpack
  framing.
"""

from contextlib import contextmanager

from tfmesos_amd.spec import Job
from tfmesos_amd.scheduler import LocalScheduler

__VERSION__ = "0.1.0"
__all__ = ["cluster", "Job", "LocalScheduler", "__VERSION__"]


@contextmanager
def cluster(jobs, **kw):
    """Launch a training cluster and yield the scheduler.

    API parity with reference ``tfmesos/__init__.py:7-22``: ``jobs`` may be
    a dict of Job kwargs, a single ``Job``, or a list of either; extra
    kwargs go to the scheduler. The yielded scheduler exposes ``.targets``
    (device name -> endpoint URL) and ``.finished()``.
    """
    if isinstance(jobs, dict):
        jobs = [Job(**jobs)]
    if isinstance(jobs, Job):
        jobs = [jobs]
    jobs = [job if isinstance(job, Job) else Job(**job) for job in jobs]
    s = LocalScheduler(jobs, **kw)
    try:
        s.start()
        yield s
    finally:
        s.stop()
