"""Per-task agent: remote bootstrap + both operating modes.

Launched by the scheduler as ``python -m tfmesos_amd.agent <task_id>
<scheduler_addr>`` (reference launch line ``tfmesos/scheduler.py:163-167``,
agent logic ``tfmesos/server.py:14-117``). Two modes, as in the reference:

* **serve mode** (``Job.cmd is None``): run the built-in runtime server
  (remote-execution RPC + parameter-server service) and block — the
  client drives it through ``targets`` (reference ``server.py:51-66``
  ran a bare ``tf.train.Server``).
* **cmd mode**: export the env-var contract, substitute ``{ps_hosts}``
  ``{worker_hosts}`` ``{job_name}`` ``{task_index}`` placeholders, run the
  user command under a shell, stream its stdout to the local stdout and
  an optional log-forward socket with a ``[job:index]`` prefix, honour
  ``extra_config['initializer'/'finalizer']`` hooks (reference
  ``server.py:68-109``). Env prefix is ``TFA_`` (renamed from
  ``TFMESOS_``).

Fixes the reference's port-reservation race (``server.py:18-21`` reserved
an ephemeral port by bind-without-listen and hoped TF could rebind it):
here the agent binds its real serving socket up front and advertises the
bound port.
"""

import json
import logging
import os
import socket
import subprocess
import sys
import threading

from tfmesos_amd import wire
from tfmesos_amd.utils import setup_logger

logger = logging.getLogger(__name__)


def _forward_stream(proc, prefix, forward_sock):
    for line in iter(proc.stdout.readline, b""):
        sys.stdout.buffer.write(line)
        sys.stdout.buffer.flush()
        if forward_sock is not None:
            try:
                forward_sock.sendall(prefix + line)
            except OSError:
                forward_sock = None


def _watch_control(sock, secret, on_shutdown):
    """Exit when the scheduler says shutdown or hangs up."""
    try:
        while True:
            msg = wire.recv_msg(sock, secret)
            if isinstance(msg, dict) and msg.get("cmd") == "shutdown":
                break
    except wire.WireError:
        pass
    on_shutdown()


def main(argv):
    setup_logger(logger)
    task_id, sched_addr = argv[1], argv[2]
    secret = bytes.fromhex(os.environ.get("TFA_SECRET", ""))

    # Bind the REAL serving socket first; advertise its actual port.
    lsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    lsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    lsock.bind(("127.0.0.1", 0))
    lsock.listen(64)
    addr = "127.0.0.1:%d" % lsock.getsockname()[1]

    c = wire.connect(sched_addr, timeout=60)
    wire.send_msg(c, {"task_id": task_id, "addr": addr}, secret)
    config = wire.recv_msg(c, secret)
    if not isinstance(config, dict) or config.get("cmd") != "config":
        raise RuntimeError("unexpected config message: %r" % (config,))

    job_name = config["job_name"]
    task_index = config["task_index"]
    cluster_def = config["cluster_def"]
    forward_addresses = config.get("forward_addresses") or {}
    extra_config = config.get("extra_config") or {}
    user_cmd = config.get("user_cmd")
    cwd = config.get("cwd") or None

    forward_fd = None
    target_name = "/job:%s/task:%s" % (job_name, task_index)
    forward_addr = forward_addresses.get(target_name)
    if forward_addr:
        forward_fd = wire.connect(forward_addr, timeout=30)

    wire.send_msg(c, "ok", secret)
    logger.info("task %s (%s) configured; serving at %s", target_name,
                "cmd" if user_cmd else "serve", addr)

    if user_cmd is None:
        # ---- serve mode: built-in runtime server ----
        from tfmesos_amd.rt.executor import Executor
        ex = Executor(lsock, secret, config)
        threading.Thread(target=_watch_control,
                         args=(c, secret, ex.request_stop),
                         daemon=True).start()
        ex.serve_forever()
        return 0

    # ---- cmd mode ----
    lsock.close()
    initializer = extra_config.get("initializer")
    if initializer:
        subprocess.check_call(initializer, shell=True, cwd=cwd)

    ps_hosts = ",".join(cluster_def.get("ps", []))
    worker_hosts = ",".join(cluster_def.get("worker", []))
    env = dict(os.environ)
    env.update({
        "TFA_PS_HOSTS": ps_hosts,
        "TFA_WORKER_HOSTS": worker_hosts,
        "TFA_JOB_NAME": job_name,
        "TFA_TASK_INDEX": str(task_index),
        "TFA_DISTRIBUTED": "yes",
        "TFA_NUM_PS": str(len(cluster_def.get("ps", [])) or 1),
        "TFA_RANK": str(config.get("rank", 0)),
        "TFA_WORLD_SIZE": str(config.get("world_size", 1)),
        "TFA_CLUSTER_DEF": json.dumps(cluster_def),
        "TFA_PROTOCOL": config.get("protocol", "rccl"),
        "PYTHONUNBUFFERED": "1",
    })
    dist_master = config.get("dist_master")
    if dist_master:
        host, port = dist_master.rsplit(":", 1)
        env.setdefault("MASTER_ADDR", host)
        env.setdefault("MASTER_PORT", port)
        env.setdefault("RANK", str(config.get("rank", 0)))
        env.setdefault("WORLD_SIZE", str(config.get("world_size", 1)))

    cmd = user_cmd.format(
        ps_hosts=ps_hosts, worker_hosts=worker_hosts,
        job_name=job_name, task_index=task_index)
    logger.info("exec: %s", cmd)

    proc = subprocess.Popen(cmd, shell=True, cwd=cwd, env=env,
                            stdout=subprocess.PIPE,
                            stderr=subprocess.STDOUT)
    stop_evt = threading.Event()
    threading.Thread(target=_watch_control,
                     args=(c, secret, stop_evt.set), daemon=True).start()

    def _killer():
        stop_evt.wait()
        if proc.poll() is None:
            proc.terminate()
    threading.Thread(target=_killer, daemon=True).start()

    prefix = ("[%s:%s] " % (job_name, task_index)).encode()
    _forward_stream(proc, prefix, forward_fd)
    rc = proc.wait()

    finalizer = extra_config.get("finalizer")
    if finalizer:
        subprocess.call(finalizer, shell=True, cwd=cwd)
    return rc


if __name__ == "__main__":
    sys.exit(main(sys.argv))
