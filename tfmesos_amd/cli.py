"""``tfa_run`` — CLI launcher (parity with the reference's ``script/tfrun``).

Same surface as ``tfrun`` (reference ``script/tfrun:11-33``): builds a
2-job ps+worker cluster where every task runs the user command with
``{ps_hosts}/{worker_hosts}/{job_name}/{task_index}`` substituted
(reference ``:58-75``), binds a log-sink socket and multiplexes the
selected workers' forwarded stdout to the local stdout until the cluster
finishes (reference ``:83-115``).

Usage:
  tfa_run -w 2 -s 1 -- python my_replica.py --ps_hosts {ps_hosts} \\
      --worker_hosts {worker_hosts} --job_name {job_name} \\
      --worker_index {task_index}
"""

import argparse
import json
import select
import socket
import sys
import threading

from tfmesos_amd import cluster
from tfmesos_amd.utils import setup_logger


def build_parser():
    p = argparse.ArgumentParser(
        prog="tfa_run",
        description="Launch a ps/worker training cluster on the local "
                    "MI355X node (tfrun-compatible flags).")
    p.add_argument("-w", "--nworker", type=int, required=True,
                   help="number of worker tasks")
    p.add_argument("-s", "--nserver", type=int, required=True,
                   help="number of ps tasks")
    p.add_argument("-m", "--master", default=None,
                   help="accepted for tfrun parity (no Mesos master here)")
    p.add_argument("-n", "--name", default="tfa_run")
    p.add_argument("-C", "--containerizer_type", default=None,
                   choices=[None, "MESOS", "DOCKER"],
                   help="accepted for tfrun parity (tasks run as local "
                        "processes)")
    p.add_argument("-f", "--force_pull_image", action="store_true",
                   help="accepted for tfrun parity")
    p.add_argument("-Cw", "--worker_cpus", type=float, default=1.0)
    p.add_argument("-Gw", "--worker_gpus", type=float, default=0)
    p.add_argument("-Mw", "--worker_mem", type=float, default=1024.0)
    p.add_argument("-Cs", "--server_cpus", type=float, default=1.0)
    p.add_argument("-Gs", "--server_gpus", type=float, default=0)
    p.add_argument("-Ms", "--server_mem", type=float, default=1024.0)
    p.add_argument("-v", "--verbose", action="store_true")
    p.add_argument("-V", "--volume", action="append", default=[],
                   metavar="SRC:DST", help="repeatable volume mapping")
    p.add_argument("-r", "--role", default=None)
    p.add_argument("-e", "--extra_config", default=None,
                   help="JSON file with extra config (initializer/"
                        "finalizer hooks etc.)")
    p.add_argument("--worker-logs", default="0",
                   help="comma-separated worker indices whose stdout is "
                        "forwarded here, or '*' for all (default: 0)")
    p.add_argument("cmd", nargs=argparse.REMAINDER,
                   help="command to run on every task (after --)")
    return p


def parse_worker_logs(value, nworker):
    if value.strip() == "*":
        return list(range(nworker))
    ids = []
    for part in value.split(","):
        part = part.strip()
        if part:
            ids.append(int(part))
    return [i for i in ids if 0 <= i < nworker]


class LogSink(object):
    """Accepts forward connections from agents and copies their byte
    stream to stdout (reference ``script/tfrun:83-94,101-112``)."""

    def __init__(self):
        self.lsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self.lsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self.lsock.bind(("127.0.0.1", 0))
        self.lsock.listen(64)
        self.addr = "127.0.0.1:%d" % self.lsock.getsockname()[1]
        self._stop = threading.Event()
        self.thread = threading.Thread(target=self._pump, daemon=True)

    def start(self):
        self.thread.start()

    def _pump(self):
        conns = []
        while not self._stop.is_set():
            rs, _, _ = select.select([self.lsock] + conns, [], [], 0.2)
            for fd in rs:
                if fd is self.lsock:
                    conn, _ = self.lsock.accept()
                    conns.append(conn)
                    continue
                try:
                    data = fd.recv(65536)
                except OSError:
                    data = b""
                if not data:
                    conns.remove(fd)
                    fd.close()
                    continue
                sys.stdout.buffer.write(data)
                sys.stdout.buffer.flush()

    def stop(self):
        self._stop.set()
        self.thread.join(timeout=2)
        try:
            self.lsock.close()
        except OSError:
            pass


def main(argv=None):
    args = build_parser().parse_args(argv)
    cmd = list(args.cmd)
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        print("tfa_run: no command given (put it after --)", file=sys.stderr)
        return 2
    cmd_str = " ".join(cmd)

    extra_config = {}
    if args.extra_config:
        with open(args.extra_config) as f:
            extra_config = json.load(f)

    volumes = {}
    for vol in args.volume:
        if ":" not in vol:
            print("tfa_run: bad volume %r (want SRC:DST)" % vol,
                  file=sys.stderr)
            return 2
        src, dst = vol.split(":", 1)
        volumes[src] = dst

    jobs_def = [
        dict(name="ps", num=args.nserver, cpus=args.server_cpus,
             gpus=args.server_gpus, mem=args.server_mem, cmd=cmd_str),
        dict(name="worker", num=args.nworker, cpus=args.worker_cpus,
             gpus=args.worker_gpus, mem=args.worker_mem, cmd=cmd_str),
    ]

    if args.verbose:
        import logging
        setup_logger(logging.getLogger("tfmesos_amd"))

    sink = LogSink()
    sink.start()
    forward_addresses = {
        "/job:worker/task:%d" % i: sink.addr
        for i in parse_worker_logs(args.worker_logs, args.nworker)
    }

    try:
        with cluster(jobs_def, name=args.name, master=args.master,
                     role=args.role, quiet=not args.verbose,
                     volumes=volumes, extra_config=extra_config,
                     containerizer_type=args.containerizer_type,
                     force_pull_image=args.force_pull_image,
                     forward_addresses=forward_addresses) as c:
            c.join()
    finally:
        sink.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
