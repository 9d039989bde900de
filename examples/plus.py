#!/usr/bin/env python
"""Smoke test: 2 ps + 2 worker, remote constant placement, prints 42.

Port of the reference acceptance test (``/root/reference/examples/
plus.py``: constants on ps:0/ps:1, add on worker:1, run via worker:0's
session -> 42; README.rst:50-65 "Successfully running the test should
result in an output of 42").
"""

import argparse
import sys

from tfmesos_amd import cluster
from tfmesos_amd import rt


def main(argv):
    parser = argparse.ArgumentParser()
    parser.add_argument("-n", "--name", default="plus")
    parser.add_argument("-m", "--master", default=None)  # accepted for parity
    args = parser.parse_args(argv)

    jobs_def = [
        dict(name="ps", num=2),
        dict(name="worker", num=2),
    ]
    with cluster(jobs_def, name=args.name, master=args.master, quiet=False) as c:
        a = rt.constant(24.0, device="/job:ps/task:0")
        b = rt.constant(18.0, device="/job:ps/task:1")
        op = rt.add(a, b, device="/job:worker/task:1")
        with rt.Session(c.targets["/job:worker/task:0"], targets=c.targets,
                        secret=c.secret) as sess:
            result = sess.run(op)
            print(int(result))
            assert int(result) == 42


if __name__ == "__main__":
    main(sys.argv[1:])
