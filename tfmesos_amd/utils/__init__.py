"""Shared utilities: logging, device inventory, address helpers."""

import logging
import os
import socket
import subprocess
import sys


def setup_logger(logger):
    """Console logger matching the reference's format intent
    (``tfmesos/utils.py:18-27``)."""
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(logging.Formatter(
        "[%(asctime)s %(levelname)s %(name)s] %(message)s"))
    logger.addHandler(handler)
    logger.setLevel(logging.INFO)
    logger.propagate = False


def hostname():
    # Container hostnames may not resolve; prefer loopback for single-node.
    return os.environ.get("TFA_HOSTNAME", "127.0.0.1")


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def gpu_count():
    """Number of visible AMD GPUs. Works without importing torch (cheap),
    falls back to torch if the sysfs probe finds nothing."""
    override = os.environ.get("TFA_FAKE_GPUS")
    if override is not None:
        return int(override)
    n = 0
    try:
        out = subprocess.run(
            ["rocm-smi", "--showid", "--json"],
            capture_output=True, text=True, timeout=10)
        if out.returncode == 0:
            import json
            data = json.loads(out.stdout or "{}")
            n = sum(1 for k in data if k.startswith("card"))
    except (OSError, ValueError, subprocess.TimeoutExpired):
        n = 0
    if n == 0:
        try:
            import torch
            if torch.cuda.is_available():
                n = torch.cuda.device_count()
        except ImportError:
            n = 0
    return n


def cpu_count():
    return os.cpu_count() or 1


def mem_mb():
    try:
        with open("/proc/meminfo") as f:
            for line in f:
                if line.startswith("MemTotal:"):
                    return int(line.split()[1]) // 1024
    except OSError:
        pass
    return 8192
