"""Remote-execution runtime (the "in-graph mode" equivalent).

The reference's fine-grained mode let a client place TF ops on remote
devices and drive them through ``tf.Session(c.targets[...])``
(reference ``examples/plus.py:22-33``, ``README.rst:121-127``). PyTorch
has no remote graph placement, so the MI355X-native equivalent is an
RPC executor on every serve-mode task: the client builds a tiny lazy
graph of device-placed nodes and ``Session.run`` evaluates it by
shipping each node's function to its device's agent.
"""

from tfmesos_amd.rt.client import Session, RemoteCall
from tfmesos_amd.rt.graph import constant, add, mul, matmul, apply_fn, Node

__all__ = ["Session", "RemoteCall", "constant", "add", "mul", "matmul",
           "apply_fn", "Node"]
