"""Tiny lazy graph with device placement, for in-graph-mode parity.

Mirrors what the reference's examples do with ``tf.device(...)`` +
``tf.constant``/``tf.add`` (``examples/plus.py:23-30``): nodes carry a
device name; ``Session.run`` executes each op node on its device's agent.
"""

_counter = [0]


class Node(object):
    def __init__(self, kind, fn=None, inputs=(), device=None, value=None,
                 name=None):
        _counter[0] += 1
        self.kind = kind          # constant | placeholder | op
        self.fn = fn              # op: callable(ctx, *input_values)
        self.inputs = list(inputs)
        self.device = device
        self.value = value
        self.name = name or ("%s_%d" % (kind, _counter[0]))

    def __repr__(self):
        return "<Node %s %s dev=%s>" % (self.kind, self.name, self.device)

    def __hash__(self):
        return id(self)

    def __eq__(self, other):
        return self is other


def constant(value, device=None, name=None):
    return Node("constant", value=value, device=device, name=name)


def placeholder(name=None):
    return Node("placeholder", name=name)


def _binary(opname, torch_op_name, pyop):
    def make(a, b, device=None, name=None):
        def fn(ctx, x, y):
            try:
                import torch
                if isinstance(x, torch.Tensor) or isinstance(y, torch.Tensor):
                    x = x.to(ctx.device) if isinstance(x, torch.Tensor) else x
                    y = y.to(ctx.device) if isinstance(y, torch.Tensor) else y
                    return getattr(torch, torch_op_name)(x, y)
            except ImportError:
                pass
            return pyop(x, y)
        fn.__name__ = opname
        return Node("op", fn=fn, inputs=[a, b], device=device, name=name)
    return make


add = _binary("add", "add", lambda x, y: x + y)
mul = _binary("mul", "mul", lambda x, y: x * y)
matmul = _binary("matmul", "matmul", lambda x, y: x @ y)


def apply_fn(fn, *inputs, device=None, name=None):
    """General op node: fn(ctx, *input_values) runs on `device`."""
    return Node("op", fn=fn, inputs=list(inputs), device=device, name=name)
