"""Client side of the remote-execution runtime: Session / RemoteCall."""

import pickle

import cloudpickle

from tfmesos_amd import wire


def _strip_scheme(target):
    return target.split("://", 1)[1] if "://" in target else target


class RemoteCall(object):
    """Raw RPC handle to one task's executor."""

    def __init__(self, target, secret):
        self.addr = _strip_scheme(target)
        self.secret = secret
        self.sock = wire.connect(self.addr, timeout=60)

    def request(self, req):
        wire.send_msg(self.sock, req, self.secret)
        reply = wire.recv_msg(self.sock, self.secret)
        if isinstance(reply, dict) and "err" in reply:
            raise RuntimeError("remote %s: %s" % (self.addr, reply["err"]))
        result = reply["ok"] if isinstance(reply, dict) and "ok" in reply else reply
        if isinstance(result, dict) and b"__cp__" in result:
            result = pickle.loads(result[b"__cp__"])
        return result

    def ping(self):
        return self.request({"op": "ping"})

    def put(self, key, value):
        return self.request({"op": "put", "key": key, "value": value})

    def get(self, key):
        return self.request({"op": "get", "key": key})

    def run(self, fn, *args):
        return self.request({
            "op": "run",
            "fn": cloudpickle.dumps(fn),
            "args": pickle.dumps(args) if args else b"",
        })

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass


class Session(object):
    """Drives graph evaluation against one primary target.

    Parity with the reference's ``tf.Session(c.targets['/job:worker/...'])``
    driving remotely-placed ops (``examples/plus.py:31-33``). ``targets``
    (the scheduler's full map) enables cross-device graphs: each node runs
    on its own device's agent; the session ships intermediate values.
    """

    def __init__(self, target, targets=None, secret=None):
        import os
        if secret is None:
            env = os.environ.get("TFA_SECRET", "")
            secret = bytes.fromhex(env) if env else b""
        self.secret = secret
        self.primary = _strip_scheme(target)
        self.targets = {k: _strip_scheme(v) for k, v in (targets or {}).items()}
        self._conns = {}

    def _conn(self, addr):
        if addr not in self._conns:
            self._conns[addr] = RemoteCall(addr, self.secret)
        return self._conns[addr]

    def _addr_for_device(self, device):
        if device is None:
            return self.primary
        if device in self.targets:
            return self.targets[device]
        raise KeyError("unknown device %r (known: %s)" %
                       (device, sorted(self.targets)))

    def run(self, node_or_nodes, feed_dict=None):
        from tfmesos_amd.rt.graph import Node
        from tfmesos_amd.rt.ref import RemoteRef
        feeds = feed_dict or {}
        cache = {}

        def ev(node):
            if not isinstance(node, Node):
                return node
            if id(node) in cache:
                return cache[id(node)]
            if node in feeds:
                val = feeds[node]
            elif node.kind == "placeholder":
                raise ValueError("placeholder %r not fed" % (node.name,))
            elif node.kind == "constant":
                if node.device is not None and node.device in self.targets:
                    # place the constant on its device's store once;
                    # ops fetch it peer-to-peer
                    key = "const/" + node.name
                    if not getattr(node, "_placed", False):
                        self._conn(self.targets[node.device]).put(key, node.value)
                        node._placed = True
                    val = RemoteRef(node.device, key)
                else:
                    val = node.value
            else:
                args = [ev(i) for i in node.inputs]
                addr = self._addr_for_device(node.device)
                val = self._conn(addr).run(node.fn, *args)
            cache[id(node)] = val
            return val

        def materialize(val):
            if isinstance(val, RemoteRef):
                return self._conn(self._addr_for_device(val.target)).get(val.key)
            return val

        if isinstance(node_or_nodes, (list, tuple)):
            return [materialize(ev(n)) for n in node_or_nodes]
        return materialize(ev(node_or_nodes))

    def call(self, fn, *args, device=None):
        """Run a raw callable on a device's agent: fn(ctx, *args)."""
        return self._conn(self._addr_for_device(device)).run(fn, *args)

    def close(self):
        for c in self._conns.values():
            c.close()
        self._conns = {}

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
