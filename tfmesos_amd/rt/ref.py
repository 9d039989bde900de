"""RemoteRef: a handle to a value stored in some task's executor store.

Lets graph constants genuinely live on their placed device (the
reference placed ``tf.constant`` on ps tasks, ``examples/plus.py:23-30``):
the client puts the value once, ships only the ref, and the op's
executor fetches it peer-to-peer over the authenticated channel.
"""


class RemoteRef(object):
    __slots__ = ("target", "key")

    def __init__(self, target, key):
        self.target = target  # "/job:ps/task:0"
        self.key = key

    def __repr__(self):
        return "<RemoteRef %s %s>" % (self.target, self.key)

    def __reduce__(self):
        return (RemoteRef, (self.target, self.key))
