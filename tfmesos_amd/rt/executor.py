"""Serve-mode runtime server: authenticated RPC executor + PS service.

One Executor runs per serve-mode task (the reference ran a bare
``tf.train.Server`` here, ``tfmesos/server.py:51-66``). It owns

* a tensor/object store (``put``/``get``/``delete``),
* a generic ``run`` op executing a client-shipped callable on this
  task's device (cloudpickle payloads, deserialized only after the
  frame's HMAC verifies — see ``tfmesos_amd/wire.py``),
* the parameter-server service (``ps_*`` ops) backed by
  ``tfmesos_amd.ps.store.PStore`` with fused HIP apply kernels on GPU.
"""

import logging
import pickle
import socket
import threading
import traceback

import cloudpickle

from tfmesos_amd import wire

logger = logging.getLogger(__name__)


class ExecContext(object):
    """Passed as first argument to every remotely-run callable."""

    def __init__(self, executor):
        self._ex = executor
        self.config = executor.config
        self.store = executor.store
        self.device = executor.device
        self.job_name = executor.config.get("job_name")
        self.task_index = executor.config.get("task_index")

    def rpc(self, target_name, request):
        """Call another task's executor (peer-to-peer)."""
        addr = self._ex.peer_addr(target_name)
        sock = wire.connect(addr, timeout=60)
        try:
            wire.send_msg(sock, request, self._ex.secret)
            reply = wire.recv_msg(sock, self._ex.secret)
        finally:
            sock.close()
        if isinstance(reply, dict) and "err" in reply:
            raise RuntimeError("peer %s: %s" % (target_name, reply["err"]))
        return reply["ok"] if isinstance(reply, dict) and "ok" in reply else reply

    def fetch(self, target_name, key):
        return self.rpc(target_name, {"op": "get", "key": key})


class Executor(object):

    def __init__(self, lsock, secret, config):
        self.lsock = lsock
        self.secret = secret
        self.config = config
        self.store = {}
        self.store_lock = threading.RLock()
        self._stop = threading.Event()
        self._ps = None  # lazy PStore

        gpus = config.get("gpus", 0)
        if gpus and gpus > 0:
            import torch
            if not torch.cuda.is_available():
                raise RuntimeError(
                    "task was granted %s GPUs but torch sees none "
                    "(HIP_VISIBLE_DEVICES=%r)" % (gpus, __import__("os")
                                                  .environ.get("HIP_VISIBLE_DEVICES")))
            self.device = "cuda:0"
        else:
            self.device = "cpu"

    # ------------------------------------------------------------- serving

    def request_stop(self):
        self._stop.set()
        # unblock accept()
        try:
            s = socket.create_connection(self.lsock.getsockname(), timeout=5)
            s.close()
        except OSError:
            pass

    def serve_forever(self):
        self.lsock.settimeout(1.0)
        while not self._stop.is_set():
            try:
                conn, _ = self.lsock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            threading.Thread(target=self._handle_conn, args=(conn,),
                             daemon=True).start()
        self.lsock.close()

    def _handle_conn(self, conn):
        try:
            while not self._stop.is_set():
                try:
                    req = wire.recv_msg(conn, self.secret)
                except wire.WireError:
                    return
                try:
                    reply = {"ok": self._dispatch(req)}
                except Exception as e:  # report, don't kill the server
                    logger.warning("op failed: %s\n%s", e,
                                   traceback.format_exc())
                    reply = {"err": "%s: %s" % (type(e).__name__, e)}
                try:
                    wire.send_msg(conn, reply, self.secret)
                except TypeError as e:
                    wire.send_msg(conn, {"err": "unencodable reply: %s" % e},
                                  self.secret)
        finally:
            conn.close()

    def peer_addr(self, target_name):
        # "/job:ps/task:0" -> cluster_def["ps"][0]
        job, idx = target_name.split("/job:")[1].split("/task:")
        return self.config["cluster_def"][job][int(idx)]

    # ------------------------------------------------------------ dispatch

    def _dispatch(self, req):
        op = req.get("op")
        if op == "ping":
            return "pong"
        if op == "put":
            with self.store_lock:
                self.store[req["key"]] = self._to_device(req["value"])
            return True
        if op == "get":
            with self.store_lock:
                return self._from_device(self.store[req["key"]])
        if op == "delete":
            with self.store_lock:
                self.store.pop(req["key"], None)
            return True
        if op == "keys":
            with self.store_lock:
                return sorted(self.store)
        if op == "run":
            fn = cloudpickle.loads(req["fn"])
            args = pickle.loads(req["args"]) if req.get("args") else ()
            ctx = ExecContext(self)
            args = tuple(self._resolve_ref(ctx, a) for a in args)
            result = fn(ctx, *args)
            return self._encode_result(result)
        if op.startswith("ps_"):
            return self._ps_dispatch(op, req)
        raise ValueError("unknown op %r" % op)

    def _resolve_ref(self, ctx, a):
        from tfmesos_amd.rt.ref import RemoteRef
        if isinstance(a, RemoteRef):
            me = "/job:%s/task:%s" % (self.config.get("job_name"),
                                      self.config.get("task_index"))
            if a.target == me:
                with self.store_lock:
                    return self.store[a.key]
            return self._to_device(ctx.fetch(a.target, a.key))
        return a

    def _to_device(self, v):
        import torch
        if isinstance(v, torch.Tensor):
            return v.to(self.device)
        return v

    def _from_device(self, v):
        import torch
        if isinstance(v, torch.Tensor):
            return v.detach().cpu()
        return v

    def _encode_result(self, result):
        """Prefer wire-native encoding; fall back to cloudpickle tag."""
        try:
            wire.pack(self._from_device_tree(result))
            return self._from_device_tree(result)
        except TypeError:
            return {b"__cp__": cloudpickle.dumps(result)}

    def _from_device_tree(self, v):
        import torch
        if isinstance(v, torch.Tensor):
            return v.detach().cpu()
        if isinstance(v, (list, tuple)):
            return [self._from_device_tree(x) for x in v]
        if isinstance(v, dict):
            return {k: self._from_device_tree(x) for k, x in v.items()}
        return v

    # ------------------------------------------------------------ PS ops

    def _ps_dispatch(self, op, req):
        from tfmesos_amd.ps.store import PStore
        if op == "ps_init":
            self._ps = PStore(device=self.device)
            self._ps.init_params(req["params"], optimizer=req.get("optimizer", "sgd"),
                                 **(req.get("hparams") or {}))
            return True
        if self._ps is None:
            raise RuntimeError("PS not initialized (call ps_init first)")
        if op == "ps_pull":
            return self._ps.pull(req.get("names"), dtype=req.get("dtype"))
        if op == "ps_push":
            return self._ps.push_apply(req["grads"])
        if op == "ps_step":
            return self._ps.global_step
        if op == "ps_save":
            self._ps.save(req["path"])
            return True
        if op == "ps_load":
            self._ps.load(req["path"])
            return True
        raise ValueError("unknown ps op %r" % op)
