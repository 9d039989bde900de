"""hipGraph capture for launch-bound inner loops.

The mnist_replica step is 8 short kernels (~60 us wall); capturing it as
ONE hipGraph replay removes the per-kernel CPU launch work and packet
building. Uses torch.cuda.CUDAGraph, which is hipGraph on ROCm.

Only the single-process (world==1) hot path is graphed: collectives are
not graph-captured (RCCL inside capture is not supported across
versions), so multi-rank steps run eager and overlap with the
collectives instead.
"""

import torch


class GraphedStep(object):
    """Capture ``fn()`` (a GPU-only step: no host sync, fixed shapes)
    into a hipGraph after ``warmup`` eager runs; call the instance to
    replay. Falls back to eager on CPU."""

    def __init__(self, fn, warmup=3, tune_iters=200):
        self.fn = fn
        self.graph = None
        if not torch.cuda.is_available():
            return
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(warmup):
                fn()
        torch.cuda.current_stream().wait_stream(stream)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            fn()
        # self-tune: replay has a fixed per-graph submission cost, so
        # for SHORT steps (few kernels) eager can win — measure both
        # and keep the faster mode (the graph captures state either
        # way, so dropping it is safe)
        if tune_iters:
            import time

            def clock(step):
                for _ in range(20):
                    step()
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(tune_iters):
                    step()
                torch.cuda.synchronize()
                return time.perf_counter() - t0

            t_graph = clock(self.graph.replay)
            t_eager = clock(fn)
            # prefer eager unless replay wins CLEARLY: short steps are
            # noisy to clock and replay has been measured slower than
            # 5-6 eager launches on this stack
            if not (t_graph < t_eager * 0.95):
                self.graph = None

    def __call__(self):
        if self.graph is not None:
            self.graph.replay()
        else:
            self.fn()
