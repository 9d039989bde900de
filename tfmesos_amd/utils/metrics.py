"""Step timing / throughput metrics (the reference's only
observability was wall-clock prints inside the example; here timing is a
small reusable helper that the launcher-injected runner, examples and
bench share). Emits the benchmark metric (steps/sec)."""

import time


class StepTimer(object):
    """Wall-clock step timer with periodic steps/sec emission."""

    def __init__(self, report_every=50, emit=print, prefix=""):
        self.report_every = report_every
        self.emit = emit
        self.prefix = prefix
        self.count = 0
        self.t0 = None
        self._last_t = None
        self._last_count = 0

    def start(self):
        self.t0 = self._last_t = time.perf_counter()
        return self

    def step(self):
        if self.t0 is None:
            self.start()
        self.count += 1
        if self.report_every and self.count % self.report_every == 0:
            now = time.perf_counter()
            rate = (self.count - self._last_count) / (now - self._last_t)
            self.emit("%sstep %d: %.1f steps/s" % (self.prefix, self.count,
                                                   rate))
            self._last_t = now
            self._last_count = self.count

    @property
    def elapsed(self):
        return time.perf_counter() - self.t0 if self.t0 else 0.0

    def summary(self):
        e = self.elapsed
        return {"steps": self.count, "elapsed_s": e,
                "steps_per_sec": self.count / e if e > 0 else 0.0}
