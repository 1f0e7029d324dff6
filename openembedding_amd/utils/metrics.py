"""Accumulators, scoped stage timers and the periodic reporter.

MI355X rebuild of the reference's observability layer:
- ``VTIMER(1, group, name, ms)`` scoped timers at every operator stage
  (reference EmbeddingPullOperator.cpp:43,119,214 and push/store/init
  likewise) -> :class:`stage_timer` context manager feeding named
  accumulators;
- pull dedup-rate accumulators ``pull_indices`` / ``pull_unique`` gated on a
  perf flag (reference EmbeddingPullOperator.cpp:208-247) -> counters fed by
  ShardedVariable, report prints the dedup rate that diagnoses all-to-all
  sizing;
- periodic accumulator report every ``server.report_interval`` seconds on
  worker 0 (reference WorkerContext.cpp:24-41,140-163) ->
  :class:`Reporter` daemon thread.

Timers around GPU work measure host wall time of the enqueue+sync window;
for per-kernel truth use rocprofv3 (profiles/). These counters are for the
always-on, run-level view the reference had.
"""

from __future__ import annotations

import contextlib
import threading
import time
from typing import Dict, List, Optional


class Accumulator:
    """Sum/count/min/max accumulator (reference pico-core Accumulator)."""

    __slots__ = ("name", "n", "total", "vmin", "vmax", "_lock")

    def __init__(self, name: str):
        self.name = name
        self.n = 0
        self.total = 0.0
        self.vmin = float("inf")
        self.vmax = float("-inf")
        self._lock = threading.Lock()

    def add(self, value: float, count: int = 1) -> None:
        with self._lock:
            self.n += count
            self.total += value
            if value < self.vmin:
                self.vmin = value
            if value > self.vmax:
                self.vmax = value

    @property
    def mean(self) -> float:
        return self.total / self.n if self.n else 0.0

    def snapshot_and_reset(self):
        with self._lock:
            snap = (self.n, self.total, self.vmin, self.vmax)
            self.n = 0
            self.total = 0.0
            self.vmin = float("inf")
            self.vmax = float("-inf")
        return snap


class MetricRegistry:
    """Process-global named accumulators; cheap enough to stay always-on,
    detailed timing gated on :func:`set_perf` (the reference's
    pico_is_evaluate_performance())."""

    def __init__(self):
        self._acc: Dict[str, Accumulator] = {}
        self._lock = threading.Lock()
        self.perf = False

    def accumulator(self, name: str) -> Accumulator:
        a = self._acc.get(name)
        if a is None:
            with self._lock:
                a = self._acc.setdefault(name, Accumulator(name))
        return a

    def add(self, name: str, value: float, count: int = 1) -> None:
        self.accumulator(name).add(value, count)

    def names(self) -> List[str]:
        return sorted(self._acc)

    def report_lines(self, reset: bool = True) -> List[str]:
        lines = []
        for name in self.names():
            acc = self._acc[name]
            n, total, vmin, vmax = (acc.snapshot_and_reset() if reset else
                                    (acc.n, acc.total, acc.vmin, acc.vmax))
            if not n:
                continue
            lines.append(
                f"{name}: n={n} total={total:.6g} mean={total / n:.6g} "
                f"min={vmin:.6g} max={vmax:.6g}")
        # derived: dedup rate (reference pull_indices/pull_unique pairing)
        return lines


REGISTRY = MetricRegistry()


def set_perf(enabled: bool = True) -> None:
    """Enable detailed stage timing (reference evaluate_performance flag)."""
    REGISTRY.perf = enabled


@contextlib.contextmanager
def stage_timer(group: str, name: str, always: bool = False):
    """Scoped timer -> accumulator '<group>.<name>_ms' (reference VTIMER).
    No-op unless perf mode or ``always``."""
    if not (REGISTRY.perf or always):
        yield
        return
    t0 = time.perf_counter()
    try:
        yield
    finally:
        REGISTRY.add(f"{group}.{name}_ms", (time.perf_counter() - t0) * 1e3)


class Reporter:
    """Daemon thread printing the accumulator report every ``interval``
    seconds on rank 0 (reference accumulator monitor,
    WorkerContext.cpp:140-163). Started from Context when
    ``server.report_interval > 0``."""

    def __init__(self, interval: float, rank: int = 0,
                 registry: Optional[MetricRegistry] = None, out=None):
        self.interval = interval
        self.rank = rank
        self.registry = registry or REGISTRY
        self.out = out
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self) -> None:
        if self.rank != 0 or self.interval <= 0 or self._thread:
            return
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="oe-metrics-reporter")
        self._thread.start()

    def _run(self) -> None:
        while not self._stop.wait(self.interval):
            self.report_once()

    def report_once(self) -> None:
        lines = self.registry.report_lines(reset=True)
        if not lines:
            return
        msg = "[openembedding_amd metrics]\n  " + "\n  ".join(lines)
        print(msg, flush=True, file=self.out)

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2.0)
            self._thread = None
