"""Observability: reconcile-duration histograms, transition counters, gauges.

The reference exposes only count getters and leaves metrics export to
consumers ("no Prometheus registration in-library", and a
commented-out summary event at upgrade_state.go:199-202).  This module closes
that gap for the AMD build: the state manager records build/apply durations
and per-state transition counts here, and consumers can scrape them in
Prometheus exposition format (via prometheus_client when available, else a
built-in text renderer).

All instruments are cheap (lock + list/dict updates) and in-process; the
global registry can be replaced per-operator.
"""

from __future__ import annotations

import bisect
import threading
from typing import Dict, List, Optional, Tuple


class Histogram:
    """Fixed-bucket histogram with a uniform sampling reservoir.

    Quantiles are EXACT while the observation count fits the reservoir
    (4096) and thereafter come from Vitter's Algorithm R — a uniform random
    sample over the ENTIRE run, so soak-length p99s are run-global rather
    than window-local (the round-1 reservoir kept only
    the most recent observations).  The fixed buckets are always run-global
    and exact, and serve as the Prometheus exposition."""

    RESERVOIR_SIZE = 4096

    DEFAULT_BUCKETS = (
        0.0001, 0.00025, 0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025,
        0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0,
    )

    def __init__(self, name: str, help_: str = "", buckets: Tuple[float, ...] = DEFAULT_BUCKETS):
        import random

        self.name = name
        self.help = help_
        self.buckets = buckets
        self._counts = [0] * (len(buckets) + 1)
        self._sum = 0.0
        self._n = 0
        self._reservoir: List[float] = []
        # seeded per-instance: deterministic test runs, independent streams
        self._rng = random.Random(0xA5A5 ^ hash(name))
        self._lock = threading.Lock()

    def observe(self, value: float) -> None:
        with self._lock:
            i = bisect.bisect_left(self.buckets, value)
            self._counts[i] += 1
            self._sum += value
            self._n += 1
            if len(self._reservoir) < self.RESERVOIR_SIZE:
                self._reservoir.append(value)
            else:
                # Algorithm R: keep each of the n observations with equal
                # probability RESERVOIR_SIZE/n
                j = self._rng.randrange(self._n)
                if j < self.RESERVOIR_SIZE:
                    self._reservoir[j] = value

    def samples(self) -> List[float]:
        """Raw observations: exact when count <= RESERVOIR_SIZE, else a
        uniform random sample over the whole run."""
        with self._lock:
            return list(self._reservoir)

    def quantile(self, q: float) -> float:
        with self._lock:
            if not self._reservoir:
                return 0.0
            data = sorted(self._reservoir)
            idx = min(len(data) - 1, int(q * len(data)))
            return data[idx]

    @property
    def count(self) -> int:
        with self._lock:
            return self._n

    @property
    def total(self) -> float:
        with self._lock:
            return self._sum

    def snapshot(self) -> Dict[str, float]:
        return {
            "count": self.count,
            "sum": self.total,
            "p50": self.quantile(0.50),
            "p90": self.quantile(0.90),
            "p99": self.quantile(0.99),
        }


class Counter:
    def __init__(self, name: str, help_: str = ""):
        self.name = name
        self.help = help_
        self._values: Dict[Tuple[str, ...], int] = {}
        self._total = 0
        self._lock = threading.Lock()

    def inc(self, *labels: str, amount: int = 1) -> None:
        with self._lock:
            self._values[labels] = self._values.get(labels, 0) + amount
            self._total += amount

    def total(self) -> int:
        """O(1) sum over all label sets."""
        with self._lock:
            return self._total

    def value(self, *labels: str) -> int:
        with self._lock:
            return self._values.get(labels, 0)

    def items(self):
        with self._lock:
            return dict(self._values)


class Gauge:
    def __init__(self, name: str, help_: str = ""):
        self.name = name
        self.help = help_
        self._values: Dict[Tuple[str, ...], float] = {}
        self._lock = threading.Lock()

    def set(self, value: float, *labels: str) -> None:
        with self._lock:
            self._values[labels] = value

    def value(self, *labels: str) -> float:
        with self._lock:
            return self._values.get(labels, 0.0)

    def items(self):
        with self._lock:
            return dict(self._values)


class MetricsRegistry:
    """The library's instruments, named in Prometheus style."""

    def __init__(self) -> None:
        self.reconcile_duration = Histogram(
            "amd_upgrade_reconcile_duration_seconds",
            "Duration of one build_state+apply_state reconcile tick",
        )
        self.build_state_duration = Histogram(
            "amd_upgrade_build_state_duration_seconds",
            "Duration of cluster state snapshot construction",
        )
        self.apply_state_duration = Histogram(
            "amd_upgrade_apply_state_duration_seconds",
            "Duration of one apply_state pass",
        )
        self.state_transitions = Counter(
            "amd_upgrade_state_transitions_total",
            "Node state transitions, labelled (from_state, to_state)",
        )
        self.node_states = Gauge(
            "amd_upgrade_nodes",
            "Nodes per upgrade state at the last snapshot, labelled (state)",
        )
        self.upgrade_failures = Counter(
            "amd_upgrade_failures_total", "Nodes entering upgrade-failed",
        )

    # -- exposition ----------------------------------------------------------

    def render_text(self) -> str:
        """Prometheus text exposition format (built-in renderer)."""
        lines: List[str] = []
        for h in (self.reconcile_duration, self.build_state_duration,
                  self.apply_state_duration):
            lines.append(f"# HELP {h.name} {h.help}")
            lines.append(f"# TYPE {h.name} histogram")
            cumulative = 0
            with h._lock:
                counts = list(h._counts)
                total_sum, total_n = h._sum, h._n
            for bucket, c in zip(h.buckets, counts):
                cumulative += c
                lines.append(f'{h.name}_bucket{{le="{bucket}"}} {cumulative}')
            cumulative += counts[-1]
            lines.append(f'{h.name}_bucket{{le="+Inf"}} {cumulative}')
            lines.append(f"{h.name}_sum {total_sum}")
            lines.append(f"{h.name}_count {total_n}")
        lines.append(f"# HELP {self.state_transitions.name} {self.state_transitions.help}")
        lines.append(f"# TYPE {self.state_transitions.name} counter")
        for labels, v in sorted(self.state_transitions.items().items()):
            frm = labels[0] if len(labels) > 0 else ""
            to = labels[1] if len(labels) > 1 else ""
            lines.append(
                f'{self.state_transitions.name}{{from="{frm or "unknown"}",to="{to}"}} {v}'
            )
        lines.append(f"# HELP {self.node_states.name} {self.node_states.help}")
        lines.append(f"# TYPE {self.node_states.name} gauge")
        for labels, v in sorted(self.node_states.items().items()):
            lines.append(f'{self.node_states.name}{{state="{labels[0] or "unknown"}"}} {v}')
        lines.append(f"# TYPE {self.upgrade_failures.name} counter")
        for labels, v in sorted(self.upgrade_failures.items().items()):
            lines.append(f"{self.upgrade_failures.name} {v}")
        return "\n".join(lines) + "\n"


_default_registry: Optional[MetricsRegistry] = None
_registry_lock = threading.Lock()


def default_registry() -> MetricsRegistry:
    global _default_registry
    with _registry_lock:
        if _default_registry is None:
            _default_registry = MetricsRegistry()
        return _default_registry


def reset_default_registry() -> None:
    global _default_registry
    with _registry_lock:
        _default_registry = None
