"""Per-phase training tracer (SURVEY §5.1 MI355X equivalent).

The reference's observability is an append-only JSONL bracket audit plus a
wall-clock benchmark harness (tools/simulation_engine_benchmark.py:84-128).
Here the training loop itself is instrumented: HIP events bracket each phase
(rollout+GAE vs. update) so the timings are device wall-clock even when the
phases are single hipGraph replays, and one JSONL record per update is
appended to `trace_file`.  On CPU the same API falls back to perf_counter.

Events are resolved lazily (`drain()` syncs once), so tracing adds only
event-record overhead to the hot loop, never a mid-update sync.
"""
from __future__ import annotations

import json
from contextlib import contextmanager
from time import perf_counter
from typing import Any, Dict, List

import torch


class PhaseTimer:
    """Device-aware phase timer: `with timer.phase("rollout"): ...`."""

    def __init__(self, device: torch.device):
        self.device = device
        self._cuda = device.type == "cuda"
        self._pending: List[tuple] = []  # (name, start, end) events or floats

    @contextmanager
    def phase(self, name: str):
        if self._cuda:
            ev0 = torch.cuda.Event(enable_timing=True)
            ev1 = torch.cuda.Event(enable_timing=True)
            ev0.record()
            try:
                yield
            finally:
                ev1.record()
                self._pending.append((name, ev0, ev1))
        else:
            t0 = perf_counter()
            try:
                yield
            finally:
                self._pending.append((name, t0, perf_counter()))

    def drain(self) -> Dict[str, float]:
        """Resolve all pending phases to milliseconds (one sync on GPU)."""
        if self._cuda and self._pending:
            torch.cuda.synchronize(self.device)
        out: Dict[str, float] = {}
        for name, a, b in self._pending:
            ms = a.elapsed_time(b) if self._cuda else (b - a) * 1e3
            out[name] = out.get(name, 0.0) + ms
        self._pending.clear()
        return out


class TraceWriter:
    """Append-only JSONL trace (one record per drained step)."""

    def __init__(self, path: str):
        self._fh = open(path, "a", encoding="utf-8")

    def write(self, record: Dict[str, Any]) -> None:
        self._fh.write(json.dumps(record) + "\n")
        self._fh.flush()

    def close(self) -> None:
        self._fh.close()
