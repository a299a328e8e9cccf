"""Per-phase timing records for executor observability.

The reference has logging only (SURVEY.md §5 "Metrics").  Here every task
produces a :class:`TaskRecord` with dispatcher-side phase durations
(connect / stage / dispatch / fetch / cleanup) merged with the remote
stub's own meta JSON (gpu probe, warm-up, user-fn and staging timings).
``bench.py`` derives electrons/sec and p50 latency from these records.
"""

from __future__ import annotations

import json
import time
from contextlib import contextmanager
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, Optional


class PhaseTimer:
    def __init__(self) -> None:
        self._phases: Dict[str, float] = {}
        self._t0 = time.perf_counter()

    @contextmanager
    def phase(self, name: str):
        start = time.perf_counter()
        try:
            yield
        finally:
            self._phases[name] = self._phases.get(name, 0.0) + (
                time.perf_counter() - start
            )

    def snapshot(self) -> Dict[str, float]:
        return dict(self._phases)

    def total(self) -> float:
        return time.perf_counter() - self._t0


@dataclass
class TaskRecord:
    operation_id: str
    gpu_id: Optional[int] = None
    #: dispatcher-side phase durations, seconds
    phases: Dict[str, float] = field(default_factory=dict)
    total_s: float = 0.0
    #: remote stub meta (phases_ms, gpu probe, staging stats)
    remote_meta: Optional[dict] = None

    def load_meta(self, meta_bytes: bytes) -> None:
        try:
            self.remote_meta = json.loads(meta_bytes.decode())
        except (ValueError, UnicodeDecodeError):
            self.remote_meta = None

    def load_meta_file(self, path: str) -> None:
        self.load_meta(Path(path).read_bytes())

    def to_json(self) -> str:
        return json.dumps(
            {
                "operation_id": self.operation_id,
                "gpu_id": self.gpu_id,
                "total_s": round(self.total_s, 6),
                "phases": {k: round(v, 6) for k, v in self.phases.items()},
                "remote_meta": self.remote_meta,
            }
        )


def summarize(records) -> dict:
    """Aggregate a list of TaskRecords into dispatch statistics
    (throughput over the records' total span is the caller's job — this
    reports latency percentiles and phase means)."""
    import statistics

    if not records:
        return {"count": 0}
    totals = [r.total_s for r in records]
    phases: Dict[str, list] = {}
    for r in records:
        for name, v in r.phases.items():
            phases.setdefault(name, []).append(v)
    out = {
        "count": len(records),
        "p50_ms": statistics.median(totals) * 1000.0,
        "p90_ms": sorted(totals)[max(0, int(len(totals) * 0.9) - 1)] * 1000.0,
        "mean_ms": statistics.fmean(totals) * 1000.0,
        "phase_mean_ms": {
            k: statistics.fmean(v) * 1000.0 for k, v in phases.items()
        },
        "gpu_spread": {},
    }
    for r in records:
        if r.gpu_id is not None:
            out["gpu_spread"][r.gpu_id] = out["gpu_spread"].get(r.gpu_id, 0) + 1
    return out
