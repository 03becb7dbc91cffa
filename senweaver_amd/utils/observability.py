"""Observability services: metrics capture, token-usage tracking, perf monitor.

Rebuilds the reference's telemetry surface:
- MetricsService (common/metricsService.ts + electron-main/metricsMainService.ts):
  capture(event, props) with shape-only metadata, debug-info dump;
- TokenUsageTracker (common/tokenUsageTracker.ts): per-request token records
  + aggregate stats;
- PerformanceMonitor (common/performanceMonitor.ts): prep-pipeline timings
  against the reference's SLO thresholds (sysmsg gen <=2s, dir traversal
  <=2s, trimming <=200ms, total prep <=3s), disabled by default, plus the
  PerfTimer helper and the 4-chars/token estimate (:244-247).

The GPU-side analog of the reference's perf spans is rocprofv3 (kernel
traces live in profiles/); these classes cover the host-side pipeline.
"""

from __future__ import annotations

import time
from collections import defaultdict
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional


class MetricsService:
    """capture(event, props) — in-process sink with pluggable exporters."""

    def __init__(self) -> None:
        self.events: List[Dict[str, Any]] = []
        self._exporters: List[Callable[[str, Dict[str, Any]], None]] = []

    def add_exporter(self, fn: Callable[[str, Dict[str, Any]], None]) -> None:
        self._exporters.append(fn)

    def capture(self, event: str, props: Optional[Dict[str, Any]] = None) -> None:
        rec = {"event": event, "ts": int(time.time() * 1000), **(props or {})}
        self.events.append(rec)
        for fn in self._exporters:
            try:
                fn(event, rec)
            except Exception:
                pass

    def debug_info(self) -> Dict[str, Any]:
        counts: Dict[str, int] = defaultdict(int)
        for e in self.events:
            counts[e["event"]] += 1
        return {"totalEvents": len(self.events), "byEvent": dict(counts)}


@dataclass
class TokenUsageRecord:
    request_id: str
    model: str
    input_tokens: int
    output_tokens: int
    timestamp: int


class TokenUsageTracker:
    def __init__(self, max_records: int = 1000) -> None:
        self._records: List[TokenUsageRecord] = []
        self._max = max_records

    def record(self, request_id: str, model: str, input_tokens: int, output_tokens: int) -> None:
        self._records.append(TokenUsageRecord(
            request_id, model, input_tokens, output_tokens, int(time.time() * 1000)))
        if len(self._records) > self._max:
            self._records = self._records[-self._max:]

    def stats(self) -> Dict[str, Any]:
        total_in = sum(r.input_tokens for r in self._records)
        total_out = sum(r.output_tokens for r in self._records)
        by_model: Dict[str, Dict[str, int]] = defaultdict(lambda: {"input": 0, "output": 0, "requests": 0})
        for r in self._records:
            m = by_model[r.model]
            m["input"] += r.input_tokens
            m["output"] += r.output_tokens
            m["requests"] += 1
        return {"totalRequests": len(self._records), "totalInputTokens": total_in,
                "totalOutputTokens": total_out, "byModel": dict(by_model)}


# SLO thresholds — performanceMonitor.ts:46-53
PERF_THRESHOLDS_MS = {
    "systemMessageGeneration": 2000,
    "directoryTraversal": 2000,
    "messageTrimming": 200,
    "totalPreparation": 3000,
}
CACHE_HIT_RATE_TARGET = 0.5


class PerfTimer:
    def __init__(self, name: str, monitor: Optional["PerformanceMonitor"] = None) -> None:
        self.name = name
        self._monitor = monitor
        self._t0 = 0.0
        self.elapsed_ms = 0.0

    def __enter__(self) -> "PerfTimer":
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc) -> None:
        self.elapsed_ms = (time.perf_counter() - self._t0) * 1000
        if self._monitor is not None:
            self._monitor.record(self.name, self.elapsed_ms)


class PerformanceMonitor:
    """Disabled by default like the reference (:57)."""

    def __init__(self, enabled: bool = False) -> None:
        self.enabled = enabled
        self._samples: Dict[str, List[float]] = defaultdict(list)
        self.violations: List[Dict[str, Any]] = []

    def record(self, metric: str, elapsed_ms: float) -> None:
        if not self.enabled:
            return
        self._samples[metric].append(elapsed_ms)
        thr = PERF_THRESHOLDS_MS.get(metric)
        if thr is not None and elapsed_ms > thr:
            self.violations.append({"metric": metric, "elapsedMs": elapsed_ms, "thresholdMs": thr})

    def timer(self, metric: str) -> PerfTimer:
        return PerfTimer(metric, self)

    @staticmethod
    def estimate_tokens(text: str) -> int:
        return len(text) // 4  # 4 chars/token (:244-247)

    def summary(self) -> Dict[str, Any]:
        return {
            m: {"count": len(v), "avgMs": sum(v) / len(v), "maxMs": max(v)}
            for m, v in self._samples.items() if v
        }


class MetricsPoller:
    """Periodic metrics heartbeat (reference MetricsPollService: a 15-min
    interval that captures a poll event so dashboards see liveness even in
    idle sessions).  Thread-timer based; start()/stop(); the capture payload
    includes the aggregate debug counts so each heartbeat is self-contained.
    """

    DEFAULT_INTERVAL_S = 15 * 60

    def __init__(self, metrics: MetricsService,
                 interval_s: float = DEFAULT_INTERVAL_S,
                 extra_props: Optional[Callable[[], Dict[str, Any]]] = None) -> None:
        self._metrics = metrics
        self._interval = interval_s
        self._extra = extra_props
        self._timer = None
        self._running = False
        self.polls = 0

    def start(self) -> None:
        if self._running:
            return
        self._running = True
        self._schedule()

    def _schedule(self) -> None:
        import threading
        self._timer = threading.Timer(self._interval, self._fire)
        self._timer.daemon = True
        self._timer.start()

    def _fire(self) -> None:
        if not self._running:
            return
        self.poll_once()
        self._schedule()

    def poll_once(self) -> None:
        props: Dict[str, Any] = {"poll": self.polls, **self._metrics.debug_info()}
        if self._extra is not None:
            try:
                props.update(self._extra())
            except Exception:
                pass
        self._metrics.capture("metrics_poll_heartbeat", props)
        self.polls += 1

    def stop(self) -> None:
        self._running = False
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None


class trace_range:
    """Host-side analog of the reference's perf spans that also shows up in
    rocprofv3 timelines: wraps a region in a ROCtx/NVTX range when CUDA(=HIP)
    is available, and records the wall time into a PerformanceMonitor.

    Usage: ``with trace_range("score_batch", monitor): ...``
    """

    def __init__(self, name: str, monitor: Optional[PerformanceMonitor] = None) -> None:
        self.name = name
        self._monitor = monitor
        self._nvtx = False
        self._t0 = 0.0

    def __enter__(self) -> "trace_range":
        try:
            import torch
            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(self.name)
                self._nvtx = True
        except Exception:
            pass
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc) -> None:
        if self._monitor is not None:
            self._monitor.record(self.name, (time.perf_counter() - self._t0) * 1000)
        if self._nvtx:
            import torch
            torch.cuda.nvtx.range_pop()
