"""Built-in timing/throughput observability.

The reference carries llama.cpp timing fields but never reports them
(SURVEY §5.1); here tokens/sec and per-stage wall times are first-class:
`DistributedLLM.generate` exposes a `ThroughputMeter`, and `StageTimer`
wraps stages with CUDA-event timing on GPU (host perf_counter on CPU) so
pipeline stages can be profiled without rocprof.
"""
from __future__ import annotations

import time
from collections import defaultdict
from typing import Dict, Optional


class ThroughputMeter:
    """Counts items (tokens) over wall time; report() -> dict."""

    def __init__(self):
        self.reset()

    def reset(self) -> None:
        self.t0: Optional[float] = None
        self.t1: Optional[float] = None
        self.count = 0

    def tick(self, n: int = 1) -> None:
        now = time.perf_counter()
        if self.t0 is None:
            self.t0 = now
        self.t1 = now
        self.count += n

    def report(self) -> Dict[str, float]:
        if self.t0 is None or self.t1 is None or self.t1 <= self.t0:
            return {"count": float(self.count), "seconds": 0.0,
                    "per_second": 0.0}
        dt = self.t1 - self.t0
        return {"count": float(self.count), "seconds": dt,
                "per_second": self.count / dt}


class StageTimer:
    """Named stage timing; on CUDA uses events so device work is measured
    without host synchronization in the hot loop (events are resolved at
    report() time)."""

    def __init__(self, device: str = "cpu"):
        self.device = device
        self._events = defaultdict(list)   # name -> [(start_ev, end_ev)]
        self._host = defaultdict(float)    # name -> seconds
        self._counts = defaultdict(int)
        self._open = {}

    def start(self, name: str) -> None:
        if self.device == "cuda":
            import torch
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._open[name] = ev
        else:
            self._open[name] = time.perf_counter()

    def stop(self, name: str) -> None:
        if name not in self._open:
            raise KeyError(f"stage {name!r} was not started")
        if self.device == "cuda":
            import torch
            end = torch.cuda.Event(enable_timing=True)
            end.record()
            self._events[name].append((self._open.pop(name), end))
        else:
            self._host[name] += time.perf_counter() - self._open.pop(name)
        self._counts[name] += 1

    def report(self) -> Dict[str, Dict[str, float]]:
        out: Dict[str, Dict[str, float]] = {}
        if self.device == "cuda":
            import torch
            torch.cuda.synchronize()
            for name, pairs in self._events.items():
                total = sum(s.elapsed_time(e) for s, e in pairs) / 1e3
                out[name] = {"seconds": total, "count": self._counts[name],
                             "mean_ms": total * 1e3 / max(1, len(pairs))}
        for name, total in self._host.items():
            out[name] = {"seconds": total, "count": self._counts[name],
                         "mean_ms": total * 1e3 / max(1, self._counts[name])}
        return out
