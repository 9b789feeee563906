"""Per-op GPU timing + chrome-trace export.

The reference's only tracing was commented-out wall-clock pairs
(SURVEY.md §5). Here: named phases are bracketed with hipEvents
(torch.cuda.Event) per epoch; `dump_chrome()` writes a chrome://tracing
JSON. Kernel-level detail comes from rocprofv3 (profiles/).
"""
from __future__ import annotations

import json
import time
from contextlib import contextmanager
from typing import List

import torch


class Tracer:
    def __init__(self, enabled: bool = True, device: str = "cuda:0"):
        self.enabled = enabled and torch.cuda.is_available()
        self.device = device
        self.events: List[tuple] = []   # (name, start_ev, end_ev, epoch)
        self.cpu_spans: List[tuple] = []
        self.epoch = 0

    @contextmanager
    def span(self, name: str):
        if not self.enabled:
            t0 = time.perf_counter()
            yield
            self.cpu_spans.append((name, t0, time.perf_counter(), self.epoch))
            return
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        s.record()
        try:
            yield
        finally:
            e.record()
            self.events.append((name, s, e, self.epoch))

    def next_epoch(self):
        self.epoch += 1

    def summarize(self) -> dict:
        """ms totals per phase name (synchronizes)."""
        if self.enabled:
            torch.cuda.synchronize()
        out = {}
        for name, s, e, _ in self.events:
            out[name] = out.get(name, 0.0) + s.elapsed_time(e)
        for name, t0, t1, _ in self.cpu_spans:
            out[name] = out.get(name, 0.0) + (t1 - t0) * 1e3
        return out

    def dump_chrome(self, path: str, rank: int = 0):
        if self.enabled:
            torch.cuda.synchronize()
        evs = []
        t = 0.0
        for name, s, e, ep in self.events:
            dur = s.elapsed_time(e) * 1e3  # us
            evs.append({"name": name, "ph": "X", "ts": t, "dur": dur,
                        "pid": rank, "tid": 0, "args": {"epoch": ep}})
            t += dur
        for name, t0, t1, ep in self.cpu_spans:
            evs.append({"name": name, "ph": "X", "ts": t0 * 1e6,
                        "dur": (t1 - t0) * 1e6, "pid": rank, "tid": 1,
                        "args": {"epoch": ep}})
        with open(path, "w") as f:
            json.dump({"traceEvents": evs}, f)
