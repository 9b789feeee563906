"""Per-op GPU timing + chrome-trace export.

The reference's only tracing was commented-out wall-clock pairs
(SURVEY.md §5). Here: named phases are bracketed with hipEvents
(torch.cuda.Event) per epoch; `dump_chrome()` writes a chrome://tracing
JSON with REAL GPU timestamps (each span placed at its hipEvent offset
from a base event recorded at tracer creation, so concurrency — e.g. a
halo all-to-all overlapping interior SpMM — is visible as overlapping
slices). Kernel-level detail comes from rocprofv3 (profiles/).
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
        self.events: List[tuple] = []   # (name, start_ev, end_ev, epoch, tid)
        self.cpu_spans: List[tuple] = []
        self.epoch = 0
        self._base = None
        if self.enabled:
            # timestamp origin: every GPU span is placed at its real
            # offset from this event (hipEventElapsedTime)
            self._base = torch.cuda.Event(enable_timing=True)
            self._base.record()
        self._t0_wall = time.perf_counter()

    @contextmanager
    def span(self, name: str, stream: torch.cuda.Stream = None):
        """Bracket a phase. Pass `stream` to record the events on a
        side stream (comm spans), giving them their own trace row."""
        if not self.enabled:
            t0 = time.perf_counter()
            yield
            self.cpu_spans.append((name, t0, time.perf_counter(), self.epoch))
            return
        s = torch.cuda.Event(enable_timing=True)
        e = torch.cuda.Event(enable_timing=True)
        tid = 0
        if stream is not None:
            with torch.cuda.stream(stream):
                s.record()
            tid = 2
        else:
            s.record()
        try:
            yield
        finally:
            if stream is not None:
                with torch.cuda.stream(stream):
                    e.record()
            else:
                e.record()
            self.events.append((name, s, e, self.epoch, tid))

    def next_epoch(self):
        self.epoch += 1

    def summarize(self) -> dict:
        """ms totals per phase name (synchronizes)."""
        if self.enabled:
            torch.cuda.synchronize()
        out = {}
        for name, s, e, _, _ in self.events:
            out[name] = out.get(name, 0.0) + s.elapsed_time(e)
        for name, t0, t1, _ in self.cpu_spans:
            out[name] = out.get(name, 0.0) + (t1 - t0) * 1e3
        return out

    def dump_chrome(self, path: str, rank: int = 0):
        if self.enabled:
            torch.cuda.synchronize()
        evs = []
        for name, s, e, ep, tid in self.events:
            # real placement: offset of the start event from the base
            ts = self._base.elapsed_time(s) * 1e3  # us since base
            dur = s.elapsed_time(e) * 1e3
            evs.append({"name": name, "ph": "X", "ts": ts, "dur": dur,
                        "pid": rank, "tid": tid, "args": {"epoch": ep}})
        for name, t0, t1, ep in self.cpu_spans:
            evs.append({"name": name, "ph": "X",
                        "ts": (t0 - self._t0_wall) * 1e6,
                        "dur": (t1 - t0) * 1e6, "pid": rank, "tid": 1,
                        "args": {"epoch": ep}})
        meta = [{"name": "thread_name", "ph": "M", "pid": rank, "tid": t,
                 "args": {"name": n}}
                for t, n in ((0, "gpu"), (1, "cpu"), (2, "gpu-comm"))]
        with open(path, "w") as f:
            json.dump({"traceEvents": meta + evs}, f)
