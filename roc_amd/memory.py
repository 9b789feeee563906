"""Host-DRAM activation offload (capacity tier).

For graphs whose activations exceed the 288 GB HBM budget (papers100M-
scale, BASELINE.json config #5), saved-for-backward activations are
copied to pinned host memory on a dedicated HIP stream as soon as the
producing op finishes, and copied back when autograd needs them. The
reference kept ALL activations in pinned host (ZC) memory and paid
H2D+D2H on every op (`types.cu:22-32`); here HBM residency is the
default and host offload is opt-in spill, double-buffered via
hipMemcpyAsync on a side stream.

Implementation: torch.autograd.graph.saved_tensors_hooks + a pinned
buffer pool. `record_stream` keeps the GPU source alive until the D2H
copy completes.
"""
from __future__ import annotations

import torch

from . import streamcheck


class ActivationOffload:
    """Context manager: offload saved activations >= min_bytes to host.

    with ActivationOffload(min_bytes=1 << 22):
        loss = model(...); loss.backward()
    """

    def __init__(self, min_bytes: int = 1 << 22, enabled: bool = True,
                 prefetch_window: int = 4):
        import os
        self.min_bytes = min_bytes
        self.enabled = enabled and torch.cuda.is_available()
        if os.environ.get("ROC_OFFLOAD_PREFETCH", "1") == "0":
            prefetch_window = 0
        self.prefetch_window = prefetch_window
        # Hard budget on page-locked host memory: beyond it tensors stay
        # HBM-resident (graceful degradation). Unbounded pinning can
        # destabilize the HOST (hipHostMalloc is non-swappable; a 200+GB
        # pinned footprint took a box down). ROC_OFFLOAD_MAX_GB overrides.
        env_gb = os.environ.get("ROC_OFFLOAD_MAX_GB")
        self.max_pinned_bytes = (int(float(env_gb) * (1 << 30)) if env_gb
                                 else (64 << 30))
        self._pinned_total = 0
        self.stream = torch.cuda.Stream() if self.enabled else None
        self._pool = {}   # (shape, dtype) -> list of free pinned tensors
        self._stats = {"offloaded_bytes": 0, "tensors": 0,
                       "budget_skipped": 0}
        self._ctx = None
        self._entries = []  # live offloaded entries, forward order

    # -- pinned buffer pool -------------------------------------------------
    def _acquire(self, t: torch.Tensor):
        key = (tuple(t.shape), t.dtype)
        free = self._pool.get(key)
        if free:
            return free.pop()
        nbytes = t.numel() * t.element_size()
        if self._pinned_total + nbytes > self.max_pinned_bytes:
            return None  # budget reached: caller keeps the tensor resident
        self._pinned_total += nbytes
        return torch.empty(t.shape, dtype=t.dtype, device="cpu",
                           pin_memory=True)

    def _release(self, cpu: torch.Tensor):
        key = (tuple(cpu.shape), cpu.dtype)
        self._pool.setdefault(key, []).append(cpu)

    # -- hooks ---------------------------------------------------------------
    def _pack(self, t: torch.Tensor):
        if (not self.enabled or not t.is_cuda
                or not t.is_floating_point()  # CSR indices etc stay put
                or t.numel() * t.element_size() < self.min_bytes
                or (t.is_leaf and t.requires_grad)):  # keep params resident
            return t
        cur = torch.cuda.current_stream(t.device)
        cpu = self._acquire(t)
        if cpu is None:
            self._stats["budget_skipped"] += 1
            return t  # pinned budget exhausted: stay HBM-resident
        self.stream.wait_stream(cur)          # producer finished
        with torch.cuda.stream(self.stream):
            cpu.copy_(t, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(self.stream)
        streamcheck.producer(ev, "offload-d2h")
        t.record_stream(self.stream)          # allocator: defer reuse
        self._stats["offloaded_bytes"] += t.numel() * t.element_size()
        self._stats["tensors"] += 1
        entry = {"cpu": cpu, "device": t.device, "ev": ev,
                 "gpu": None, "ev2": None}
        self._entries.append(entry)
        return ("roc_offloaded", entry)

    def prefetch(self, window: int = 0):
        """Start H2D copies for the next `window` offloaded tensors in
        LIFO order (backward consumes saved tensors roughly in reverse
        forward order); each consumption triggers the next copy. Keeps at
        most `window` prefetched activations resident — capacity stays
        bounded. Call right before loss.backward()."""
        if not self.enabled or self.prefetch_window <= 0:
            return
        window = window or self.prefetch_window
        self._order = list(reversed(self._entries))
        self._pf_ptr = 0
        for _ in range(min(window, len(self._order))):
            self._start_next()

    _order = ()
    _pf_ptr = 0

    def _start_next(self):
        while self._pf_ptr < len(self._order):
            e = self._order[self._pf_ptr]
            self._pf_ptr += 1
            if e["gpu"] is None and e["cpu"] is not None:
                self._start_h2d(e)
                return

    def _start_h2d(self, entry):
        streamcheck.consumer(entry["ev"], "offload-h2d")
        with torch.cuda.stream(self.stream):
            entry["ev"].wait(self.stream)     # D2H done before H2D
            entry["gpu"] = entry["cpu"].to(entry["device"], non_blocking=True)
            ev2 = torch.cuda.Event()
            ev2.record(self.stream)
            entry["ev2"] = ev2
        streamcheck.producer(ev2, "offload-h2d")

    def _unpack(self, packed):
        if not isinstance(packed, tuple) or not packed or \
                packed[0] != "roc_offloaded":
            return packed
        entry = packed[1]
        if entry["cpu"] is None:
            raise RuntimeError(
                "offloaded activation already consumed (double backward is "
                "not supported with ActivationOffload)")
        if entry["gpu"] is None:
            self._start_h2d(entry)
        streamcheck.consumer(entry["ev2"], "backward-consume")
        cur = torch.cuda.current_stream(entry["device"])
        cur.wait_event(entry["ev2"])
        gpu = entry["gpu"]
        gpu.record_stream(cur)
        self._release(entry["cpu"])
        entry["cpu"] = None
        entry["gpu"] = None
        # identity-based removal: dict equality would compare the tensors
        # held by OTHER entries elementwise (ambiguous-bool RuntimeError)
        self._entries = [e for e in self._entries if e is not entry]
        self._start_next()                    # keep the pipeline full
        return gpu

    def __enter__(self):
        self._ctx = torch.autograd.graph.saved_tensors_hooks(
            self._pack, self._unpack)
        self._ctx.__enter__()
        return self

    def __exit__(self, *a):
        self._ctx.__exit__(*a)
        self._ctx = None
        return False

    @property
    def stats(self):
        return dict(self._stats)
