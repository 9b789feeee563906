"""Happens-before checking for the cross-stream/async edges.

The reference got race-freedom for free: Legion serializes conflicting
region accesses statically (EXCLUSIVE coherence on every region
requirement, `scattergather.cc:59-78`), so its tasks could not race by
construction. This framework schedules work on HIP streams and async
collectives instead, so the ordering guarantees live in explicit
event/work edges — and a missing edge is a silent data race.

This module is the debug-mode replacement for Legion's static check
(SURVEY.md §5 "race detection"). Every async producer REGISTERS the
sync object that orders its completion (a hipEvent or a c10d Work
handle) together with a tag; every consumer DECLARES the edge it
depends on before touching the data. With stream debug enabled:

- a consumer naming a sync object that was never registered raises
  `StreamOrderViolation` (the edge does not exist -> race);
- consuming the same edge twice raises (use-after-release);
- each declared edge is made BLOCKING (event/work synchronized on the
  host) so that any *undeclared* ordering assumption elsewhere loses
  its accidental timing cover and surfaces deterministically;
- `edge_log()` exposes the observed happens-before DAG for tests.

With stream debug off (default), `producer()`/`consumer()` are
zero-cost no-ops: the hot path keeps its purely asynchronous edges.

Instrumented edges:
- memory.ActivationOffload: producer-compute -> D2H copy (`ev`),
  D2H -> H2D prefetch, H2D -> backward consumer (`ev2`).
- parallel.halo._HaloAggregate: all_to_all work -> boundary SpMM
  (forward) and reverse all_to_all work -> index_add (backward).
"""
from __future__ import annotations

_ENABLED = False
_LIVE: dict[int, str] = {}    # id(sync_obj) -> producer tag
_LOG: list[tuple[str, str]] = []   # (producer_tag, consumer_tag)


class StreamOrderViolation(RuntimeError):
    pass


def enable_stream_debug(enable: bool = True) -> None:
    global _ENABLED
    _ENABLED = enable
    _LIVE.clear()
    del _LOG[:]


def stream_debug_enabled() -> bool:
    return _ENABLED


def producer(sync_obj, tag: str) -> None:
    """Register `sync_obj` (hipEvent / dist Work) as the completion
    marker of async work `tag`. No-op unless stream debug is on."""
    if not _ENABLED:
        return
    _LIVE[id(sync_obj)] = tag


def consumer(sync_obj, tag: str, *, release: bool = True) -> None:
    """Declare that the caller (`tag`) depends on `sync_obj` having
    completed. Validates the edge exists, logs it, and synchronizes the
    sync object on the host so undeclared orderings lose their timing
    cover. `release=False` keeps the edge live for further consumers."""
    if not _ENABLED:
        return
    key = id(sync_obj)
    if key not in _LIVE:
        raise StreamOrderViolation(
            f"consumer '{tag}' waits on an unregistered sync object "
            f"({type(sync_obj).__name__}): either the producer edge was "
            "never declared (races with whatever wrote the data) or it "
            "was already consumed (use-after-release)")
    _LOG.append((_LIVE[key], tag))
    if release:
        del _LIVE[key]
    # make the dependency blocking: events/works both expose one of these
    for meth in ("synchronize", "wait"):
        fn = getattr(sync_obj, meth, None)
        if fn is not None:
            fn()
            break


def edge_log() -> list[tuple[str, str]]:
    """Observed (producer, consumer) happens-before edges this run."""
    return list(_LOG)


def pending() -> list[str]:
    """Producer tags registered but never consumed (leaked edges)."""
    return list(_LIVE.values())
