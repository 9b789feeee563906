"""Happens-before stream/async-edge checker (roc_amd.streamcheck).

The reference prevented races statically via Legion's EXCLUSIVE region
coherence (`scattergather.cc:59-78`); this framework's explicit
event/work edges get a debug-mode validator instead.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd import streamcheck as sc


class _FakeWork:
    def __init__(self):
        self.waited = 0

    def wait(self):
        self.waited += 1


def test_registry_semantics():
    sc.enable_stream_debug(True)
    try:
        w = _FakeWork()
        sc.producer(w, "a2a")
        sc.consumer(w, "spmm")
        assert w.waited == 1               # debug mode made it blocking
        assert sc.edge_log() == [("a2a", "spmm")]
        assert sc.pending() == []
        # consuming a released edge = use-after-release
        with pytest.raises(sc.StreamOrderViolation):
            sc.consumer(w, "again")
        # undeclared producer = missing edge
        with pytest.raises(sc.StreamOrderViolation):
            sc.consumer(_FakeWork(), "orphan")
        # non-releasing consume keeps the edge live for a second reader
        w2 = _FakeWork()
        sc.producer(w2, "d2h")
        sc.consumer(w2, "h2d", release=False)
        sc.consumer(w2, "verify")
        assert ("d2h", "h2d") in sc.edge_log()
        # leaked edge shows up in pending()
        w3 = _FakeWork()
        sc.producer(w3, "leaky")
        assert sc.pending() == ["leaky"]
    finally:
        sc.enable_stream_debug(False)


def test_disabled_is_noop():
    assert not sc.stream_debug_enabled()
    w = _FakeWork()
    sc.producer(w, "x")
    sc.consumer(w, "y")     # no registration, no raise, no wait
    assert w.waited == 0
    assert sc.edge_log() == []


WS = 2


def _worker(rank, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = "halo"
        os.environ["ROC_OVERLAP"] = "1"
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        from roc_amd.graph import synthetic_dataset
        from roc_amd.parallel.partition import (build_shard,
                                                edge_balanced_bounds)
        from roc_amd import build_model, AdamOptimizer, Trainer
        sc.enable_stream_debug(True)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08,
                                                      seed=12)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        model = build_model("gcn", [feats.shape[1], 16, c],
                            dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        tr.train_epoch()
        edges = sc.edge_log()
        q.put((rank, edges, sc.pending(), None))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, None, None, traceback.format_exc()))
    finally:
        sc.enable_stream_debug(False)
        for k in ("ROC_OVERLAP", "ROC_COMM_MODE"):
            os.environ.pop(k, None)
        if dist.is_initialized():
            dist.destroy_process_group()


def test_overlap_path_declares_its_edges():
    """One epoch of the comm/compute-overlap halo path logs a validated
    a2a->compute edge per layer per direction, and leaks none."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 29571, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, edges, leaked, err in res:
        assert err is None, f"rank {rank}: {err}"
        fwd = [e for e in edges if e == ("halo-a2a-fwd", "boundary-spmm")]
        bwd = [e for e in edges if e == ("halo-a2a-bwd", "grad-scatter")]
        assert len(fwd) == 2, edges     # 2 GCN layers forward
        assert len(bwd) >= 1, edges     # layer-1 dX is skipped (input)
        assert leaked == [], leaked
