"""Source-strip-blocked SpMM plan (CPU correctness; the GPU win is
measured in profiles/r21)."""
import numpy as np
import torch

from roc_amd import build_shard, synthetic_dataset
from roc_amd.ops import reference as ref
from roc_amd.parallel.partition import build_strip_plan


def test_strip_plan_partitions_edges_exactly():
    g, *_ = synthetic_dataset("cora", scale=0.05, seed=3)
    sh = build_shard(g, 0, 1)
    n = g.num_nodes
    for width in (37, 101, 1000):
        strips = build_strip_plan(sh.rowptr, sh.colidx, n, width)
        total = 0
        for s, (srp, sci) in enumerate(strips):
            assert srp.numel() == sh.rowptr.numel()
            lo, hi = s * width, min((s + 1) * width, n)
            if sci.numel():
                assert int(sci.min()) >= lo and int(sci.max()) < hi
            total += sci.numel()
        assert total == g.num_edges


def test_strip_passes_sum_to_full_aggregation():
    torch.manual_seed(3)
    g, feats, *_ = synthetic_dataset("cora", scale=0.05, seed=3)
    sh = build_shard(g, 0, 1)
    n = g.num_nodes
    full = ref.spmm(feats, sh.rowptr, sh.colidx, n)
    acc = torch.zeros_like(full)
    for srp, sci in build_strip_plan(sh.rowptr, sh.colidx, n, 64):
        acc += ref.spmm(feats, srp, sci, n)
    assert torch.allclose(acc, full, atol=1e-5)


def test_strip_autobuild_threshold(monkeypatch):
    monkeypatch.setenv("ROC_SPMM_STRIP_MIN_EDGES", "0")
    monkeypatch.setenv("ROC_SPMM_STRIP_WIDTH", "50")
    g, *_ = synthetic_dataset("cora", scale=0.05, seed=3)
    sh = build_shard(g, 0, 1)
    assert sh.fwd_strips is not None and sh.bwd_strips is not None
    assert sum(sci.numel() for _, sci in sh.fwd_strips) == g.num_edges
    assert sum(sci.numel() for _, sci in sh.bwd_strips) == g.num_edges
    # transpose strips are over the ORIGINAL row space (cols < n_local)
    moved = sh.to("cpu")  # .to must carry the strip lists
    assert moved.fwd_strips is not None
    # default thresholds: small graph -> no strips
    monkeypatch.delenv("ROC_SPMM_STRIP_MIN_EDGES")
    sh2 = build_shard(g, 0, 1)
    assert sh2.fwd_strips is None
