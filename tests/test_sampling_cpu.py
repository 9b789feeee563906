"""Mini-batch neighbor sampling (roc_amd.sampling): sampler invariants,
exactness at full fanout, and sampled training end-to-end."""
import numpy as np
import pytest
import torch

from roc_amd import build_model, build_shard, AdamOptimizer
from roc_amd.graph import synthetic_dataset
from roc_amd.sampling import Block, MiniBatchTrainer, sample_blocks


def _graph():
    return synthetic_dataset("cora", scale=0.2, seed=7,
                             learnable_labels=True)


def test_sampler_invariants():
    g, *_ = _graph()
    rng = np.random.default_rng(3)
    targets = rng.choice(g.num_nodes, size=64, replace=False)
    fanouts = [5, 3]
    blocks = sample_blocks(g, targets, fanouts, rng)
    assert len(blocks) == 2
    rp = g.rowptr.numpy()
    ci = g.colidx.numpy()
    for i, (blk, fo) in enumerate(zip(blocks, fanouts)):
        deg = (blk.rowptr[1:] - blk.rowptr[:-1])
        assert int(deg.max()) <= fo  # fanout bound
        # dst nodes are the src prefix
        assert blk.n_dst <= blk.n_src
        # every sampled edge exists in the real graph
        src_ids = blk.src_ids.numpy()
        for v_local in range(blk.n_dst):
            v = int(src_ids[v_local])
            neigh = set(ci[rp[v]:rp[v + 1]].tolist())
            for e in range(int(blk.rowptr[v_local]),
                           int(blk.rowptr[v_local + 1])):
                assert int(src_ids[int(blk.colidx[e])]) in neigh
    # layer chaining: inner block's dst ids ARE the outer block's srcs
    assert blocks[0].n_dst == blocks[1].n_src
    assert torch.equal(blocks[0].src_ids[:blocks[0].n_dst],
                       blocks[1].src_ids)
    assert blocks[1].n_dst == len(targets)


def test_full_fanout_matches_full_graph_forward():
    """With fanout >= max indegree nothing is dropped: the sampled
    forward must equal the full-graph forward on the batch rows."""
    g, feats, labels, mask, c = _graph()
    sh = build_shard(g, 0, 1)
    model = build_model("sage", [feats.shape[1], 16, c],
                        dropout=0.0, seed=2)
    model.eval()
    full = model(feats.float(), sh)
    max_deg = int((g.rowptr[1:] - g.rowptr[:-1]).max())
    targets = np.arange(0, g.num_nodes, 7)
    blocks = sample_blocks(g, targets, [max_deg, max_deg])
    x = feats.float()[blocks[0].src_ids]
    out = model.forward_blocks(x, blocks)
    want = full[torch.from_numpy(targets)]
    assert torch.allclose(out, want, atol=1e-4, rtol=1e-4), \
        (out - want).abs().max()


def test_minibatch_training_learns():
    g, feats, labels, mask, c = _graph()
    model = build_model("sage", [feats.shape[1], 32, c], dropout=0.1,
                        seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.02, weight_decay=1e-4)
    tr = MiniBatchTrainer(model, g, feats, labels, mask, opt,
                          fanouts=[10, 10], batch_size=128, seed=5)
    first = tr.train_epoch()
    for _ in range(7):
        last = tr.train_epoch()
    assert last < first, (first, last)
    sh = build_shard(g, 0, 1)
    md = tr.evaluate(sh)
    assert md["train_acc"] > 2.0 / c, md  # >> chance on teacher labels


def test_sample_cli(tmp_path):
    """train.py --sample end-to-end (the user-facing sampled tier)."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "train.py"),
         "--dataset", "cora-synthetic", "--scale", "0.2",
         "--learnable-labels", "--model", "sage", "--sample", "10,10",
         "--batch-size", "128", "--epochs", "4", "--eval-every", "2",
         "--lr", "0.02", "--checkpoint", str(tmp_path / "ck.pt")],
        capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-1000:]
    assert "epoch     4" in r.stdout, r.stdout[-500:]
    assert (tmp_path / "ck.pt").exists()


def test_native_sampler_matches_python_at_full_fanout():
    """At fanout >= max degree both implementations keep every edge in
    the same order -> bit-identical blocks; under sampling the native
    hop (graph_cpu.cpp) is deterministic in its seed."""
    from roc_amd import _C
    from roc_amd.sampling import _sample_hop
    g, *_ = _graph()
    targets = np.arange(0, g.num_nodes, 5, dtype=np.int64)
    max_deg = int((g.rowptr[1:] - g.rowptr[:-1]).max())
    rp_n, ci_n, src_n = _C.sample_hop(g.rowptr, g.colidx,
                                      torch.from_numpy(targets),
                                      max_deg, 123)
    rng = np.random.default_rng(0)
    # force the python path by exceeding the native fanout clamp? no —
    # call the fallback body via a fanout the clamp allows but compare
    # to an independently-built python hop:
    import roc_amd.sampling as S
    rp_p, ci_p, src_p = None, None, None
    # python reference inline (full fanout keeps all edges in order)
    rpn = g.rowptr.numpy(); cin = g.colidx.numpy()
    local = {int(v): i for i, v in enumerate(targets)}
    rows, cols, extras = [], [], []
    for v in targets:
        seg = cin[rpn[v]:rpn[v + 1]]
        rows.append(len(seg))
        for u in seg:
            u = int(u)
            j = local.get(u)
            if j is None:
                j = len(targets) + len(extras)
                local[u] = j
                extras.append(u)
            cols.append(j)
    rp_p = torch.zeros(len(targets) + 1, dtype=torch.int64)
    torch.cumsum(torch.tensor(rows, dtype=torch.int64), 0, out=rp_p[1:])
    ci_p = torch.tensor(cols, dtype=torch.int32)
    src_p = torch.from_numpy(np.concatenate(
        [targets, np.asarray(extras, dtype=np.int64)]))
    assert torch.equal(rp_n, rp_p)
    assert torch.equal(ci_n, ci_p)
    assert torch.equal(src_n, src_p)
    # determinism under real sampling: same seed -> same result
    a = _C.sample_hop(g.rowptr, g.colidx, torch.from_numpy(targets), 4, 7)
    b = _C.sample_hop(g.rowptr, g.colidx, torch.from_numpy(targets), 4, 7)
    for x, y in zip(a, b):
        assert torch.equal(x, y)
    c = _C.sample_hop(g.rowptr, g.colidx, torch.from_numpy(targets), 4, 8)
    assert not all(torch.equal(x, y) for x, y in zip(a, c))


def test_minibatch_determinism():
    """Same seed -> identical loss trajectory (native sampler streams
    are deterministic; dropout/init are seeded)."""
    def run():
        g, feats, labels, mask, c = _graph()
        torch.manual_seed(0)
        model = build_model("sage", [feats.shape[1], 16, c],
                            dropout=0.3, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.02)
        tr = MiniBatchTrainer(model, g, feats, labels, mask, opt,
                              fanouts=[6, 6], batch_size=64, seed=9)
        return [tr.train_epoch() for _ in range(3)]
    a, b = run(), run()
    assert a == b, (a, b)
