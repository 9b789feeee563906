import numpy as np
import pytest
import torch

from roc_amd.graph import (CSRGraph, load_lux, save_lux, synthetic_graph,
                           synthetic_dataset, build_transpose)


def test_synthetic_graph_valid():
    g = synthetic_graph(100, 1000, seed=3)
    assert g.num_nodes == 100
    assert g.rowptr[-1].item() == g.num_edges
    assert g.colidx.max().item() < 100
    assert g.colidx.min().item() >= 0
    # self edges present
    rp = g.rowptr.numpy()
    ci = g.colidx.numpy()
    for v in range(100):
        assert v in ci[rp[v]:rp[v + 1]]
    # columns sorted within each row
    for v in range(100):
        row = ci[rp[v]:rp[v + 1]]
        assert np.all(np.diff(row) >= 0)


def test_lux_roundtrip(tmp_path):
    g = synthetic_graph(64, 500, seed=7)
    p = str(tmp_path / "g.add_self_edge.lux")
    save_lux(p, g)
    g2 = load_lux(p)
    assert g2.num_nodes == g.num_nodes
    assert g2.num_edges == g.num_edges
    assert torch.equal(g2.rowptr, g.rowptr)
    assert torch.equal(g2.colidx, g.colidx)


def test_transpose_correct():
    g = synthetic_graph(50, 400, seed=11)
    t_rowptr, t_colidx = build_transpose(50, g.rowptr, g.colidx)
    # dense check: A[v][u] = count of edge u->v in CSR
    A = np.zeros((50, 50))
    rp, ci = g.rowptr.numpy(), g.colidx.numpy()
    for v in range(50):
        for e in range(rp[v], rp[v + 1]):
            A[v, ci[e]] += 1
    At = np.zeros((50, 50))
    trp, tci = t_rowptr.numpy(), t_colidx.numpy()
    for u in range(50):
        for e in range(trp[u], trp[u + 1]):
            At[u, tci[e]] += 1
    assert np.array_equal(At, A.T)


def test_synthetic_dataset_shapes():
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.1)
    assert feats.shape[0] == g.num_nodes
    assert labels.shape[0] == g.num_nodes
    assert mask.shape[0] == g.num_nodes
    assert labels.max().item() < c


def test_indegree_clamped():
    g = synthetic_graph(30, 100, seed=5, add_self_edges=False)
    deg = g.indegree()
    assert (deg >= 1.0).all()


def test_community_locality():
    g = synthetic_graph(1000, 30000, seed=2, locality=0.9,
                        num_communities=10)
    rp, ci = g.rowptr.numpy(), g.colidx.numpy()
    row = np.repeat(np.arange(1000), np.diff(rp))
    same_block = (row // 100) == (ci // 100)
    # ~90% locality + self edges + 10% random
    assert same_block.mean() > 0.85
    # halo_fraction counts UNIQUE remote coverage: on a SPARSE community
    # graph the cut is small -> halo mode; dense/uniform -> allgather
    from roc_amd.parallel.partition import build_shard
    gs = synthetic_graph(20000, 100000, seed=2, locality=0.9,
                         num_communities=20)
    sh = build_shard(gs, 0, 2)
    assert sh.halo_fraction < 0.5 and sh.comm_mode == "halo", \
        (sh.halo_fraction, sh.comm_mode)
    g2 = synthetic_graph(1000, 30000, seed=2)  # dense uniform control
    sh2 = build_shard(g2, 0, 2)
    assert sh2.comm_mode == "allgather"


def test_label_and_mask_text_loaders(tmp_path):
    """Reference on-disk formats: `.label` one class id per line
    (`load_task.cu:110-123`), `.mask` Train|Val|Test|None strings
    (`load_task.cu:160-183`)."""
    from roc_amd.graph import (load_labels, load_mask, MASK_TRAIN, MASK_VAL,
                               MASK_TEST, MASK_NONE)
    lp = tmp_path / "d.label"
    lp.write_text("3\n0\n7\n1\n")
    labels = load_labels(str(lp), 4)
    assert labels.dtype == torch.int64
    assert labels.tolist() == [3, 0, 7, 1]
    mp = tmp_path / "d.mask"
    mp.write_text("Train\nVal\nTest\nNone\n")
    mask = load_mask(str(mp), 4)
    assert mask.tolist() == [MASK_TRAIN, MASK_VAL, MASK_TEST, MASK_NONE]
    # row-count mismatch must fail loudly, not truncate
    with pytest.raises(AssertionError):
        load_labels(str(lp), 5)
    with pytest.raises(AssertionError):
        load_mask(str(mp), 3)


def test_reorder_graph_preserves_structure():
    from roc_amd.graph import reorder_graph, degree_order
    from roc_amd.parallel.partition import build_shard
    from roc_amd.ops import functional as F
    g = synthetic_graph(80, 600, seed=9)
    perm = degree_order(g)
    g2 = reorder_graph(g, perm)
    x = torch.randn(80, 6)
    out1 = F.scatter_gather(x, build_shard(g, 0, 1))
    out2 = F.scatter_gather(x[perm], build_shard(g2, 0, 1))
    # aggregation commutes with relabeling
    assert torch.allclose(out2, out1[perm], atol=1e-5)


def test_rcm_order_path_bandwidth():
    # a label-shuffled path graph must come back to bandwidth <= 2
    from roc_amd.graph import rcm_order, reorder_graph
    n = 300
    rng = np.random.default_rng(0)
    shuf = rng.permutation(n)
    rows = [[] for _ in range(n)]
    for i in range(n - 1):
        rows[shuf[i + 1]].append(int(shuf[i]))
    rowptr = np.zeros(n + 1, dtype=np.int64)
    cols = []
    for v in range(n):
        rowptr[v + 1] = rowptr[v] + len(rows[v])
        cols.extend(sorted(rows[v]))
    g = CSRGraph(num_nodes=n, num_edges=len(cols),
                 rowptr=torch.from_numpy(rowptr),
                 colidx=torch.tensor(cols, dtype=torch.int32))
    perm = rcm_order(g)
    assert sorted(perm.tolist()) == list(range(n))
    g2 = reorder_graph(g, perm)
    rp, ci = g2.rowptr.numpy(), g2.colidx.numpy()
    r = np.repeat(np.arange(n), np.diff(rp))
    assert np.abs(r - ci).max() <= 2


def _window_frac(g, w=8192):
    rp, ci = g.rowptr.numpy(), g.colidx.numpy()
    rows = np.repeat(np.arange(g.num_nodes), np.diff(rp))
    return float((np.abs(rows - ci) < w // 2).mean())


def test_cluster_order_recovers_communities():
    # shuffle a community graph; LP clustering must recover most of the
    # gather locality the shuffle destroyed
    from roc_amd.graph import cluster_order, reorder_graph
    rng = np.random.default_rng(1)
    n = 20000
    g = synthetic_graph(n, 200_000, seed=3, locality=0.85,
                        num_communities=20)
    gshuf = reorder_graph(g, torch.from_numpy(rng.permutation(n)))
    perm = cluster_order(gshuf)
    assert sorted(perm.tolist()) == list(range(n))
    glp = reorder_graph(gshuf, perm)
    nat, shuf, rec = _window_frac(g), _window_frac(gshuf), _window_frac(glp)
    assert shuf < 0.5 * nat          # shuffle really destroyed locality
    assert rec > shuf + 0.5 * (nat - shuf)  # LP recovered >half the gap


def test_apply_ordering_training_equivalent():
    # full-graph training is permutation-equivariant: same loss trajectory
    # (fp32, no dropout) on the relabeled dataset
    from roc_amd.graph import apply_ordering, synthetic_dataset
    from roc_amd import build_shard, build_model, AdamOptimizer, Trainer
    from roc_amd.parallel.partition import edge_balanced_bounds

    def run(g, feats, labels, mask, c):
        torch.manual_seed(0)
        shard = build_shard(g, 0, 1, edge_balanced_bounds(g.rowptr, 1))
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01)
        tr = Trainer(model, shard, feats, labels, mask, opt)
        return [tr.train_epoch() for _ in range(3)]

    g, feats, labels, mask, c = synthetic_dataset("cora", seed=4)
    base = run(g, feats, labels, mask, c)
    for kind in ("degree", "rcm", "cluster"):
        g2, f2, l2, m2, perm = apply_ordering(g, feats, labels, mask, kind)
        assert sorted(perm.tolist()) == list(range(g.num_nodes))
        out = run(g2, f2, l2, m2, c)
        for a, b in zip(base, out):  # metrics[1] = CE loss sum
            la, lb = float(a[1]), float(b[1])
            assert abs(la - lb) < 1e-3 * max(1.0, abs(la))


def test_cluster_order_shrinks_halo():
    # the multi-GPU payoff: on a community graph with scrambled ids,
    # cluster reordering makes contiguous-range partitions cut far fewer
    # edges, so the 2-way halo shrinks
    from roc_amd.graph import cluster_order, reorder_graph
    from roc_amd.parallel.partition import build_shard, edge_balanced_bounds
    rng = np.random.default_rng(5)
    n = 20000
    g = synthetic_graph(n, 200_000, seed=7, locality=0.9,
                        num_communities=16)
    gshuf = reorder_graph(g, torch.from_numpy(rng.permutation(n)))
    glp = reorder_graph(gshuf, cluster_order(gshuf))

    def halo_rows(gg):
        tot = 0
        bounds = edge_balanced_bounds(gg.rowptr, 2)
        for r in range(2):
            tot += build_shard(gg, r, 2, bounds).n_halo
        return tot

    h_shuf, h_lp = halo_rows(gshuf), halo_rows(glp)
    assert h_lp < 0.6 * h_shuf, (h_shuf, h_lp)


def test_apply_ordering_auto():
    # auto: reorders a scrambled community graph, leaves a uniform graph
    # (and an already-ordered one) alone
    # (graph must be ≫ the ±8192-row locality window to discriminate)
    from roc_amd.graph import apply_ordering, reorder_graph
    rng = np.random.default_rng(3)
    n = 100_000
    gc = synthetic_graph(n, 1_000_000, seed=3, locality=0.85,
                         num_communities=25)
    feats = torch.randn(n, 4)
    labels = torch.zeros(n, dtype=torch.int64)
    mask = torch.ones(n, dtype=torch.int32)
    gshuf = reorder_graph(gc, torch.from_numpy(rng.permutation(n)))
    _, _, _, _, perm = apply_ordering(gshuf, feats, labels, mask, "auto")
    assert perm is not None  # scrambled community graph -> reordered
    _, _, _, _, perm = apply_ordering(gc, feats, labels, mask, "auto")
    assert perm is None      # already ordered -> kept
    gu = synthetic_graph(n, 1_000_000, seed=4)
    _, _, _, _, perm = apply_ordering(gu, feats, labels, mask, "auto")
    assert perm is None      # uniform -> kept
