"""Property-based invariants for the graph machinery (hypothesis):
random CSR graphs -> transpose/reorder/ordering utilities must hold
their structural contracts for ANY input, not just the fixture shapes."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from roc_amd.graph import (CSRGraph, build_transpose, reorder_graph,
                           rcm_order, cluster_order, degree_order,
                           synthetic_graph)


@st.composite
def graphs(draw):
    n = draw(st.integers(min_value=1, max_value=120))
    e = draw(st.integers(min_value=0, max_value=6 * n))
    seed = draw(st.integers(min_value=0, max_value=2**31 - 1))
    self_edges = draw(st.booleans())
    loc = draw(st.sampled_from([0.0, 0.5, 0.9]))
    return synthetic_graph(n, max(e, n if self_edges else 0), seed=seed,
                           add_self_edges=self_edges, locality=loc,
                           num_communities=draw(st.integers(1, 8)))


def _degree_multiset(g):
    return sorted(np.diff(g.rowptr.numpy()).tolist())


@settings(max_examples=40, deadline=None, derandomize=True)
@given(graphs())
def test_transpose_involution(g):
    # (A^T)^T == A as a multigraph: edge multiset preserved
    trp, tci = build_transpose(g.num_nodes, g.rowptr, g.colidx)
    trp2, tci2 = build_transpose(g.num_nodes, trp, tci)
    rows = np.repeat(np.arange(g.num_nodes), np.diff(g.rowptr.numpy()))
    e1 = sorted(zip(rows.tolist(), g.colidx.numpy().tolist()))
    rows2 = np.repeat(np.arange(g.num_nodes), np.diff(trp2.numpy()))
    e2 = sorted(zip(rows2.tolist(), tci2.numpy().tolist()))
    assert e1 == e2
    assert int(trp[-1]) == g.num_edges


@settings(max_examples=40, deadline=None, derandomize=True)
@given(graphs())
def test_orderings_are_permutations_preserving_structure(g):
    for fn in (degree_order, rcm_order, cluster_order):
        perm = fn(g)
        assert sorted(perm.tolist()) == list(range(g.num_nodes)), fn.__name__
        g2 = reorder_graph(g, perm)
        assert g2.num_edges == g.num_edges
        # degree multiset invariant under relabeling
        assert _degree_multiset(g2) == _degree_multiset(g), fn.__name__
        # spot-check edge preservation: edge (u, v) <-> (pos(u), pos(v))
        inv = np.empty(g.num_nodes, dtype=np.int64)
        inv[perm.numpy()] = np.arange(g.num_nodes)
        rows = np.repeat(np.arange(g.num_nodes), np.diff(g.rowptr.numpy()))
        e1 = sorted(zip(inv[rows].tolist(),
                        inv[g.colidx.numpy()].tolist()))
        rows2 = np.repeat(np.arange(g.num_nodes), np.diff(g2.rowptr.numpy()))
        e2 = sorted(zip(rows2.tolist(), g2.colidx.numpy().tolist()))
        assert e1 == e2, fn.__name__


@settings(max_examples=25, deadline=None, derandomize=True)
@given(st.integers(2, 60), st.integers(0, 2**31 - 1), st.integers(1, 6))
def test_edge_balanced_bounds_cover(n, seed, parts):
    from roc_amd.parallel.partition import edge_balanced_bounds
    g = synthetic_graph(n, 4 * n, seed=seed)
    b = edge_balanced_bounds(g.rowptr, parts)
    assert b[0] == 0 and b[-1] == g.num_nodes
    assert all(b[i] <= b[i + 1] for i in range(len(b) - 1))


@settings(max_examples=25, deadline=None, derandomize=True)
@given(graphs(), st.integers(1, 5))
def test_halo_plan_invariants(g, world):
    """Structural contracts of the per-rank halo plan for ANY graph and
    world size (no process group: the full-graph scan path)."""
    import os
    from roc_amd.parallel.partition import build_shard, edge_balanced_bounds
    world = min(world, g.num_nodes)
    bounds = edge_balanced_bounds(g.rowptr, world)
    os.environ["ROC_COMM_MODE"] = "halo"  # allgather has no send plan
    try:
        shards = [build_shard(g, r, world, bounds) for r in range(world)]
    finally:
        del os.environ["ROC_COMM_MODE"]
    assert bounds[0] == 0 and bounds[-1] == g.num_nodes

    for sh in shards:
        # halo ids are remote, unique, grouped by owner rank
        halo = sh.halo_ids.numpy()
        assert len(np.unique(halo)) == len(halo)
        assert not ((halo >= sh.lo) & (halo < sh.hi)).any()
        owners = np.searchsorted(np.asarray(bounds), halo, side="right") - 1
        assert (np.diff(owners) >= 0).all()  # owner-major grouping
        assert sum(sh.recv_splits) == sh.n_halo
        # local CSR columns live in the ext space; transpose in local space
        if sh.colidx.numel():
            assert int(sh.colidx.max()) < sh.n_ext
        if sh.t_colidx.numel():
            assert int(sh.t_colidx.max()) < sh.n_local
        # every local row's edge count matches the global CSR
        rp = g.rowptr.numpy()
        assert np.array_equal(np.diff(sh.rowptr.numpy()),
                              np.diff(rp)[sh.lo:sh.hi])

    # send plans mirror recv plans: what rank r sends to s is exactly
    # the slice of s's halo owned by r
    for r, sh_r in enumerate(shards):
        off = 0
        for s, cnt in enumerate(sh_r.send_splits):
            sent = sh_r.send_idx[off:off + cnt].numpy() + sh_r.lo
            off += cnt
            sh_s = shards[s]
            ho = 0
            want = np.empty(0, dtype=np.int64)
            for o, rc in enumerate(sh_s.recv_splits):
                if o == r:
                    want = sh_s.halo_ids[ho:ho + rc].numpy()
                ho += rc
            assert np.array_equal(np.sort(sent), np.sort(want)), (r, s)


@settings(max_examples=40, deadline=None, derandomize=True)
@given(graphs(), st.integers(min_value=1, max_value=12),
       st.integers(min_value=0, max_value=2**31 - 1))
def test_native_sampler_invariants(g, fanout, seed):
    """sample_hop contracts for ANY graph/fanout/seed: degrees bounded
    by min(fanout, true degree), every sampled edge exists in the
    graph, targets form the src-id prefix, ids in range."""
    from roc_amd import _C
    rng = np.random.default_rng(seed)
    k = min(g.num_nodes, 1 + seed % 16)
    targets = torch.from_numpy(
        rng.choice(g.num_nodes, size=k, replace=False).astype(np.int64))
    rp_s, ci_s, src = _C.sample_hop(g.rowptr, g.colidx, targets,
                                    fanout, seed)
    assert rp_s.numel() == k + 1 and int(rp_s[0]) == 0
    assert torch.equal(src[:k], targets)
    assert src.unique().numel() == src.numel()  # local ids are distinct
    rp = g.rowptr.numpy()
    ci = g.colidx.numpy()
    for i in range(k):
        v = int(targets[i])
        deg_s = int(rp_s[i + 1] - rp_s[i])
        deg_true = int(rp[v + 1] - rp[v])
        assert deg_s == min(fanout, deg_true)
        neigh = ci[rp[v]:rp[v + 1]].tolist()
        picked = [int(src[int(ci_s[e])])
                  for e in range(int(rp_s[i]), int(rp_s[i + 1]))]
        for u in picked:
            assert u in neigh
        if deg_true > fanout:  # sampled WITHOUT replacement
            assert len(set(picked)) == len(picked) or \
                len(set(neigh)) < len(neigh)  # unless graph has dup edges
