"""Edge-tensor capability (CPU): factory shapes, weighted aggregation
fwd/bwd vs torch autograd, equivalence of gcn_norm edge weights with
the fused symmetric-norm path, and halo-extended (ws=2-shaped) use.

The reference declares EDGE tensors (`gnn.cc:475-623`
create_edge_tensor) but ships no op over them; here the surface is
live: roc_amd.edge_tensor + ops.functional.scatter_gather_weighted.
"""
import numpy as np
import torch

from roc_amd import build_shard, edge_tensor, synthetic_dataset
from roc_amd.ops import functional as F
from roc_amd.ops import reference as ref
from roc_amd.parallel.partition import edge_balanced_bounds


def _setup(seed=5, scale=0.05):
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=scale,
                                                  seed=seed)
    sh = build_shard(g, 0, 1)
    return g, feats, sh


def test_edge_tensor_factory():
    g, feats, sh = _setup()
    e = sh.num_local_edges
    assert e == g.num_edges
    assert edge_tensor(sh).shape == (e,)
    assert edge_tensor(sh, dim=4).shape == (e, 4)
    assert edge_tensor(sh, init="ones").sum() == e
    w = edge_tensor(sh, init="gcn_norm")
    assert w.shape == (e,) and (w > 0).all()


def test_weighted_aggregation_matches_dense():
    torch.manual_seed(3)
    g, feats, sh = _setup()
    n = g.num_nodes
    w = torch.rand(sh.num_local_edges)
    x = feats.clone().requires_grad_(True)
    out = F.scatter_gather_weighted(x, w, sh)
    # dense reference: A[v,u] = sum of w over edges (v<-u)
    A = torch.zeros(n, n)
    row = torch.repeat_interleave(torch.arange(n),
                                  (sh.rowptr[1:] - sh.rowptr[:-1]).long())
    A.index_put_((row, sh.colidx.long()), w, accumulate=True)
    assert torch.allclose(out, A @ feats, atol=1e-5)


def test_weighted_aggregation_grads_vs_autograd():
    torch.manual_seed(4)
    g, feats, sh = _setup()
    w = torch.rand(sh.num_local_edges, requires_grad=True)
    x = feats.clone().double().float().requires_grad_(True)
    out = F.scatter_gather_weighted(x, w, sh,
                                    dst_scale=sh.rsqrt_deg_local)
    gy = torch.randn_like(out)
    out.backward(gy)
    # autograd reference via index ops
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    row = torch.repeat_interleave(torch.arange(g.num_nodes),
                                  (sh.rowptr[1:] - sh.rowptr[:-1]).long())
    out2 = torch.zeros_like(out)
    out2 = out2.index_add(0, row, x2[sh.colidx.long()] * w2.unsqueeze(1))
    out2 = out2 * sh.rsqrt_deg_local.unsqueeze(1)
    out2.backward(gy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4), \
        (w.grad - w2.grad).abs().max()


def test_gcn_norm_weights_equal_fused_norm_path():
    g, feats, sh = _setup()
    w = edge_tensor(sh, init="gcn_norm")
    got = F.scatter_gather_weighted(feats, w, sh)
    want = F.scatter_gather(feats, sh, normalize=True)
    assert torch.allclose(got, want, atol=1e-5), (got - want).abs().max()


def test_weighted_aggregation_halo_shard():
    """ws=2-shaped shard (halo-extended x, rectangular CSR): weighted
    aggregation must reproduce the single-rank slice with per-rank edge
    windows of a global weight vector."""
    torch.manual_seed(6)
    g, feats, _ = _setup()
    w_global = torch.rand(g.num_edges)
    full = ref.spmm_weighted(feats, g.rowptr, g.colidx, w_global,
                             g.num_nodes)
    bounds = edge_balanced_bounds(g.rowptr, 2)
    for rank in range(2):
        sh = build_shard(g, rank, 2, bounds)
        e0 = int(g.rowptr[sh.lo])
        w_loc = w_global[e0:e0 + sh.num_local_edges]
        x_ext = torch.cat([feats[sh.lo:sh.hi], feats[sh.halo_ids]]) \
            if sh.n_halo else feats[sh.lo:sh.hi]
        out = F.scatter_gather_weighted(x_ext, w_loc, sh)
        assert torch.allclose(out, full[sh.lo:sh.hi], atol=1e-5)


def test_t_edge_perm_roundtrip():
    g, feats, sh = _setup()
    p = sh.t_edge_perm()
    # permuted colidx must be sorted (the transpose groups by source)
    ci = sh.colidx.long()[p]
    assert (ci[1:] >= ci[:-1]).all()
    assert np.array_equal(np.sort(p.numpy()), np.arange(g.num_edges))
