"""torch.autograd.gradcheck over the CPU op path on RANDOM graphs
(hypothesis): verifies every hand-written backward formula against
finite differences — including empty rows, hubs and multi-edges the
fixed-shape numerics tests never draw."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from roc_amd.graph import synthetic_graph
from roc_amd.parallel.partition import build_shard
from roc_amd.ops import functional as F


@st.composite
def shard_and_dim(draw):
    n = draw(st.integers(min_value=2, max_value=40))
    e = draw(st.integers(min_value=0, max_value=5 * n))
    seed = draw(st.integers(min_value=0, max_value=2**31 - 1))
    g = synthetic_graph(n, max(e, n), seed=seed,
                        add_self_edges=draw(st.booleans()))
    d = draw(st.integers(min_value=1, max_value=6))
    return build_shard(g, 0, 1), d


@settings(max_examples=15, deadline=None, derandomize=True)
@given(shard_and_dim())
def test_scatter_gather_gradcheck(sd):
    shard, d = sd
    x = torch.randn(shard.n_local, d, dtype=torch.float64,
                    requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: F.scatter_gather(t, shard), (x,), eps=1e-6, atol=1e-5)


@settings(max_examples=15, deadline=None, derandomize=True)
@given(shard_and_dim())
def test_scatter_gather_normalized_gradcheck(sd):
    shard, d = sd
    x = torch.randn(shard.n_local, d, dtype=torch.float64,
                    requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: F.scatter_gather(
            t, shard, dst_scale=shard.rsqrt_deg_local.double()),
        (x,), eps=1e-6, atol=1e-5)


@settings(max_examples=15, deadline=None, derandomize=True)
@given(shard_and_dim())
def test_indegree_norm_gradcheck(sd):
    shard, d = sd
    x = torch.randn(shard.n_local, d, dtype=torch.float64,
                    requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: F.indegree_norm(t, shard), (x,), eps=1e-6, atol=1e-5)


@settings(max_examples=10, deadline=None, derandomize=True)
@given(st.integers(1, 20), st.integers(1, 8), st.integers(1, 8),
       st.integers(0, 2**31 - 1))
def test_linear_gradcheck(rows, din, dout, seed):
    torch.manual_seed(seed)
    x = torch.randn(rows, din, dtype=torch.float64, requires_grad=True)
    w = torch.randn(din, dout, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda a, b: F.linear(a, b), (x, w), eps=1e-6, atol=1e-5)
    rs = torch.rand(rows, dtype=torch.float64) + 0.5
    assert torch.autograd.gradcheck(
        lambda a, b: F.linear(a, b, row_scale=rs), (x, w),
        eps=1e-6, atol=1e-5)


@settings(max_examples=10, deadline=None, derandomize=True)
@given(st.integers(1, 30), st.integers(1, 6), st.integers(0, 2**31 - 1))
def test_elementwise_gradcheck(rows, d, seed):
    torch.manual_seed(seed)
    a = torch.randn(rows, d, dtype=torch.float64, requires_grad=True)
    b = torch.randn(rows, d, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(lambda u, v: F.add(u, v), (a, b))
    assert torch.autograd.gradcheck(lambda u, v: F.mul(u, v), (a, b))
    # relu at exactly 0 is non-differentiable; keep inputs away from it
    c = (torch.randn(rows, d, dtype=torch.float64) + 0.0)
    c = torch.where(c.abs() < 1e-3, torch.full_like(c, 0.5), c)
    c.requires_grad_(True)
    assert torch.autograd.gradcheck(F.relu, (c,))
    assert torch.autograd.gradcheck(F.sigmoid, (a,))


@settings(max_examples=10, deadline=None, derandomize=True)
@given(st.integers(2, 30), st.integers(2, 7), st.integers(0, 2**31 - 1))
def test_softmax_ce_gradcheck(rows, classes, seed):
    # mask-zeroed CE grad (softmax - onehot on Train rows only) vs finite
    # differences of the returned loss
    torch.manual_seed(seed)
    rng = np.random.default_rng(seed)
    labels = torch.from_numpy(rng.integers(0, classes, rows))
    mask = torch.from_numpy(
        rng.choice([0, 1, 2, 3], size=rows).astype(np.int32))
    if (mask == 1).sum() == 0:
        mask[0] = 1  # at least one Train row so the loss is non-trivial
    logits = torch.randn(rows, classes, dtype=torch.float64,
                         requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: F.softmax_cross_entropy(t, labels, mask, 1.0)[0],
        (logits,), eps=1e-6, atol=1e-5)


@settings(max_examples=10, deadline=None, derandomize=True)
@given(st.integers(1, 30), st.integers(1, 6), st.integers(0, 2**31 - 1))
def test_dropout_gradcheck(rows, d, seed):
    # fixed call_id -> fixed Philox mask -> differentiable given the mask
    F.set_dropout_seed(seed)
    x = torch.randn(rows, d, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda t: F.dropout(t, 0.4, True, call_id=3), (x,),
        eps=1e-6, atol=1e-5)
