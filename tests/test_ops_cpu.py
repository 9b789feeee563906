"""CPU reference-op correctness: each op vs a naive/analytic formulation,
plus autograd gradients vs finite differences / torch equivalents."""
import math

import numpy as np
import pytest
import torch

from roc_amd.graph import synthetic_graph, MASK_TRAIN, MASK_VAL, MASK_TEST
from roc_amd.parallel.partition import build_shard
from roc_amd.ops import functional as F
from roc_amd.ops import reference as ref


def dense_adj(g):
    A = torch.zeros(g.num_nodes, g.num_nodes)
    rp, ci = g.rowptr, g.colidx
    for v in range(g.num_nodes):
        for e in range(rp[v], rp[v + 1]):
            A[v, ci[e]] += 1.0
    return A


@pytest.fixture(scope="module")
def small():
    g = synthetic_graph(40, 300, seed=2)
    shard = build_shard(g, 0, 1)
    return g, shard


def test_spmm_matches_dense(small):
    g, shard = small
    x = torch.randn(40, 8)
    out = F.scatter_gather(x, shard)
    A = dense_adj(g)
    assert torch.allclose(out, A @ x, atol=1e-5)


def test_spmm_normalized_matches_dense(small):
    g, shard = small
    x = torch.randn(40, 8)
    out = F.scatter_gather(x, shard, normalize=True)
    A = dense_adj(g)
    d = A.sum(dim=1).clamp(min=1.0)
    ref_out = torch.diag(d.rsqrt()) @ A @ torch.diag(d.rsqrt()) @ x
    assert torch.allclose(out, ref_out, atol=1e-5)


def test_spmm_backward_exact_on_asymmetric(small):
    g, shard = small
    x = torch.randn(40, 8, requires_grad=True)
    out = F.scatter_gather(x, shard)
    gout = torch.randn_like(out)
    out.backward(gout)
    A = dense_adj(g)
    assert torch.allclose(x.grad, A.t() @ gout, atol=1e-5)


def test_degnorm(small):
    g, shard = small
    x = torch.randn(40, 8, requires_grad=True)
    y = F.indegree_norm(x, shard)
    d = (g.rowptr[1:] - g.rowptr[:-1]).float().clamp(min=1)
    assert torch.allclose(y, x / d.sqrt().unsqueeze(1), atol=1e-6)
    gy = torch.randn_like(y)
    y.backward(gy)
    assert torch.allclose(x.grad, gy / d.sqrt().unsqueeze(1), atol=1e-6)


def test_linear_fwd_bwd():
    x = torch.randn(16, 10, requires_grad=True)
    w = torch.nn.Parameter(torch.randn(10, 5))
    y = F.linear(x, w)
    assert torch.allclose(y, x @ w, atol=1e-6)
    gy = torch.randn_like(y)
    y.backward(gy)
    assert torch.allclose(w.grad, x.detach().t() @ gy, atol=1e-5)
    assert torch.allclose(x.grad, gy @ w.detach().t(), atol=1e-5)


def test_linear_fused_relu_grad():
    x = torch.randn(16, 10, requires_grad=True)
    w = torch.nn.Parameter(torch.randn(10, 5))
    y = F.linear(x, w, activation="relu")
    x2 = x.detach().clone().requires_grad_(True)
    w2 = torch.nn.Parameter(w.detach().clone())
    y2 = torch.relu(x2 @ w2)
    assert torch.allclose(y, y2, atol=1e-6)
    gy = torch.randn_like(y)
    y.backward(gy)
    y2.backward(gy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-5)


def test_activations():
    x = torch.randn(8, 4, requires_grad=True)
    y = F.relu(x)
    y.sum().backward()
    assert torch.allclose(x.grad, (x > 0).float(), atol=1e-6)
    x2 = torch.randn(8, 4, requires_grad=True)
    s = F.sigmoid(x2)
    s.sum().backward()
    sd = torch.sigmoid(x2.detach())
    assert torch.allclose(x2.grad, sd * (1 - sd), atol=1e-5)


def test_elementwise():
    a = torch.randn(6, 3, requires_grad=True)
    b = torch.randn(6, 3, requires_grad=True)
    (F.add(a, b)).sum().backward()
    assert torch.allclose(a.grad, torch.ones_like(a))
    a.grad = None
    b.grad = None
    (F.mul(a, b)).sum().backward()
    assert torch.allclose(a.grad, b.detach())
    assert torch.allclose(b.grad, a.detach())


def test_dropout_train_and_infer():
    F.set_dropout_seed(42)
    x = torch.ones(1000, 16)
    y = F.dropout(x, 0.5, training=False)
    assert torch.equal(y, x)  # identity at infer (dropout_kernel.cu:159-180)
    xr = x.clone().requires_grad_(True)
    y = F.dropout(xr, 0.5, training=True)
    kept = (y != 0).float().mean().item()
    assert 0.4 < kept < 0.6
    assert torch.allclose(y[y != 0], torch.full_like(y[y != 0], 2.0))
    # backward uses same mask
    y.sum().backward()
    assert torch.equal((xr.grad != 0), (y.detach() != 0))


def test_softmax_ce_grad_and_metrics():
    torch.manual_seed(0)
    n, c = 50, 7
    logits = torch.randn(n, c, requires_grad=True)
    labels = torch.randint(0, c, (n,))
    mask = torch.randint(1, 4, (n,), dtype=torch.int32)
    loss, metrics = F.softmax_cross_entropy(logits, labels, mask)
    loss.backward()
    p = torch.softmax(logits.detach(), dim=1)
    onehot = torch.zeros(n, c).scatter_(1, labels.unsqueeze(1), 1.0)
    expected = (p - onehot) * (mask == MASK_TRAIN).unsqueeze(1).float()
    assert torch.allclose(logits.grad, expected, atol=1e-5)
    md = F.decode_metrics(metrics)
    train = mask == MASK_TRAIN
    pt = p[torch.arange(n), labels]
    assert md["roc_loss"] == pytest.approx(float((1 - pt)[train].sum()), rel=1e-4)
    # ce loss vs torch
    ce = torch.nn.functional.cross_entropy(
        logits.detach()[train], labels[train], reduction="mean")
    assert md["ce_loss"] == pytest.approx(float(ce), rel=1e-4)
    pred = p.argmax(1)
    for name, mval in (("train", MASK_TRAIN), ("val", MASK_VAL), ("test", MASK_TEST)):
        sel = mask == mval
        if sel.any():
            acc = float((pred[sel] == labels[sel]).float().mean())
            assert md[f"{name}_acc"] == pytest.approx(acc, abs=1e-6)


def test_adam_matches_formula():
    torch.manual_seed(1)
    w = torch.randn(13)
    g = torch.randn(13)
    m = torch.zeros(13)
    v = torch.zeros(13)
    w0 = w.clone()
    lr, b1, b2, eps, wd = 0.01, 0.9, 0.999, 1e-8, 1e-4
    t = 1
    alpha = lr * math.sqrt(1 - b2 ** t) / (1 - b1 ** t)
    F.adam_step(w, g, m, v, alpha, b1, b2, eps, wd)
    gt = g + wd * w0
    me = (1 - b1) * gt
    ve = (1 - b2) * gt * gt
    we = w0 - alpha * me / (ve.sqrt() + eps)
    assert torch.allclose(w, we, atol=1e-6)
    assert torch.allclose(m, me, atol=1e-6)
    assert torch.allclose(v, ve, atol=1e-6)


def test_glorot_uniform_range():
    w = ref.glorot_uniform((100, 50), seed=3)
    s = math.sqrt(6.0 / 150)
    assert w.abs().max().item() <= s
    assert w.abs().max().item() > 0.5 * s
    assert abs(w.mean().item()) < 0.01
