"""hipGraph whole-epoch capture: replayed epochs must train like eager
ones (advancing dropout streams + Adam schedule on device)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)
from roc_amd.ops import functional as F


def _make(graph_capture, dropout=0.3, seed=11):
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=7)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    shard = build_shard(g, 0, 1)
    model = build_model("gcn", [feats.shape[1], 64, c], dropout=dropout,
                        seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4,
                        decay_rate=0.97, decay_steps=100)
    tr = Trainer(model, shard, feats, labels, mask, opt, device="cuda:0",
                 compute_dtype=torch.bfloat16, seed=seed)
    if graph_capture:
        tr.enable_graph_capture(warmup_epochs=2)
    return tr


def test_capture_matches_eager_no_dropout(monkeypatch):
    """Replayed graphs vs the SAME kernels run eagerly. The baseline also
    uses the device-side Adam schedule (warmup set huge so it never
    captures), and ROC_DETERMINISTIC pins the split-K dW reduction order,
    so the two 8-epoch trajectories must agree to float rounding (an
    8-step Adam trajectory is chaotic under any reduction-order noise)."""
    monkeypatch.setenv("ROC_DETERMINISTIC", "1")
    tr_e = _make(True, dropout=0.0)
    tr_e._graph_warmup = 10 ** 9  # device-schedule eager forever
    for _ in range(8):
        tr_e.train_epoch()
    assert tr_e._graph is None
    me = tr_e.evaluate()
    we = tr_e.model.weights[0].detach().cpu()

    tr_g = _make(True, dropout=0.0)
    for _ in range(8):
        tr_g.train_epoch()
    assert tr_g._graph is not None, "graph was never captured"
    mg = tr_g.evaluate()
    wg = tr_g.model.weights[0].detach().cpu()
    F.set_dropout_counter(None)
    assert torch.allclose(we, wg, atol=1e-5), (we - wg).abs().max()
    assert mg["ce_loss"] == pytest.approx(me["ce_loss"], rel=0.01)
    assert tr_g.optimizer.t == tr_e.optimizer.t == 8


def test_adam_device_schedule_matches_host():
    """adam_step with a device step counter must match the host-side
    alpha_t computation (optimizer.cc:79-85 semantics) tightly."""
    import math
    torch.manual_seed(3)
    n = 4096
    lr, b1, b2, eps, wd = 0.01, 0.9, 0.999, 1e-8, 1e-4
    decay_rate, decay_steps = 0.97, 100
    for t in (1, 5, 150, 999):
        w0 = torch.randn(n, device="cuda:0")
        g = torch.randn(n, device="cuda:0")
        m0 = torch.rand(n, device="cuda:0")
        v0 = torch.rand(n, device="cuda:0")
        # host schedule
        wa, ma, va = w0.clone(), m0.clone(), v0.clone()
        lr_t = lr * decay_rate ** (t // decay_steps)
        alpha = lr_t * math.sqrt(1 - b2 ** t) / (1 - b1 ** t)
        F.adam_step(wa, g, ma, va, alpha, b1, b2, eps, wd)
        # device schedule
        wb, mb, vb = w0.clone(), m0.clone(), v0.clone()
        step = torch.tensor([t], dtype=torch.int64, device="cuda:0")
        F.adam_step(wb, g, mb, vb, lr, b1, b2, eps, wd,
                    step=step, decay_rate=decay_rate, decay_steps=decay_steps)
        assert torch.allclose(wa, wb, atol=1e-6, rtol=1e-4), \
            (t, (wa - wb).abs().max())


def test_capture_dropout_advances():
    tr = _make(True, dropout=0.5)
    losses = []
    for _ in range(10):
        m = tr.train_epoch()
    # two consecutive replays must not produce identical metrics
    torch.cuda.synchronize()
    m1 = tr.train_epoch().cpu().clone()
    m2 = tr.train_epoch().cpu().clone()
    torch.cuda.synchronize()
    F.set_dropout_counter(None)
    assert not torch.equal(m1, m2), "dropout mask frozen across replays"
    assert torch.isfinite(m1).all() and torch.isfinite(m2).all()


def test_capture_trains():
    tr = _make(True, dropout=0.2)
    m0 = tr.evaluate()
    for _ in range(40):
        tr.train_epoch()
    m1 = tr.evaluate()
    F.set_dropout_counter(None)
    assert m1["ce_loss"] < m0["ce_loss"], (m0, m1)


def test_capture_replay_no_leak():
    """200 graph replays must not grow device memory (stable pool)."""
    tr = _make(True, dropout=0.3)
    for _ in range(10):
        tr.train_epoch()
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    for _ in range(200):
        tr.train_epoch()
    torch.cuda.synchronize()
    grown = torch.cuda.memory_allocated() - base
    F.set_dropout_counter(None)
    assert grown < (32 << 20), f"memory grew {grown/1e6:.1f} MB over replays"
