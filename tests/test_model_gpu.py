"""End-to-end GPU training on the HIP kernel path (bf16)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)


def make_trainer(name="gcn", scale=0.3, hidden=64, dtype=torch.bfloat16,
                 lr=0.01, dropout=0.2, **mkw):
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=scale, seed=2)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    shard = build_shard(g, 0, 1)
    dims = [feats.shape[1], hidden, c]
    model = build_model(name, dims, dropout=dropout, seed=1, **mkw)
    opt = AdamOptimizer(model.parameters(), lr=lr, weight_decay=1e-4)
    gs = 1.0 / max(int((mask == 1).sum()), 1)
    return Trainer(model, shard, feats, labels, mask, opt, device="cuda:0",
                   compute_dtype=dtype, grad_scale=gs)


@pytest.mark.parametrize("name", ["gcn", "sage", "gin", "sgc", "appnp"])
def test_model_trains_gpu(name):
    tr = make_trainer(name)
    m0 = tr.evaluate()
    for _ in range(30):
        tr.train_epoch()
    m1 = tr.evaluate()
    assert m1["ce_loss"] == m1["ce_loss"], "NaN loss"
    assert m1["ce_loss"] < m0["ce_loss"], (m0, m1)


def test_gcn_gpu_matches_cpu_reference():
    """One full epoch (no dropout) on GPU bf16 vs CPU fp32: loss and
    metrics must agree to bf16 tolerance."""
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=3)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    shard = build_shard(g, 0, 1)
    dims = [feats.shape[1], 32, c]

    results = {}
    for dev, dtype in (("cpu", torch.float32), ("cuda:0", torch.bfloat16)):
        model = build_model("gcn", dims, dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, shard, feats, labels, mask, opt, device=dev,
                     compute_dtype=dtype)
        tr.train_epoch()
        results[dev] = (tr.evaluate(),
                        model.weights[0].detach().cpu().clone())
    m_cpu, w_cpu = results["cpu"]
    m_gpu, w_gpu = results["cuda:0"]
    assert m_gpu["ce_loss"] == pytest.approx(m_cpu["ce_loss"], rel=0.05)
    assert m_gpu["train_total"] == m_cpu["train_total"]
    assert torch.allclose(w_gpu, w_cpu, atol=0.05), \
        (w_gpu - w_cpu).abs().max()


def test_gcn_overfits_gpu():
    tr = make_trainer("gcn", scale=0.1, hidden=64, lr=0.05, dropout=0.0)
    for _ in range(150):
        tr.train_epoch()
    m = tr.evaluate()
    assert m["train_acc"] > 0.85, m


def test_gcn_fp32_on_gpu_matches_cpu():
    """The reference's fp32-only mode, on GPU: must track the CPU fp32
    path tightly (exact-fp32 MFMA + fp32 gathers)."""
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.15, seed=9)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    shard = build_shard(g, 0, 1)
    dims = [feats.shape[1], 32, c]
    res = {}
    for dev in ("cpu", "cuda:0"):
        model = build_model("gcn", dims, dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, shard, feats, labels, mask, opt, device=dev,
                     compute_dtype=torch.float32)
        for _ in range(3):
            tr.train_epoch()
        res[dev] = (tr.evaluate(), model.weights[0].detach().cpu())
    m_c, w_c = res["cpu"]
    m_g, w_g = res["cuda:0"]
    assert torch.allclose(w_c, w_g, atol=1e-3), (w_c - w_g).abs().max()
    assert m_g["ce_loss"] == pytest.approx(m_c["ce_loss"], rel=1e-3)


def test_eval_identity_dropout_gpu():
    tr = make_trainer("gcn", dropout=0.5)
    tr.model.eval()
    x = tr.x
    y1 = tr.model(x, tr.shard)
    y2 = tr.model(x, tr.shard)
    assert torch.equal(y1, y2)  # dropout must be identity at infer


def test_full_stack_convergence_gpu():
    """Convergence oracle on the bf16 HIP path: a realizable teacher must
    be recovered to high held-out accuracy."""
    g, feats, labels, mask, c = synthetic_dataset(
        "reddit", scale=0.05, seed=1, learnable_labels=True)
    feats = torch.nn.functional.pad(feats, (0, (-feats.shape[1]) % 8))
    shard = build_shard(g, 0, 1)
    model = build_model("gcn", [feats.shape[1], 256, c + (-c) % 64],
                        dropout=0.2, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    gs = 1.0 / max(int((mask == 1).sum()), 1)
    tr = Trainer(model, shard, feats, labels, mask, opt, device="cuda:0",
                 compute_dtype=torch.bfloat16, grad_scale=gs, num_classes=c)
    for _ in range(60):
        tr.train_epoch()
    m = tr.evaluate()
    assert m["train_acc"] > 0.8, m
    assert m["val_acc"] > 0.7, m


def test_recompute_matches_standard_gpu():
    """The r17 capacity path on the HIP kernels: per-layer recompute must
    reproduce the standard path's loss and gradients (bf16, dropout
    active — the masks regenerate from per-layer Philox call_ids, and
    the HIP dropout backward returns the full grad arity).

    Tolerances are set to separate bug from noise, not to assert
    bit-equality: the split-K dW GEMM accumulates through fp32 ATOMICS,
    so even two identical standard runs differ by reduction-order noise
    (~1e-7 relative). A wrong dropout mask in the recomputed forward
    (p=0.4) would shift gradients by O(1) — far above these bounds."""

    def run(recompute):
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.3,
                                                      seed=2)
        pad = (-feats.shape[1]) % 8
        if pad:
            feats = torch.nn.functional.pad(feats, (0, pad))
        shard = build_shard(g, 0, 1)
        model = build_model("gcn", [feats.shape[1], 64, 64, c],
                            dropout=0.4, seed=1)
        model.recompute = recompute
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, shard, feats, labels, mask, opt,
                     device="cuda:0", compute_dtype=torch.bfloat16)
        metrics = tr.train_epoch().cpu()  # one epoch: same weights in
        grads = [p.grad.detach().float().cpu().clone()
                 for p in model.parameters() if p.grad is not None]
        return metrics, grads

    m0, g0 = run(False)
    m1, g1 = run(True)
    assert torch.allclose(m0, m1, rtol=1e-3, atol=1e-2), (m0, m1)
    assert len(g0) == len(g1) and len(g0) > 0
    for a, b in zip(g0, g1):
        assert torch.allclose(a, b, rtol=2e-2, atol=1e-4), \
            (a - b).abs().max()


def test_async_dw_matches_sync():
    """Side-stream dW accumulation (ROC_ASYNC_DW) must produce the same
    weights as the in-line synchronous dW path."""
    import roc_amd.ops.functional as Fn
    from roc_amd.graph import synthetic_dataset
    from roc_amd import build_shard, build_model, AdamOptimizer, Trainer

    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=4)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))

    def run(flag):
        old = Fn._ASYNC_DW
        Fn._ASYNC_DW = flag
        try:
            torch.manual_seed(0)
            sh = build_shard(g, 0, 1)
            model = build_model("gcn", [feats.shape[1], 32, c],
                                dropout=0.4, seed=1)
            opt = AdamOptimizer(model.parameters(), lr=0.01,
                                weight_decay=1e-4)
            tr = Trainer(model, sh, feats, labels, mask, opt,
                         device="cuda:0", compute_dtype=torch.bfloat16,
                         seed=3)
            for _ in range(3):
                tr.train_epoch()
            torch.cuda.synchronize()
            return [p.detach().float().cpu() for p in model.parameters()]
        finally:
            Fn._ASYNC_DW = old

    ws_sync = run(False)
    ws_async = run(True)
    assert not Fn._DW_PENDING  # every epoch flushed its events
    for a, b in zip(ws_sync, ws_async):
        # fp32 atomic accumulation order differs run-to-run either way;
        # async must stay inside that noise floor
        assert torch.allclose(a, b, atol=2e-3), (a - b).abs().max()


@pytest.mark.parametrize("name", ["sgc", "appnp"])
def test_new_family_gpu_matches_cpu(name):
    """SGC / APPNP: one epoch GPU bf16 vs CPU fp32 (no dropout) — the
    precomputed propagation (SGC) and the K-hop differentiable
    propagation (APPNP) must agree across backends."""
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=3)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    shard = build_shard(g, 0, 1)
    dims = [feats.shape[1], 32, c]

    results = {}
    for dev, dtype in (("cpu", torch.float32), ("cuda:0", torch.bfloat16)):
        model = build_model(name, dims, dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, shard, feats, labels, mask, opt, device=dev,
                     compute_dtype=dtype)
        tr.train_epoch()
        results[dev] = (tr.evaluate(),
                        next(model.parameters()).detach().cpu().clone())
    m_cpu, w_cpu = results["cpu"]
    m_gpu, w_gpu = results["cuda:0"]
    assert m_gpu["ce_loss"] == pytest.approx(m_cpu["ce_loss"], rel=0.07)
    assert m_gpu["train_total"] == m_cpu["train_total"]
    assert torch.allclose(w_gpu, w_cpu, atol=0.05), \
        (w_gpu - w_cpu).abs().max()


DEV = "cuda:0"


def test_minibatch_sampled_gpu():
    """Sampled MFG blocks run the same HIP kernels: one sampled epoch
    on GPU bf16 trains, and the full-fanout block forward matches the
    full-graph forward (bf16 tolerance)."""
    import numpy as np
    from roc_amd.sampling import MiniBatchTrainer, sample_blocks
    g, feats, labels, mask, c = synthetic_dataset(
        "cora", scale=0.2, seed=7, learnable_labels=True)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    model = build_model("sage", [feats.shape[1], 32, c], dropout=0.0,
                        seed=2)
    # exactness: full fanout == full-graph forward on the batch rows
    sh = build_shard(g, 0, 1).to(DEV)
    model_gpu = model.to(DEV)
    model_gpu.eval()
    xf = feats.to(DEV).to(torch.bfloat16)
    full = model_gpu(xf, sh)
    max_deg = int((g.rowptr[1:] - g.rowptr[:-1]).max())
    targets = np.arange(0, g.num_nodes, 11)
    blocks = [b.to(DEV) for b in sample_blocks(g, targets,
                                               [max_deg, max_deg])]
    out = model_gpu.forward_blocks(xf[blocks[0].src_ids], blocks)
    want = full[torch.from_numpy(targets).to(DEV)]
    tol = want.float().abs().max().item() * 2 ** -6 + 1e-2
    assert torch.allclose(out.float(), want.float(), atol=tol, rtol=0.05), \
        (out.float() - want.float()).abs().max()
    # sampled training step on GPU decreases batch loss over epochs
    opt = AdamOptimizer(model.parameters(), lr=0.02, weight_decay=1e-4)
    tr = MiniBatchTrainer(model, g, feats, labels, mask, opt,
                          fanouts=[10, 10], batch_size=256, device=DEV,
                          compute_dtype=torch.bfloat16, seed=5)
    first = tr.train_epoch()
    for _ in range(4):
        last = tr.train_epoch()
    assert last < first, (first, last)
