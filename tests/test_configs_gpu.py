"""BASELINE.json configs #3-#5 at reduced scale on one GPU: the model
family + aggregation path + (for #5) host-DRAM offload all run and train."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)


def run_config(dataset, model_name, layers, hidden, scale, offload=False,
               epochs=4):
    g, feats, labels, mask, c = synthetic_dataset(dataset, scale=scale, seed=5)
    pad = (-feats.shape[1]) % 8
    if pad:
        feats = torch.nn.functional.pad(feats, (0, pad))
    c_pad = c + ((-c) % 64)
    shard = build_shard(g, 0, 1)
    dims = [feats.shape[1]] + [hidden] * (layers - 1) + [c_pad]
    model = build_model(model_name, dims, dropout=0.3, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    gs = 1.0 / max(int((mask == 1).sum()), 1)
    tr = Trainer(model, shard, feats, labels, mask, opt, device="cuda:0",
                 compute_dtype=torch.bfloat16, grad_scale=gs, num_classes=c)
    if offload:
        tr.enable_offload(min_bytes=1 << 20)
    m0 = tr.evaluate()
    for _ in range(epochs):
        tr.train_epoch()
    m1 = tr.evaluate()
    assert m1["ce_loss"] == m1["ce_loss"], "NaN loss"
    assert m1["ce_loss"] < m0["ce_loss"] * 1.05, (m0, m1)
    return tr


def test_config3_amazon_sage():
    run_config("amazon", "sage", layers=3, hidden=256, scale=0.01)


def test_config4_products_gin():
    run_config("ogbn-products", "gin", layers=3, hidden=256, scale=0.01)


def test_config5_papers_gcn_offload():
    tr = run_config("papers-synth-small", "gcn", layers=4, hidden=128,
                    scale=0.01, offload=True)
    assert tr.offload.stats["tensors"] > 0
