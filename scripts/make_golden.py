#!/usr/bin/env python3
"""Generate the checked-in golden regression dataset + expected trajectory.

A small fixed teacher-labeled graph in the reference's on-disk formats
(`.add_self_edge.lux` + `.feats.bin` + `.label` + `.mask`,
`gnn.cc:758-801` / `load_task.cu:25-183`), with the CPU-fp32 loss and
accuracy trajectory pinned in golden_expected.json. tests/test_golden.py
retrains on these files and fails on ANY numerics drift — the committed
stand-in for the reference's tkipf/gcn-Reddit accuracy oracle
(`gnn.cc:93-94`), which needs a network we don't have.

Run from the repo root: python scripts/make_golden.py
"""
import json
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from roc_amd.graph import (synthetic_graph, _learnable_labels, save_lux,
                           MASK_TRAIN, MASK_VAL, MASK_TEST)

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "tests", "data")
PREFIX = os.path.join(OUT, "golden")

# fixed recipe — tests must reproduce this EXACTLY
N, E, D, C = 2048, 16384, 32, 7
SEED = 7
EPOCHS = 30
CHECK_EPOCHS = (10, 20, 30)


def generate():
    os.makedirs(OUT, exist_ok=True)
    g = synthetic_graph(N, E, seed=SEED, locality=0.3, num_communities=16)
    rng = np.random.default_rng(SEED + 1)
    feats = torch.from_numpy(rng.standard_normal((N, D)).astype(np.float32))
    labels = _learnable_labels(g, feats, C, SEED)
    u = rng.random(N)
    mask = np.full(N, MASK_TRAIN, dtype=np.int32)
    mask[u >= 0.70] = MASK_VAL
    mask[u >= 0.85] = MASK_TEST
    save_lux(PREFIX + ".add_self_edge.lux", g)
    feats.numpy().tofile(PREFIX + ".feats.bin")
    np.savetxt(PREFIX + ".label", labels.numpy(), fmt="%d")
    names = {1: "Train", 2: "Val", 3: "Test", 0: "None"}
    with open(PREFIX + ".mask", "w") as f:
        f.write("\n".join(names[int(v)] for v in mask) + "\n")
    return g, feats, labels, torch.from_numpy(mask)


def train_trajectory(device="cpu", dtype=torch.float32):
    """The pinned training recipe (also used by the tests)."""
    from roc_amd import build_model, AdamOptimizer, Trainer
    from roc_amd.graph import (load_lux, load_features, load_labels,
                               load_mask)
    from roc_amd.parallel.partition import build_shard
    torch.manual_seed(0)
    g = load_lux(PREFIX + ".add_self_edge.lux")
    feats = load_features(PREFIX, g.num_nodes, D)
    labels = load_labels(PREFIX + ".label", g.num_nodes)
    mask = load_mask(PREFIX + ".mask", g.num_nodes)
    sh = build_shard(g, 0, 1)
    model = build_model("gcn", [D, 32, C], dropout=0.5, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4,
                        decay_rate=0.97, decay_steps=100)
    n_train = int((mask == 1).sum())
    tr = Trainer(model, sh, feats, labels, mask, opt, device=device,
                 compute_dtype=dtype, grad_scale=1.0 / n_train, seed=1)
    traj = {}
    for _ in range(EPOCHS):
        tr.train_epoch()
        if tr.epoch in CHECK_EPOCHS:
            md = tr.evaluate()
            traj[str(tr.epoch)] = {
                "ce_loss": round(md["ce_loss"], 6),
                "train_acc": round(md["train_acc"], 6),
                "val_acc": round(md["val_acc"], 6),
            }
    return traj


def main():
    generate()
    torch.set_num_threads(1)  # deterministic reduction order
    traj = train_trajectory()
    expected = {
        "recipe": {"n": N, "e": E, "d": D, "c": C, "seed": SEED,
                   "epochs": EPOCHS, "model": "gcn-32", "dropout": 0.5,
                   "lr": 0.01, "wd": 1e-4, "torch_seed": 0},
        "cpu_fp32": traj,
        # filled in after the first GPU run (bf16 is looser: different
        # dropout stream + rounding; thresholds, not point values)
        "gpu_bf16_min_train_acc": None,
        "gpu_bf16_max_ce_loss": None,
    }
    with open(PREFIX + "_expected.json", "w") as f:
        json.dump(expected, f, indent=1)
    print(json.dumps(traj, indent=1))
    sizes = {f: os.path.getsize(os.path.join(OUT, f))
             for f in os.listdir(OUT) if f.startswith("golden")}
    print("files:", sizes, "total KB:", sum(sizes.values()) // 1024)


if __name__ == "__main__":
    main()
