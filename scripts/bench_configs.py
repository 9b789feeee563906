#!/usr/bin/env python3
"""Measure all five BASELINE.json configs at FULL synthetic scale on one
GPU (the driver's multi-GPU scaling run covers N>1). Emits one JSON line
per config."""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)

CONFIGS = [
    # (name, dataset, model, layers, hidden, dtype, offload)
    ("1-cora-gcn-cpu-shape", "cora", "gcn", 2, 16, "bf16", False),
    ("2-reddit-gcn", "reddit", "gcn", 2, 256, "bf16", False),
    ("3-amazon-sage", "amazon", "sage", 3, 256, "bf16", False),
    ("4-products-gin", "ogbn-products", "gin", 3, 256, "bf16", False),
    ("5-papers-gcn-offload", "papers-synth-small", "gcn", 4, 128, "bf16",
     True),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--scale", type=float, default=1.0)
    ap.add_argument("--only", default=None)
    args = ap.parse_args()
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"

    for name, ds, model_name, layers, hidden, dtype, offload in CONFIGS:
        if args.only and args.only not in name:
            continue
        t0 = time.perf_counter()
        g, feats, labels, mask, c = synthetic_dataset(ds, seed=1,
                                                      scale=args.scale)
        pad = (-feats.shape[1]) % 8
        if pad:
            feats = torch.nn.functional.pad(feats, (0, pad))
        c_pad = c + ((-c) % 64)
        shard = build_shard(g, 0, 1)
        gen_s = time.perf_counter() - t0
        dims = [feats.shape[1]] + [hidden] * (layers - 1) + [c_pad]
        model = build_model(model_name, dims, dropout=0.5, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        cdtype = torch.bfloat16 if (dtype == "bf16" and dev != "cpu") \
            else torch.float32
        tr = Trainer(model, shard, feats, labels, mask, opt, device=dev,
                     compute_dtype=cdtype, num_classes=c,
                     grad_scale=1.0 / max(int((mask == 1).sum()), 1))
        if offload:
            tr.enable_offload(min_bytes=1 << 24)
        for _ in range(args.warmup):
            tr.train_epoch()
        elapsed = tr.timed_epochs(args.steps)
        md = tr.evaluate()
        print(json.dumps({
            "config": name, "dataset": ds, "model": model_name,
            "dims": dims, "nodes": g.num_nodes, "edges": g.num_edges,
            "ms_per_epoch": elapsed / args.steps * 1e3,
            "gen_s": round(gen_s, 1), "dtype": dtype, "n_gpus": 1,
            "offload": offload, "ce_loss": round(md["ce_loss"], 4),
            "train_acc": round(md["train_acc"], 4),
        }), flush=True)
        del tr, model, opt, g, feats, labels, mask, shard
        if dev != "cpu":
            torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
