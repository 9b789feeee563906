#!/usr/bin/env python3
"""Mini-batch tier microbench: native sampler throughput + per-batch
step time. CPU-runnable (sampler numbers are host-side by design);
pass --device cuda:0 on a GPU box for the full sampled step.

  python scripts/bench_sampling.py --dataset reddit --batch 4096 \
      --fanouts 25,10 [--device cuda:0]
"""
import argparse
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from roc_amd import AdamOptimizer, build_model
from roc_amd.graph import synthetic_dataset
from roc_amd.sampling import MiniBatchTrainer, sample_blocks


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--dataset", default="reddit")
    ap.add_argument("--scale", type=float, default=0.3)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--fanouts", default="25,10")
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--reps", type=int, default=5)
    args = ap.parse_args()
    fanouts = [int(f) for f in args.fanouts.split(",")]

    g, feats, labels, mask, c = synthetic_dataset(
        args.dataset, scale=args.scale, seed=1, learnable_labels=True)
    print(f"graph: {g.num_nodes} nodes / {g.num_edges} edges; "
          f"batch {args.batch} fanouts {fanouts}", flush=True)

    rng = np.random.default_rng(0)
    targets = rng.choice(g.num_nodes, size=args.batch, replace=False)

    # sampler alone
    t0 = time.perf_counter()
    for _ in range(args.reps):
        blocks = sample_blocks(g, targets, fanouts, rng)
    dt = (time.perf_counter() - t0) / args.reps * 1e3
    srcs = blocks[0].n_src
    edges = sum(int(b.colidx.numel()) for b in blocks)
    print(f"sample_blocks: {dt:.1f} ms/batch "
          f"({srcs} input rows, {edges} sampled edges)", flush=True)

    # full sampled step (sample + H2D + fwd + bwd + Adam)
    dims = [feats.shape[1], 256, c]
    model = build_model("sage", dims, dropout=0.5, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    dtype = (torch.bfloat16 if args.device.startswith("cuda")
             else torch.float32)
    tr = MiniBatchTrainer(model, g, feats, labels, mask, opt,
                          fanouts=fanouts, batch_size=args.batch,
                          device=args.device, compute_dtype=dtype, seed=1)
    tr.train_epoch()  # warmup
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    loss = tr.train_epoch()
    if args.device.startswith("cuda"):
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    nb = (tr.train_ids.size + args.batch - 1) // args.batch
    print(f"train_epoch: {dt * 1e3:.0f} ms ({nb} batches, "
          f"{dt / nb * 1e3:.1f} ms/batch, loss {loss:.4f})", flush=True)


if __name__ == "__main__":
    main()
