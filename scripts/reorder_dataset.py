#!/usr/bin/env python3
"""Offline dataset reordering (preprocessing for real on-disk datasets).

Reads a reference-format dataset (<prefix>.add_self_edge.lux,
<prefix>.feats.csv|.bin, <prefix>.label, <prefix>.mask), applies a
locality ordering from roc_amd.graph.ORDERINGS, and writes the relabeled
dataset to a new prefix (graph + .feats.bin + .label + .mask +
.perm.npy). Training on the result is the same mathematical problem
(permutation equivariance — tested in tests/test_graph.py), but gathers
are community-local and contiguous-range partitions cut fewer edges
(profiles/r16). Unlike train.py --reorder (whole-graph, in-memory),
this runs ONCE and the output works with windowed multi-rank loading
(`train.py --file ... world>1`), which cannot reorder on the fly.

  python scripts/reorder_dataset.py --in data/reddit --out data/reddit-lp \
      --order cluster --in-dim 602
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--in", dest="inp", required=True,
                    help="input dataset prefix")
    ap.add_argument("--out", dest="out", required=True,
                    help="output dataset prefix")
    ap.add_argument("--order", default="cluster",
                    help="degree | rcm | cluster")
    ap.add_argument("--in-dim", type=int, required=True,
                    help="feature dimension (needed to read .feats)")
    args = ap.parse_args()

    from roc_amd.graph import (load_lux, save_lux, load_features,
                               load_labels, load_mask, reorder_graph,
                               ORDERINGS, MASK_NAMES)

    g = load_lux(args.inp + ".add_self_edge.lux")
    print(f"[load] {g.num_nodes} nodes / {g.num_edges} edges", flush=True)
    feats = load_features(args.inp, g.num_nodes, args.in_dim)
    labels = load_labels(args.inp + ".label", g.num_nodes)
    mask = load_mask(args.inp + ".mask", g.num_nodes)

    perm = ORDERINGS[args.order](g)
    g2 = reorder_graph(g, perm)
    print(f"[order] {args.order} done", flush=True)

    os.makedirs(os.path.dirname(os.path.abspath(args.out)), exist_ok=True)
    save_lux(args.out + ".add_self_edge.lux", g2)
    feats.numpy()[perm.numpy()].astype(np.float32).tofile(
        args.out + ".feats.bin")
    with open(args.out + ".label", "w") as f:
        f.write("\n".join(str(int(v)) for v in labels[perm]) + "\n")
    with open(args.out + ".mask", "w") as f:
        f.write("\n".join(MASK_NAMES[int(v)] for v in mask[perm]) + "\n")
    np.save(args.out + ".perm.npy", perm.numpy())  # old id -> position
    print(f"[write] {args.out}.{{add_self_edge.lux,feats.bin,label,mask,"
          f"perm.npy}}", flush=True)


if __name__ == "__main__":
    main()
