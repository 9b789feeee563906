#!/usr/bin/env python3
"""Ordering x SpMM locality A/B on a community-structured graph.

Answers the question profiles/r04 left open: the tiling-ceiling probe
showed the gather SpMM caps at ~1.6x even with a perfectly windowed
source set, and reordering can't help a LOCALITY-FREE uniform graph —
but real graphs are community-heavy. Here we generate a Reddit-shaped
graph WITH planted communities, destroy the ordering with a random
relabeling (what an arbitrary dataset numbering looks like), and
measure the hot SpMM under each recovery ordering from
roc_amd.graph.ORDERINGS (degree / RCM / LP-cluster).

Run on a GPU box:
  python scripts/bench_reorder.py [--nodes N --edges E --locality F]
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def time_spmm(g, D, rounds, schedule="degree", dev="cuda:0"):
    """schedule: 'degree' (global degree-desc row order), 'natural'
    (no reorder — preserves community grouping in launch order), or
    'degree-block' (degree-desc WITHIN each 4096-row block — wide-row
    balance AND block locality)."""
    from roc_amd import _C
    rowptr = g.rowptr.to(dev)
    colidx = g.colidx.to(dev)
    deg = (g.rowptr[1:] - g.rowptr[:-1]).float().clamp(min=1)
    rsq = deg.rsqrt().to(dev)
    if schedule == "degree":
        row_order = torch.argsort(-deg).int().to(dev)
    elif schedule == "degree-block":
        blk = torch.arange(g.num_nodes, dtype=torch.float64) // 4096
        key = blk * 1e9 - deg.double()  # block-major, degree-desc inside
        row_order = torch.argsort(key).int().to(dev)
    elif schedule == "xcd-split":
        # round-robin dispatch sends consecutive workgroups to different
        # XCDs; interleave 8 contiguous regions so XCD k streams region k
        # (each XCD L2 holds ITS OWN window instead of 8 copies of one)
        n = g.num_nodes
        chunk = (n + 7) // 8
        pad = np.full(chunk * 8, -1, dtype=np.int64)
        pad[:n] = np.arange(n)
        ro = pad.reshape(8, chunk).T.reshape(-1)
        row_order = torch.from_numpy(ro[ro >= 0].astype(np.int32)).to(dev)
    else:
        row_order = None
    x = torch.randn(g.num_nodes, D).to(torch.bfloat16).to(dev)
    out = torch.empty_like(x)
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    _C.spmm(out, x, rowptr, colidx, rsq, None, row_order)  # warm
    torch.cuda.synchronize()
    times = []
    for _ in range(rounds):
        s.record()
        _C.spmm(out, x, rowptr, colidx, rsq, None, row_order)
        e.record()
        torch.cuda.synchronize()
        times.append(s.elapsed_time(e))
    times.sort()
    return times[len(times) // 2]


def window_frac(g, w):
    rp, ci = g.rowptr.numpy(), g.colidx.numpy()
    rows = np.repeat(np.arange(g.num_nodes), np.diff(rp))
    return float((np.abs(rows - ci) < w // 2).mean())


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=232965)
    ap.add_argument("--edges", type=int, default=114848857)
    ap.add_argument("--locality", type=float, default=0.8)
    ap.add_argument("--communities", type=int, default=57)  # ~4k rows each
    ap.add_argument("--rounds", type=int, default=5)
    ap.add_argument("--dims", type=int, nargs="+", default=[256])
    ap.add_argument("--orderings", nargs="+",
                    default=["natural", "shuffled", "cluster", "rcm"])
    ap.add_argument("--schedules", nargs="+",
                    default=["degree", "natural", "degree-block"])
    args = ap.parse_args()

    from roc_amd.graph import (synthetic_graph, reorder_graph, ORDERINGS)

    t0 = time.time()
    g_nat = synthetic_graph(args.nodes, args.edges, seed=1,
                            locality=args.locality,
                            num_communities=args.communities)
    print(f"[gen] {g_nat.num_edges} edges, locality={args.locality}, "
          f"{args.communities} communities ({time.time()-t0:.0f}s)",
          flush=True)
    rng = np.random.default_rng(0)
    shuf_perm = torch.from_numpy(rng.permutation(args.nodes))
    g_shuf = reorder_graph(g_nat, shuf_perm)

    graphs = {}
    for name in args.orderings:
        t0 = time.time()
        if name == "natural":
            graphs[name] = (g_nat, 0.0)
        elif name == "shuffled":
            graphs[name] = (g_shuf, 0.0)
        else:
            perm = ORDERINGS[name](g_shuf)
            graphs[name] = (reorder_graph(g_shuf, perm), time.time() - t0)
        print(f"[order] {name}: ready ({time.time()-t0:.0f}s)", flush=True)

    results = []
    for D in args.dims:
        gb = args.edges * D * 2 / 1e9
        for name, (g, order_s) in graphs.items():
            wf = window_frac(g, 4096 * 16)  # 64k-row ~ aggregate-L2 window
            if not torch.cuda.is_available():  # CPU dry-run: fractions only
                print(json.dumps({"ordering": name, "D": D,
                                  "frac_in_64k_window": round(wf, 3)}),
                      flush=True)
                continue
            for sched in args.schedules:
                ms = time_spmm(g, D, args.rounds, schedule=sched)
                r = {"ordering": name, "schedule": sched, "D": D,
                     "ms": round(ms, 2), "eff_TBs": round(gb / ms, 2),
                     "frac_in_64k_window": round(wf, 3),
                     "ordering_cost_s": round(order_s, 1)}
                results.append(r)
                print(json.dumps(r), flush=True)

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/reorder_ab.json", "w") as f:
        json.dump({"args": vars(args), "results": results}, f, indent=1)
    print("done")


if __name__ == "__main__":
    main()
