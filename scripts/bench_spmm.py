#!/usr/bin/env python3
"""Standalone SpMM kernel A/B microbench (within-process, interleaved
rounds per CDNA guide §5.4 rule 24). Run on a GPU box:

  python scripts/bench_spmm.py [--nodes N --edges E --rounds R]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def time_variant(fn, rounds, stream=None):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    fn()  # warm
    torch.cuda.synchronize()
    times = []
    for _ in range(rounds):
        s.record()
        fn()
        e.record()
        torch.cuda.synchronize()
        times.append(s.elapsed_time(e))
    times.sort()
    return times[len(times) // 2], times[0]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=232965)
    ap.add_argument("--edges", type=int, default=114848857)
    ap.add_argument("--rounds", type=int, default=5)
    ap.add_argument("--dims", type=int, nargs="+", default=[48, 256])
    args = ap.parse_args()

    from roc_amd.graph import synthetic_graph
    from roc_amd import _C

    g = synthetic_graph(args.nodes, args.edges, seed=1)
    dev = "cuda:0"
    rowptr = g.rowptr.to(dev)
    colidx = g.colidx.to(dev)
    deg = (g.rowptr[1:] - g.rowptr[:-1]).float().clamp(min=1)
    rsq = deg.rsqrt().to(dev)
    row_order = torch.argsort(-deg).int().to(dev)

    for D in args.dims:
        x = torch.randn(args.nodes, D).to(torch.bfloat16).to(dev)
        out = torch.empty_like(x)
        gb = args.edges * D * 2 / 1e9
        print(f"== D={D}  gather volume {gb:.1f} GB ==")
        # locality ceiling probe: confine sources to an L2-scale window
        for window in (4096, 16384, 65536):
            ci_w = torch.remainder(colidx, window)
            med, best = time_variant(
                lambda: _C.spmm(out, x, rowptr, ci_w, rsq, None, None),
                args.rounds)
            mb = window * D * 2 / 1e6
            print(f"  src-window {window:6d} rows ({mb:6.1f} MB): "
                  f"median {med:8.2f} ms ({gb/med:.2f} TB/s eff)", flush=True)
        variants = {}
        for un in (8, 16):
            for use_order in (0, 1):
                os.environ["ROC_SPMM_UNROLL"] = str(un)
                _C.spmm_refresh_knobs()  # knobs are pinned at first launch
                ro = row_order if use_order else None

                def fn(ro=ro):
                    _C.spmm(out, x, rowptr, colidx, rsq, None, ro)

                med, best = time_variant(fn, args.rounds)
                key = f"un={un} order={use_order}"
                variants[key] = (med, best)
                print(f"  {key}: median {med:8.2f} ms  best {best:8.2f}"
                      f"  ({gb/med:.2f} TB/s eff)", flush=True)
        os.environ.pop("ROC_SPMM_UNROLL", None)
        _C.spmm_refresh_knobs()
    print("done")


if __name__ == "__main__":
    main()
