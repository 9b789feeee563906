#!/usr/bin/env python3
"""A/B: source-strip-blocked SpMM vs the single-pass kernel.

r04 measured a 1.58x ceiling when ALL sources fit one XCD's L2 (2 MB
window) — but that probe kept full-length rows. A real strip schedule
splits each row's edges by source range and pays K accumulate passes:
shorter segments (worse gather-queue depth) + K-1 output re-reads.
This measures the real thing at several strip widths.

Run on a GPU box:  python scripts/bench_spmm_strips.py
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from roc_amd import synthetic_dataset, build_shard
from roc_amd.ops import functional as Fn

_C = Fn._C


def build_strips(rowptr, colidx, n, width):
    """Split the CSR into ceil(n/width) strip-CSRs by source id.
    Columns are sorted within rows, so each row's strip segment is a
    contiguous slice — strip rowptrs come from searchsorted, no edge
    copy reordering needed beyond slicing."""
    rp = rowptr.numpy()
    ci = colidx.numpy().astype(np.int64)
    nrows = rp.shape[0] - 1
    bounds = list(range(width, n, width)) + [n]
    # composite sorted key (row, col): one vectorized searchsorted gives
    # every row's strip boundary position at once
    row = np.repeat(np.arange(nrows, dtype=np.int64), np.diff(rp))
    key = row * (n + 1) + ci
    rows_q = np.arange(nrows, dtype=np.int64) * (n + 1)
    pos = [rp[:-1]]
    for b in bounds[:-1]:
        pos.append(np.searchsorted(key, rows_q + b))
    pos.append(rp[1:])
    strips = []
    ci32 = colidx.numpy()
    for s in range(len(pos) - 1):
        seg_start, seg_end = pos[s], pos[s + 1]
        cnt = seg_end - seg_start
        srp = np.zeros(nrows + 1, dtype=np.int64)
        np.cumsum(cnt, out=srp[1:])
        # slice-copy each row's strip segment (vectorized index build)
        idx = np.repeat(seg_start - srp[:-1], cnt) + np.arange(
            int(srp[-1]), dtype=np.int64)
        sci = ci32[idx] if srp[-1] else np.empty(0, dtype=ci32.dtype)
        strips.append((torch.from_numpy(srp),
                       torch.from_numpy(np.ascontiguousarray(sci))))
    return strips


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--dim", type=int, default=256)
    ap.add_argument("--widths", default="65536,32768,16384,8192")
    ap.add_argument("--fp32acc", action="store_true",
                    help="production path: fp32 partial buffer + final "
                         "cast_rowscale instead of bf16 in-place passes")
    args = ap.parse_args()
    dev = "cuda:0"
    torch.manual_seed(0)
    g, feats, *_ = synthetic_dataset("reddit", seed=1)
    sh = build_shard(g, 0, 1)
    n = g.num_nodes
    D = args.dim
    x = torch.randn(n, D, device=dev).to(torch.bfloat16)
    out = torch.empty(n, D, device=dev, dtype=torch.bfloat16)
    out32 = torch.empty(n, D, device=dev, dtype=torch.float32)
    rp_d = sh.rowptr.to(dev)
    ci_d = sh.colidx.to(dev)

    def timeit(fn, reps=10):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / reps * 1e3

    base = timeit(lambda: _C.spmm(out, x, rp_d, ci_d, None, None, None))
    print(f"single-pass D={D}: {base:.2f} ms", flush=True)
    ref = out.float().cpu().clone()

    for width in (int(w) for w in args.widths.split(",")):
        t0 = time.perf_counter()
        strips = [(srp.to(dev), sci.to(dev))
                  for srp, sci in build_strips(sh.rowptr, sh.colidx, n,
                                               width)]
        prep = time.perf_counter() - t0

        if args.fp32acc:
            def run():
                for i, (srp, sci) in enumerate(strips):
                    _C.spmm(out32, x, srp, sci, None, None, None, i > 0)
                _C.cast_rowscale(out, out32, None)
        else:
            def run():
                for i, (srp, sci) in enumerate(strips):
                    _C.spmm(out, x, srp, sci, None, None, None, i > 0)

        t = timeit(run)
        err = (out.float().cpu() - ref).abs().max().item()
        print(f"strips width={width:6d} (K={len(strips):3d}): {t:.2f} ms "
              f"({base / t:.2f}x)  prep {prep:.1f}s  max|diff|={err:.4f}",
              flush=True)


if __name__ == "__main__":
    main()
