"""Comm/compute-overlap halo aggregation == sequential halo+aggregate
(gloo, world_size=2, all three model families)."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd.graph import synthetic_dataset
from roc_amd.parallel.partition import build_shard, edge_balanced_bounds
from roc_amd import build_model, AdamOptimizer, Trainer

WS = 2


def _worker(rank, port, model_name, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08,
                                                      seed=12)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        assert sh.has_overlap_split
        ws = {}
        for overlap in ("0", "1"):
            os.environ["ROC_OVERLAP"] = overlap
            model = build_model(model_name, [feats.shape[1], 16, c],
                                dropout=0.0, seed=1)
            opt = AdamOptimizer(model.parameters(), lr=0.01)
            tr = Trainer(model, sh, feats, labels, mask, opt)
            for _ in range(3):
                tr.train_epoch()
            key = "overlap" if overlap == "1" else "seq"
            p0 = next(iter(model.parameters()))
            ws[key] = p0.detach().numpy().copy()
        q.put((rank, ws, None))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, None, traceback.format_exc()))
    finally:
        os.environ.pop("ROC_OVERLAP", None)
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.parametrize("model_name,port", [("gcn", 29561), ("sage", 29563),
                                             ("gin", 29565)])
def test_overlap_matches_sequential(model_name, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, port, model_name, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, ws, err in res:
        assert err is None, f"rank {rank}: {err}"
        a = torch.from_numpy(ws["seq"])
        b = torch.from_numpy(ws["overlap"])
        assert torch.allclose(a, b, atol=1e-5), \
            (model_name, rank, (a - b).abs().max())
