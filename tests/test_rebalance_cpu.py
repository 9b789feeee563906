"""Cost-model repartitioning: live rebalance mid-training (gloo, ws=2)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd.graph import synthetic_dataset
from roc_amd.parallel.partition import build_shard, edge_balanced_bounds
from roc_amd import build_model, AdamOptimizer, Trainer

WS = 2


def _worker(rank, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08,
                                                      seed=8)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        tr.attach_full_graph(g)
        for _ in range(2):
            tr.train_epoch()
        old_bounds = list(tr.shard.bounds)
        nb = tr.measure_and_rebalance(feats, labels, mask, probe_epochs=2)
        # keep training on the new shard; loss must stay finite and the
        # model must still improve
        m0 = tr.evaluate()
        for _ in range(5):
            tr.train_epoch()
        m1 = tr.evaluate()
        q.put((rank, old_bounds, list(nb), m0["ce_loss"], m1["ce_loss"],
               None))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, None, None, None, None, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_rebalance_midtraining():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 29541, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, ob, nb, l0, l1, err in res:
        assert err is None, f"rank {rank}: {err}"
        assert nb[0] == 0 and nb[-1] == ob[-1]
        assert l1 == l1 and l1 < 10  # finite, sane
    # both ranks agreed on bounds
    assert res[0][2] == res[1][2]
