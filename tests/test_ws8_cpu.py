"""World-size-8 multi-process equality on CPU (gloo) — the exact rank
count of the driver's round-end MI355X scale run. ws=2/3 are covered in
test_dist_cpu.py; 8 ranks additionally stress the per-destination split
sizing, the 8-block allgather padding/reduce-to-owner order, and the
8-way edge-balanced bounds."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd import build_model, AdamOptimizer, Trainer
from roc_amd.graph import synthetic_dataset
from roc_amd.parallel.partition import build_shard, edge_balanced_bounds

WS = 8


def _worker(rank, port, mode, q, ws=WS):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = mode
        if mode == "halo":
            os.environ["ROC_OVERLAP"] = "1"  # overlap split path at ws=8
        dist.init_process_group("gloo", rank=rank, world_size=ws)
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2,
                                                      seed=3)
        bounds = edge_balanced_bounds(g.rowptr, ws)
        sh = build_shard(g, rank, ws, bounds)
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        for _ in range(2):
            tr.train_epoch()
        md = tr.evaluate()
        q.put((rank, md, model.weights[0].detach().numpy().copy(), None))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, None, None, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _single_rank_baseline():
    torch.manual_seed(0)
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.2, seed=3)
    sh = build_shard(g, 0, 1)
    model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    tr = Trainer(model, sh, feats, labels, mask, opt)
    for _ in range(2):
        tr.train_epoch()
    return tr.evaluate(), model.weights[0].detach()


def test_bench_contract_ws8_gloo():
    """The driver's largest scale point: torchrun ws=8 through bench.py
    end-to-end (gloo on CPU, tiny scale) — one JSON line, whole-job
    metric, agreed auto-extended step count across 8 ranks."""
    import json
    import subprocess
    import sys
    import roc_amd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(
        roc_amd.__file__)))
    env = dict(os.environ)
    env.pop("ROC_SPMM_SCHEDULE", None)
    env["ROC_BENCH_CACHE"] = env.get("TMPDIR", "/tmp")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29739", os.path.join(repo, "bench.py"),
         "--gpus", "8", "--steps", "2", "--warmup", "1", "--scale",
         "0.02", "--exact-steps"],
        capture_output=True, text=True, timeout=900, env=env, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 8 and out["steps"] == 2
    assert out["unit"] == "s/epoch" and out["value"] > 0
    assert "x8" in out["config"]["parallelism"]


@pytest.mark.parametrize("mode,nranks,port", [
    ("halo", 8, 29561), ("allgather", 8, 29563),
    ("halo", 4, 29565), ("allgather", 4, 29567),  # the driver's N=4 point
])
def test_ws8_matches_single_rank(mode, nranks, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, port, mode, q, nranks))
             for r in range(nranks)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(nranks)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=300)
    for rank, md, w, err in res:
        assert err is None, f"rank {rank}: {err}"
    ws = [torch.from_numpy(r[2]) for r in res]
    for k in range(1, nranks):  # replicated weights identical everywhere
        assert torch.allclose(ws[0], ws[k], atol=1e-6), k
    md1, w1 = _single_rank_baseline()
    assert torch.allclose(w1, ws[0], atol=1e-4), (w1 - ws[0]).abs().max()
    assert res[0][1]["train_total"] == md1["train_total"]
    assert abs(res[0][1]["ce_loss"] - md1["ce_loss"]) < 1e-3
