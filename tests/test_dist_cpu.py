"""Multi-process distributed path on CPU (gloo, world_size=2):
- halo exchange forward/backward over all_to_all_single
- full sharded training step == single-rank training step
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd.graph import synthetic_dataset
from roc_amd.parallel.partition import build_shard, edge_balanced_bounds
from roc_amd.parallel.halo import halo_exchange
from roc_amd import build_model, AdamOptimizer, Trainer

WS = 2


def _init(rank, port, comm_mode="halo"):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    # pin the exchange strategy: these tests exercise the halo plan
    # directly (auto picks allgather on uniform synthetic graphs)
    os.environ["ROC_COMM_MODE"] = comm_mode
    dist.init_process_group("gloo", rank=rank, world_size=WS)


def _halo_worker(rank, port, q):
    try:
        _init(rank, port)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05, seed=3)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        x = feats[sh.lo:sh.hi].clone().requires_grad_(True)
        xe = halo_exchange(x, sh)
        # forward: halo rows match global features
        expect = torch.cat([feats[sh.lo:sh.hi], feats[sh.halo_ids]]) \
            if sh.n_halo else feats[sh.lo:sh.hi]
        ok_fwd = torch.allclose(xe, expect, atol=1e-6)
        # backward: ones grad -> local grad = 1 + (#ranks that requested row)
        xe.backward(torch.ones_like(xe))
        counts = torch.ones(sh.n_local)
        counts.index_add_(0, sh.send_idx, torch.ones(sh.send_idx.numel()))
        ok_bwd = torch.allclose(x.grad, counts.unsqueeze(1).expand_as(x.grad))
        q.put((rank, bool(ok_fwd), bool(ok_bwd), None))
    except Exception as e:  # pragma: no cover
        q.put((rank, False, False, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _train_worker(rank, port, q):
    try:
        _init(rank, port)
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05, seed=3)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        dims = [feats.shape[1], 16, c]
        model = build_model("gcn", dims, dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        for _ in range(3):
            tr.train_epoch()
        m = tr.evaluate()
        w0 = model.weights[0].detach().numpy().copy()  # plain array: safe to pickle
        q.put((rank, m, w0, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, None, None, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def _run(workers_fn, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=workers_fn, args=(r, port, q)) for r in range(WS)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(WS)]
    for p in procs:
        p.join(timeout=120)
    return sorted(results, key=lambda t: t[0])


def test_halo_exchange_gloo():
    res = _run(_halo_worker, 29511)
    for rank, ok_fwd, ok_bwd, err in res:
        assert err is None, f"rank {rank}: {err}"
        assert ok_fwd, f"rank {rank} halo forward mismatch"
        assert ok_bwd, f"rank {rank} halo backward mismatch"


def test_sharded_training_matches_single_rank():
    res = _run(_train_worker, 29513)
    for rank, m, w, err in res:
        assert err is None, f"rank {rank}: {err}"
    # both ranks must hold identical weights after all-reduced updates
    w_r0 = torch.from_numpy(res[0][2])
    w_r1 = torch.from_numpy(res[1][2])
    assert torch.allclose(w_r0, w_r1, atol=1e-6)

    # single-rank baseline (same seeds -> same init)
    torch.manual_seed(0)
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05, seed=3)
    sh = build_shard(g, 0, 1)
    dims = [feats.shape[1], 16, c]
    model = build_model("gcn", dims, dropout=0.0, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    tr = Trainer(model, sh, feats, labels, mask, opt)
    for _ in range(3):
        tr.train_epoch()
    m_single = tr.evaluate()
    w_single = model.weights[0].detach()
    assert torch.allclose(w_single, w_r0, atol=1e-4), \
        (w_single.sum(), w_r0.sum())
    md = res[0][1]
    assert md["train_total"] == m_single["train_total"]
    assert abs(md["ce_loss"] - m_single["ce_loss"]) < 1e-3


def test_bench_contract_ws2_gloo():
    """The driver's scale run launches bench.py under torchrun with
    --gpus N: guard that contract end-to-end on CPU (gloo, ws=2) — one
    JSON line on rank 0 with the whole-job metric fields."""
    import json
    import os
    import subprocess
    import sys
    import roc_amd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(roc_amd.__file__)))
    env = dict(os.environ)
    env.pop("ROC_SPMM_SCHEDULE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29737", os.path.join(repo, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1", "--scale", "0.02"],
        capture_output=True, text=True, timeout=600, env=env, cwd=repo)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2 and out["steps"] == 2
    assert out["unit"] == "s/epoch" and out["value"] > 0
    assert out["higher_is_better"] is False
    assert "parallelism" in out["config"]


def _recompute_worker(rank, port, q):
    # sharded training with per-layer recompute: the collectives re-run
    # INSIDE backward must line up identically across ranks
    try:
        _init(rank, port)

        def run(recompute, comm_mode):
            os.environ["ROC_COMM_MODE"] = comm_mode
            torch.manual_seed(0)
            g, feats, labels, mask, c = synthetic_dataset(
                "cora", scale=0.05, seed=3)
            bounds = edge_balanced_bounds(g.rowptr, WS)
            sh = build_shard(g, rank, WS, bounds)
            model = build_model("gcn", [feats.shape[1], 16, 16, c],
                                dropout=0.4, seed=1)
            model.recompute = recompute
            opt = AdamOptimizer(model.parameters(), lr=0.01)
            tr = Trainer(model, sh, feats, labels, mask, opt)
            for _ in range(2):
                tr.train_epoch()
            return model.weights[0].detach().numpy().copy()

        for mode in ("halo", "allgather"):
            w_std = run(False, mode)
            w_rec = run(True, mode)
            if not np.allclose(w_std, w_rec, atol=1e-6):
                q.put((rank, None, None,
                       f"{mode}: recompute diverged from standard"))
                return
        q.put((rank, True, None, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, None, None, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_recompute_distributed_matches_standard():
    res = _run(_recompute_worker, 29517)
    for rank, ok, _, err in res:
        assert err is None, f"rank {rank}: {err}"
        assert ok


def _ws3_worker(rank, port, mode, q):
    # odd world size: exercises uneven bounds and the allgather block
    # padding (ag_max_rows) that even splits never stress
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = mode
        dist.init_process_group("gloo", rank=rank, world_size=3)
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08,
                                                      seed=5)
        bounds = edge_balanced_bounds(g.rowptr, 3)
        sh = build_shard(g, rank, 3, bounds)
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        for _ in range(3):
            tr.train_epoch()
        q.put((rank, model.weights[0].detach().numpy().copy(), None))
    except Exception as e:  # pragma: no cover
        q.put((rank, None, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.parametrize("mode", ["halo", "allgather"])
def test_sharded_training_ws3(mode):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29521 if mode == "halo" else 29523
    procs = [ctx.Process(target=_ws3_worker, args=(r, port, mode, q))
             for r in range(3)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(3)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, w, err in res:
        assert err is None, f"rank {rank}: {err}"
    # all ranks agree after all-reduced updates
    for r in (1, 2):
        assert np.allclose(res[0][1], res[r][1], atol=1e-6)

    # single-rank baseline: same final weights
    torch.manual_seed(0)
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08, seed=5)
    sh = build_shard(g, 0, 1)
    model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    tr = Trainer(model, sh, feats, labels, mask, opt)
    for _ in range(3):
        tr.train_epoch()
    # reduction-order noise passes through Adam's m/sqrt(v) normalization,
    # so cross-world-size agreement is approximate; ranks agree to 1e-6
    assert np.allclose(res[0][1], model.weights[0].detach().numpy(),
                       atol=2e-3)


def _elastic_worker(rank, port, ckpt, q):
    # resume a ws=1 checkpoint in a ws=2 job (weights/Adam state are
    # replicated, so restarts may change the GPU count freely)
    try:
        _init(rank, port)
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05,
                                                      seed=3)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        from roc_amd.utils import load_checkpoint
        load_checkpoint(ckpt, tr)
        for _ in range(2):
            tr.train_epoch()
        q.put((rank, model.weights[0].detach().numpy().copy(), None))
    except Exception as e:  # pragma: no cover
        q.put((rank, None, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_elastic_resume_ws1_to_ws2(tmp_path):
    from roc_amd.utils import save_checkpoint, load_checkpoint

    def fresh():
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05,
                                                      seed=3)
        sh = build_shard(g, 0, 1)
        model = build_model("gcn", [feats.shape[1], 16, c], dropout=0.0,
                            seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        return Trainer(model, sh, feats, labels, mask, opt)

    tr = fresh()
    for _ in range(3):
        tr.train_epoch()
    ckpt = str(tmp_path / "elastic.pt")
    save_checkpoint(ckpt, tr)
    for _ in range(2):  # ws=1 continuation = the oracle
        tr.train_epoch()
    w_ref = tr.model.weights[0].detach().numpy()

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_elastic_worker, args=(r, 29531, ckpt, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, w, err in res:
        assert err is None, f"rank {rank}: {err}"
    assert np.allclose(res[0][1], res[1][1], atol=1e-6)
    assert np.allclose(res[0][1], w_ref, atol=2e-3)  # Adam noise tolerance


def _widen_worker(rank, port, mode, model_name, q):
    # widening hidden layer (in < hidden): the adaptive aggregation
    # order flips to aggregate-FIRST, so the halo/allgather exchange
    # moves the in-width rows — this worker covers that comm path
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = mode
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        torch.manual_seed(0)
        g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05,
                                                      seed=3)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        sh = build_shard(g, rank, WS, bounds)
        dims = [feats.shape[1], feats.shape[1] * 2, c]
        model = build_model(model_name, dims, dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
        tr = Trainer(model, sh, feats, labels, mask, opt)
        for _ in range(3):
            tr.train_epoch()
        w = next(model.parameters()).detach().numpy().copy()
        q.put((rank, w, None))
    except Exception as e:  # pragma: no cover
        q.put((rank, None, repr(e)))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


@pytest.mark.parametrize("mode,model_name,port", [
    ("halo", "gcn", 29531), ("allgather", "gcn", 29533),
    ("halo", "sage", 29535), ("halo", "appnp", 29537),
])
def test_widening_layer_sharded_matches_single(mode, model_name, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_widen_worker,
                         args=(r, port, mode, model_name, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, w, err in res:
        assert err is None, f"rank {rank}: {err}"
    assert np.allclose(res[0][1], res[1][1], atol=1e-6)

    torch.manual_seed(0)
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.05, seed=3)
    sh = build_shard(g, 0, 1)
    dims = [feats.shape[1], feats.shape[1] * 2, c]
    model = build_model(model_name, dims, dropout=0.0, seed=1)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    tr = Trainer(model, sh, feats, labels, mask, opt)
    for _ in range(3):
        tr.train_epoch()
    w_single = next(model.parameters()).detach().numpy()
    assert np.allclose(res[0][1], w_single, atol=2e-3)
