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


# ---------------------------------------------------------------------------
# comm-aware cost model (fit + bound equalization)
# ---------------------------------------------------------------------------

def test_fit_cost_model_recovers_coefficients():
    from roc_amd.parallel.partition import fit_cost_model
    a, b = 2e-6, 5e-5
    samples = [(e, h, a * e + b * h)
               for e, h in [(1e6, 100), (2e6, 5000), (5e5, 9000),
                            (3e6, 400)]]
    af, bf = fit_cost_model(samples)
    assert af == pytest.approx(a, rel=1e-6)
    assert bf == pytest.approx(b, rel=1e-6)
    # underdetermined (2 samples) or degenerate -> edge-only fallback
    af, bf = fit_cost_model(samples[:2])
    assert bf == 0.0 and af > 0


def test_rebalance_bounds_comm_moves_edges_off_comm_heavy_rank():
    from roc_amd.parallel.partition import rebalance_bounds_comm
    n = 1000
    rowptr = torch.arange(0, (n + 1) * 10, 10, dtype=torch.int64)  # 10 e/v
    bounds = [0, 500, 1000]
    # rank 0 pays a big comm cost (many halo rows): same edges, more time
    a, b = 1e-6, 1e-4
    e = 5000.0
    times = [a * e + b * 2000, a * e + b * 0]
    samples = []
    # two rounds so the 2-coef fit has rank-2 data
    nb = rebalance_bounds_comm(rowptr, bounds, times, [2000, 0], samples)
    times2 = [a * 10 * (nb[1] - nb[0]) + b * 2000, a * 10 * (1000 - nb[1])]
    nb2 = rebalance_bounds_comm(rowptr, nb, times2, [2000, 0], samples)
    # rank 0 must end with FEWER vertices (its comm overhead is real)
    assert nb2[1] < 500
    assert nb2[0] == 0 and nb2[-1] == 1000


# ---------------------------------------------------------------------------
# rebalance under WINDOWED file loading (the path VERDICT flagged:
# measure_and_rebalance used to require full feats on every rank)
# ---------------------------------------------------------------------------

def _windowed_worker(rank, port, pref, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = "halo"
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        from roc_amd.graph import (load_lux_meta, load_features_window,
                                   load_labels, load_mask)
        from roc_amd.parallel.partition import build_shard_from_lux
        lux = pref + ".add_self_edge.lux"
        n, _, rowptr = load_lux_meta(lux)
        in_dim = 8
        sh = build_shard_from_lux(lux, rank, WS)
        feats = load_features_window(pref, n, in_dim, sh.lo, sh.hi)
        labels = load_labels(pref + ".label", n)[sh.lo:sh.hi]
        mask = load_mask(pref + ".mask", n)[sh.lo:sh.hi]
        model = build_model("gcn", [in_dim, 16, 3], dropout=0.0, seed=1)
        opt = AdamOptimizer(model.parameters(), lr=0.01)
        tr = Trainer(model, sh, feats, labels, mask, opt, local_slices=True)

        def loader(lo, hi):
            return (load_features_window(pref, n, in_dim, lo, hi),
                    load_labels(pref + ".label", n)[lo:hi],
                    load_mask(pref + ".mask", n)[lo:hi])

        tr.attach_windowed_dataset(lux, loader, rowptr)
        for _ in range(2):
            tr.train_epoch()
        old_bounds = list(tr.shard.bounds)
        nb = tr.measure_and_rebalance(probe_epochs=2)
        # x/labels/mask must match the new window exactly
        want = load_features_window(pref, n, in_dim, tr.shard.lo, tr.shard.hi)
        ok_x = torch.allclose(tr.x.float().cpu(), want)
        for _ in range(3):
            tr.train_epoch()
        m = tr.evaluate()
        q.put((rank, old_bounds, list(nb), bool(ok_x), m["ce_loss"], None))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, None, None, None, None, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_rebalance_windowed_loading(tmp_path):
    import numpy as np
    from roc_amd.graph import save_lux
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.08, seed=8)
    in_dim = 8
    pref = str(tmp_path / "mini")
    save_lux(pref + ".add_self_edge.lux", g)
    feats[:, :in_dim].contiguous().numpy().astype(np.float32).tofile(
        pref + ".feats.bin")
    np.savetxt(pref + ".label", (labels % 3).numpy(), fmt="%d")
    with open(pref + ".mask", "w") as f:
        names = {0: "None", 1: "Train", 2: "Val", 3: "Test"}
        f.write("\n".join(names[int(v)] for v in mask) + "\n")
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_windowed_worker, args=(r, 29543, pref, q))
             for r in range(WS)]
    for p in procs:
        p.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for p in procs:
        p.join(timeout=180)
    for rank, ob, nb, ok_x, loss, err in res:
        assert err is None, f"rank {rank}: {err}"
        assert ok_x, f"rank {rank}: reloaded window mismatch"
        assert loss == loss and loss < 10
    assert res[0][2] == res[1][2]  # agreed bounds
