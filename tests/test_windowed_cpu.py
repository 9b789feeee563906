"""Windowed dataset loading + comm-based halo send plan (gloo ws=2):
- load_lux_cols window == full loader slice
- build_shard_from_lux (windowed + all_to_all requests) produces the
  SAME shard as the full-graph scan builder
- feature window == full feature slice
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from roc_amd.graph import (synthetic_graph, save_lux, load_lux,
                           load_lux_meta, load_lux_cols, load_features,
                           load_features_window)
from roc_amd.parallel.partition import (build_shard, build_shard_from_lux,
                                        edge_balanced_bounds)

WS = 2


def test_windowed_cols(tmp_path):
    g = synthetic_graph(120, 1500, seed=6)
    p = str(tmp_path / "g.lux")
    save_lux(p, g)
    n, e, rowptr = load_lux_meta(p)
    assert n == 120 and e == g.num_edges
    assert torch.equal(rowptr, g.rowptr)
    cols = load_lux_cols(p, n, 100, 900)
    assert np.array_equal(cols, g.colidx[100:900].numpy())


def test_feature_window(tmp_path):
    pref = str(tmp_path / "ds")
    feats = np.random.default_rng(0).standard_normal((50, 7)).astype(np.float32)
    np.savetxt(pref + ".feats.csv", feats, delimiter=",")
    full = load_features(pref, 50, 7)
    win = load_features_window(pref, 50, 7, 10, 30)
    assert torch.allclose(win, full[10:30])


def _worker(rank, port, lux_path, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["ROC_COMM_MODE"] = "halo"  # compare the halo structures
        dist.init_process_group("gloo", rank=rank, world_size=WS)
        g = load_lux(lux_path)
        bounds = edge_balanced_bounds(g.rowptr, WS)
        scan = build_shard(g, rank, WS, bounds)
        comm = build_shard_from_lux(lux_path, rank, WS, bounds)
        same = {}
        for k in ("rowptr", "colidx", "t_rowptr", "t_colidx", "send_idx",
                  "halo_ids", "rsqrt_deg_ext", "loc_rowptr", "halo_colidx",
                  "t_halo_rowptr"):
            a, b = getattr(scan, k), getattr(comm, k)
            same[k] = bool(torch.equal(a, b))
        same["send_splits"] = scan.send_splits == comm.send_splits
        same["recv_splits"] = scan.recv_splits == comm.recv_splits
        q.put((rank, same, None))
    except Exception:  # pragma: no cover
        import traceback
        q.put((rank, None, traceback.format_exc()))
    finally:
        os.environ.pop("ROC_COMM_MODE", None)
        if dist.is_initialized():
            dist.destroy_process_group()


def test_comm_plan_matches_scan_plan(tmp_path):
    g = synthetic_graph(300, 5000, seed=14)
    p = str(tmp_path / "g.lux")
    save_lux(p, g)
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_worker, args=(r, 29581, p, q))
             for r in range(WS)]
    for pr in procs:
        pr.start()
    res = sorted([q.get() for _ in range(WS)], key=lambda t: t[0])
    for pr in procs:
        pr.join(timeout=120)
    for rank, same, err in res:
        assert err is None, f"rank {rank}: {err}"
        bad = [k for k, v in same.items() if not v]
        assert not bad, f"rank {rank}: mismatched {bad}"


def test_train_from_lux_two_ranks(tmp_path):
    """torchrun ws=2 over reference-format files: each rank reads only
    its window (build_shard_from_lux + load_features_window)."""
    import subprocess
    import sys as _sys
    import numpy as np
    n, d, c = 160, 10, 4
    g = synthetic_graph(n, 1300, seed=31)
    pref = str(tmp_path / "w")
    save_lux(pref + ".add_self_edge.lux", g)
    rng = np.random.default_rng(1)
    np.savetxt(pref + ".feats.csv",
               rng.standard_normal((n, d)).astype(np.float32), delimiter=",")
    np.savetxt(pref + ".label", rng.integers(0, c, n), fmt="%d")
    with open(pref + ".mask", "w") as f:
        f.write("\n".join(rng.choice(["Train", "Val", "Test"], size=n)) + "\n")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, PYTHONPATH=repo)
    r = subprocess.run(
        [_sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29621", "train.py", "--file", pref,
         "--layers", f"{d}-8-{c}", "--epochs", "6", "--eval-every", "3"],
        cwd=repo, env=env, capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1200:])
    assert "epoch     6" in r.stdout, r.stdout[-800:]


def test_windowed_mean_loss_matches_single_rank(tmp_path):
    """ADVICE-high regression: with windowed --file loading and the
    default mean loss, every rank must use the GLOBAL train count for
    the gradient scale — ws=2 training must produce the same weights
    as the single-rank run on the same files (halo mode exercises the
    gloo send-plan exchange too)."""
    import subprocess
    import sys
    g = synthetic_graph(400, 6000, seed=9)
    pref = str(tmp_path / "ds")
    save_lux(pref + ".add_self_edge.lux", g)
    rng = np.random.default_rng(3)
    feats = rng.standard_normal((400, 8)).astype(np.float32)
    feats.tofile(pref + ".feats.bin")
    np.savetxt(pref + ".label", rng.integers(0, 3, 400), fmt="%d")
    names = np.array(["Train", "Val", "Test", "None"])
    with open(pref + ".mask", "w") as f:
        f.write("\n".join(names[rng.integers(0, 4, 400)]) + "\n")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, ROC_COMM_MODE="halo")
    base = [os.path.join(repo, "train.py"), "--file", pref, "--layers",
            "8-16-3", "--epochs", "3", "--loss", "mean", "--dropout", "0",
            "--eval-every", "0", "--seed", "1"]
    r1 = subprocess.run([sys.executable] + base +
                        ["--checkpoint", str(tmp_path / "ck1.pt")],
                        capture_output=True, text=True, timeout=300, env=env)
    assert r1.returncode == 0, r1.stderr[-800:]
    r2 = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29741"] + base +
        ["--checkpoint", str(tmp_path / "ck2.pt")],
        capture_output=True, text=True, timeout=600, env=env, cwd=repo)
    assert r2.returncode == 0, r2.stderr[-1500:]
    s1 = torch.load(str(tmp_path / "ck1.pt"), weights_only=False)["model"]
    s2 = torch.load(str(tmp_path / "ck2.pt"), weights_only=False)["model"]
    for k in s1:
        assert torch.allclose(s1[k], s2[k], atol=1e-5), \
            (k, (s1[k] - s2[k]).abs().max())
