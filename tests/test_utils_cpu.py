"""Checkpoint/resume + tracer tests (CPU)."""
import os
import subprocess
import sys

import pytest

import torch

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)
from roc_amd.utils import save_checkpoint, load_checkpoint
from roc_amd.utils.trace import Tracer


def _mk(seed=1):
    g, feats, labels, mask, c = synthetic_dataset("cora", scale=0.04, seed=4)
    shard = build_shard(g, 0, 1)
    model = build_model("gcn", [feats.shape[1], 8, c], dropout=0.3, seed=seed)
    opt = AdamOptimizer(model.parameters(), lr=0.01, weight_decay=1e-4)
    return Trainer(model, shard, feats, labels, mask, opt, seed=seed)


def test_checkpoint_resume_bitexact(tmp_path):
    p = str(tmp_path / "ckpt.pt")
    tr = _mk()
    for _ in range(4):
        tr.train_epoch()
    save_checkpoint(p, tr)
    # continue 3 more epochs
    for _ in range(3):
        tr.train_epoch()
    w_direct = tr.model.weights[0].detach().clone()

    # fresh trainer, resume, same 3 epochs -> identical weights
    tr2 = _mk(seed=99)  # different init, must be overwritten by checkpoint
    load_checkpoint(p, tr2)
    assert tr2.epoch == 4
    for _ in range(3):
        tr2.train_epoch()
    w_resumed = tr2.model.weights[0].detach()
    assert torch.allclose(w_direct, w_resumed, atol=1e-7), \
        (w_direct - w_resumed).abs().max()


def test_tracer_cpu_spans():
    tr = _mk()
    t = tr.enable_tracing()
    tr.train_epoch()
    tr.train_epoch()
    s = t.summarize()
    for phase in ("forward", "backward", "adam", "zero_grad"):
        assert phase in s and s[phase] >= 0.0


def test_trace_dump(tmp_path):
    tr = _mk()
    t = tr.enable_tracing()
    tr.train_epoch()
    out = str(tmp_path / "trace.json")
    t.dump_chrome(out)
    import json
    with open(out) as f:
        j = json.load(f)
    assert len(j["traceEvents"]) >= 4


def test_train_cli_runs(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ck = str(tmp_path / "c.pt")
    r = subprocess.run(
        [sys.executable, "train.py", "--dataset", "cora-synthetic",
         "--scale", "0.04", "--epochs", "6", "--hidden", "8",
         "--eval-every", "5", "--checkpoint", ck],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "epoch     5" in r.stdout
    assert os.path.exists(ck)
    # resume
    r2 = subprocess.run(
        [sys.executable, "train.py", "--dataset", "cora-synthetic",
         "--scale", "0.04", "--epochs", "8", "--hidden", "8",
         "--eval-every", "0", "--resume", ck],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "resumed" in r2.stdout


def test_train_cli_rejects_bad_layers():
    """--layers is the reference's dash-separated dims string
    (`gnn.cc:130-143`); a bare count must fail with a clear message,
    not a shape error deep in the kernels."""
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "train.py", "--dataset", "cora-synthetic",
         "--scale", "0.04", "--layers", "4", "--epochs", "1"],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r.returncode != 0
    assert "--num-layers" in (r.stderr + r.stdout)


def test_auto_recover_cli(tmp_path):
    """Divergence guard + checkpoint restore: train with an absurd LR that
    NaNs out; --auto-recover must restore and finish."""
    import numpy as np
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ck = str(tmp_path / "r.pt")
    # first produce a good checkpoint
    r = subprocess.run(
        [sys.executable, "train.py", "--dataset", "cora-synthetic",
         "--scale", "0.04", "--epochs", "3", "--hidden", "8",
         "--eval-every", "0", "--checkpoint", ck],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r.returncode == 0, r.stderr[-1500:]
    # resume with a huge LR: must diverge, recover, and complete
    r2 = subprocess.run(
        [sys.executable, "train.py", "--dataset", "cora-synthetic",
         "--scale", "0.04", "--epochs", "8", "--hidden", "8",
         "--eval-every", "1", "--resume", ck, "--checkpoint", ck,
         "--auto-recover", "--lr", "1e18"],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r2.returncode == 0, (r2.stdout[-800:], r2.stderr[-800:])
    assert "[recover]" in r2.stdout, r2.stdout[-800:]


def test_train_from_reference_format_files(tmp_path):
    """Full E2E on the reference's on-disk formats: .lux + feats.csv +
    .label + .mask -> train.py --file."""
    import numpy as np
    from roc_amd.graph import synthetic_graph, save_lux
    n, d, c = 120, 12, 5
    g = synthetic_graph(n, 900, seed=21)
    pref = str(tmp_path / "tiny")
    save_lux(pref + ".add_self_edge.lux", g)
    rng = np.random.default_rng(0)
    np.savetxt(pref + ".feats.csv",
               rng.standard_normal((n, d)).astype(np.float32), delimiter=",")
    np.savetxt(pref + ".label", rng.integers(0, c, n), fmt="%d")
    masks = rng.choice(["Train", "Val", "Test", "None"], size=n,
                       p=[0.7, 0.1, 0.1, 0.1])
    with open(pref + ".mask", "w") as f:
        f.write("\n".join(masks) + "\n")

    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(
        os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "train.py", "--file", pref,
         "--layers", f"{d}-8-{c}", "--epochs", "10", "--eval-every", "5"],
        cwd=env["PYTHONPATH"], env=env, capture_output=True, text=True,
        timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "epoch    10" in r.stdout, r.stdout
    # binary feature cache was written on first load (load_task.cu:63-65)
    assert os.path.exists(pref + ".feats.bin")


def test_all_config_yamls_run(tmp_path):
    """Every shipped BASELINE config YAML parses and drives a (tiny) run."""
    import glob
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, PYTHONPATH=repo)
    for cfg in sorted(glob.glob(os.path.join(repo, "configs", "*.yaml"))):
        r = subprocess.run(
            [sys.executable, "train.py", "--config", cfg,
             "--scale", "0.002", "--epochs", "2", "--eval-every", "0",
             "--hidden", "8"],
            cwd=repo, env=env, capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, (cfg, r.stderr[-1500:])
        assert "[config]" in r.stdout, (cfg, r.stdout[-500:])


def test_reorder_dataset_tool(tmp_path):
    # offline preprocessing: scripts/reorder_dataset.py writes a relabeled
    # reference-format dataset whose aggregation commutes with the perm
    import subprocess
    import sys
    import numpy as np
    from roc_amd.graph import (synthetic_dataset, save_lux, MASK_NAMES,
                               load_lux, load_features, load_labels,
                               load_mask)
    from roc_amd.ops import functional as F
    from roc_amd.parallel.partition import build_shard
    g, feats, labels, mask, c = synthetic_dataset("cora", seed=2, scale=0.2)
    pre = str(tmp_path / "cora")
    save_lux(pre + ".add_self_edge.lux", g)
    feats.numpy().astype(np.float32).tofile(pre + ".feats.bin")
    with open(pre + ".label", "w") as f:
        f.write("\n".join(str(int(v)) for v in labels) + "\n")
    with open(pre + ".mask", "w") as f:
        f.write("\n".join(MASK_NAMES[int(v)] for v in mask) + "\n")
    import roc_amd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(roc_amd.__file__)))
    out = str(tmp_path / "cora-lp")
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts/reorder_dataset.py"),
         "--in", pre, "--out", out, "--order", "cluster",
         "--in-dim", str(feats.shape[1])],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    g2 = load_lux(out + ".add_self_edge.lux")
    f2 = load_features(out, g2.num_nodes, feats.shape[1])
    perm = torch.from_numpy(np.load(out + ".perm.npy"))
    assert torch.equal(f2, feats[perm])
    assert torch.equal(load_labels(out + ".label", g2.num_nodes),
                       labels[perm])
    assert torch.equal(load_mask(out + ".mask", g2.num_nodes), mask[perm])
    x = torch.randn(g.num_nodes, 5)
    o1 = F.scatter_gather(x, build_shard(g, 0, 1))
    o2 = F.scatter_gather(x[perm], build_shard(g2, 0, 1))
    assert torch.allclose(o2, o1[perm], atol=1e-5)


def test_bench_reorder_script_cpu_dryrun():
    """scripts/bench_reorder.py degrades to a locality-fraction dry-run
    without a GPU — guard the harness against bit-rot."""
    import roc_amd
    repo = os.path.dirname(os.path.dirname(os.path.abspath(roc_amd.__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts/bench_reorder.py"),
         "--nodes", "5000", "--edges", "50000", "--communities", "4",
         "--orderings", "natural", "shuffled", "cluster"],
        capture_output=True, text=True, timeout=300, cwd=repo)
    assert r.returncode == 0, r.stderr[-1000:]
    assert r.stdout.count("frac_in_64k_window") >= 3


def test_predict_mode(tmp_path):
    """--predict: checkpoint -> resume -> one infer forward -> .npy of
    class ids (train.py inference mode; dropout=identity)."""
    import numpy as np
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ck = str(tmp_path / "ck.pt")
    out = str(tmp_path / "preds.npy")
    base = [sys.executable, os.path.join(repo, "train.py"),
            "--dataset", "cora-synthetic", "--scale", "0.2",
            "--eval-every", "0"]
    r = subprocess.run(base + ["--epochs", "3", "--checkpoint", ck],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    r = subprocess.run(base + ["--resume", ck, "--predict", out],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    assert "[predict] wrote" in r.stdout
    p = np.load(out)
    assert p.dtype == np.int64 and p.min() >= 0 and p.max() < 7


def test_save_best_and_safetensors_export(tmp_path):
    """--save-best keeps the best-val checkpoint (with the pinned val
    metric in extra); --export-safetensors writes loadable weights."""
    import numpy as np
    from safetensors.torch import load_file
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    best = str(tmp_path / "best.pt")
    st = str(tmp_path / "w.safetensors")
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "train.py"),
         "--dataset", "cora-synthetic", "--scale", "0.2",
         "--learnable-labels", "--epochs", "12", "--eval-every", "3",
         "--save-best", best, "--export-safetensors", st],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    assert "[export] wrote" in r.stdout
    state = torch.load(best, weights_only=False)
    assert state["extra"]["best_val_acc"] >= 0
    assert state["extra"]["best_epoch"] % 3 == 0
    weights = load_file(st)
    assert weights and all(v.dtype == torch.float32
                           for v in weights.values())
    ck_names = set(state["model"])
    assert set(weights) == ck_names


def test_log_csv(tmp_path):
    """--log-csv appends a well-formed metrics row per eval."""
    import csv
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    log = str(tmp_path / "m.csv")
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "train.py"),
         "--dataset", "cora-synthetic", "--scale", "0.2",
         "--epochs", "9", "--eval-every", "3", "--log-csv", log],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    with open(log) as f:
        rows = list(csv.DictReader(f))
    assert [int(x["epoch"]) for x in rows] == [3, 6, 9]
    assert all(float(x["ce_loss"]) > 0 for x in rows)
    assert all(float(x["ms_per_epoch"]) > 0 for x in rows)


@pytest.mark.parametrize("model", ["sgc", "appnp"])
def test_new_family_checkpoint_predict(tmp_path, model):
    """SGC/APPNP round-trip through checkpoint -> resume -> --predict."""
    import numpy as np
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    ck = str(tmp_path / "ck.pt")
    out = str(tmp_path / "p.npy")
    base = [sys.executable, os.path.join(repo, "train.py"),
            "--dataset", "cora-synthetic", "--scale", "0.2",
            "--model", model, "--eval-every", "0"]
    r = subprocess.run(base + ["--epochs", "3", "--checkpoint", ck],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    r = subprocess.run(base + ["--resume", ck, "--predict", out],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    p = np.load(out)
    assert p.dtype == np.int64 and p.min() >= 0 and p.max() < 7


@pytest.mark.parametrize("extra", [
    ["--model", "sgc", "--k-hops", "3"],
    ["--model", "appnp", "--k-hops", "4", "--alpha", "0.2"],
])
def test_model_hyperparam_flags(extra):
    """--k-hops / --alpha reach the model constructors."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, os.path.join(repo, "train.py"),
         "--dataset", "cora-synthetic", "--scale", "0.2",
         "--epochs", "2", "--eval-every", "0", "-v"] + extra,
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
