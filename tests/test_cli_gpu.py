"""GPU end-to-end of the user-facing driver CLI (train.py), the path a
reference user actually runs (`gnn.cc:25-179` flag surface): windowed
.lux dataset loading -> shard build -> bf16 training on the HIP kernels
-> checkpoint -> resume -> --predict inference -> chrome trace with
real hipEvent timestamps."""
import json
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

from roc_amd.graph import synthetic_graph, save_lux

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make_dataset(tmp_path, n=600, e=9000, d=16, c=5, seed=3):
    g = synthetic_graph(n, e, seed=seed)
    pref = str(tmp_path / "ds")
    save_lux(pref + ".add_self_edge.lux", g)
    rng = np.random.default_rng(seed)
    rng.standard_normal((n, d)).astype(np.float32).tofile(
        pref + ".feats.bin")
    np.savetxt(pref + ".label", rng.integers(0, c, n), fmt="%d")
    names = np.array(["Train", "Val", "Test", "None"])
    with open(pref + ".mask", "w") as f:
        f.write("\n".join(names[rng.integers(0, 4, n)]) + "\n")
    return pref, d, c


def test_cli_train_resume_predict_gpu(tmp_path):
    pref, d, c = _make_dataset(tmp_path)
    ck = str(tmp_path / "ck.pt")
    preds = str(tmp_path / "preds.npy")
    trace = str(tmp_path / "trace.json")
    env = dict(os.environ, ROC_COMM_MODE="halo")
    base = [sys.executable, os.path.join(REPO, "train.py"),
            "--file", pref, "--layers", f"{d}-32-{c}", "--seed", "2",
            "--eval-every", "3"]
    r = subprocess.run(base + ["--epochs", "6", "--checkpoint", ck,
                               "--trace", trace],
                       capture_output=True, text=True, timeout=420, env=env)
    assert r.returncode == 0, (r.stdout[-500:], r.stderr[-1200:])
    assert "epoch     6" in r.stdout, r.stdout[-800:]

    # chrome trace: real hipEvent-placed GPU spans (begin ts + dur)
    with open(trace) as f:
        tr = json.load(f)
    spans = [ev for ev in tr["traceEvents"] if ev.get("ph") == "X"]
    assert len(spans) > 10
    assert all("ts" in ev and ev["dur"] >= 0 for ev in spans)

    # resume continues the epoch counter and trains further on GPU
    r = subprocess.run(base + ["--epochs", "9", "--resume", ck,
                               "--checkpoint", ck],
                       capture_output=True, text=True, timeout=420, env=env)
    assert r.returncode == 0, r.stderr[-1200:]
    assert "epoch     9" in r.stdout, r.stdout[-800:]

    # inference mode: per-node class ids from the trained checkpoint
    r = subprocess.run(base + ["--resume", ck, "--predict", preds],
                       capture_output=True, text=True, timeout=420, env=env)
    assert r.returncode == 0, r.stderr[-1200:]
    assert "[predict] wrote" in r.stdout
    p = np.load(preds)
    assert p.shape == (600,) and p.dtype == np.int64
    assert p.min() >= 0 and p.max() < c


def test_cli_bf16_uses_native_kernels_gpu(tmp_path):
    """The CLI on a GPU must run the in-tree HIP extension (no silent
    eager fallback): verify roc_amd._C is loaded in the train process
    and the checkpoint weights actually moved."""
    pref, d, c = _make_dataset(tmp_path, seed=9)
    ck = str(tmp_path / "ck.pt")
    code = (
        "import sys, torch; sys.argv = ['train.py', '--file', %r, "
        "'--layers', '%d-32-%d', '--epochs', '2', '--eval-every', '0', "
        "'--checkpoint', %r]; "
        "import train; train.main(); "
        "import roc_amd; assert roc_amd.ops.functional._C is not None; "
        "m = [mod for name, mod in sys.modules.items() "
        "     if name == 'roc_amd._C']; "
        "assert m and '/roc_amd/' in m[0].__file__, m"
        % (pref, d, c, ck)
    )
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=420, cwd=REPO)
    assert r.returncode == 0, (r.stdout[-500:], r.stderr[-1200:])
    st = torch.load(ck, weights_only=False)
    w0 = next(iter(st["model"].values()))
    assert torch.isfinite(w0.float()).all()
