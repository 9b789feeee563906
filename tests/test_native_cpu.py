"""Native C++ graph machinery (roc_amd._C CPU functions) vs numpy."""
import numpy as np
import pytest
import os

import torch

from roc_amd.graph import synthetic_graph

_C = pytest.importorskip("roc_amd._C")


def numpy_transpose(num_cols, rowptr, colidx):
    rp = rowptr.numpy()
    ci = colidx.numpy()
    counts = np.bincount(ci, minlength=num_cols).astype(np.int64)
    t_rowptr = np.zeros(num_cols + 1, dtype=np.int64)
    np.cumsum(counts, out=t_rowptr[1:])
    dst = np.repeat(np.arange(rp.shape[0] - 1, dtype=np.int32), np.diff(rp))
    order = np.argsort(ci, kind="stable")
    return t_rowptr, dst[order]


@pytest.mark.parametrize("n,e", [(100, 900), (500, 8000)])
def test_csr_transpose_matches_numpy(n, e):
    g = synthetic_graph(n, e, seed=17)
    trp, tci = _C.csr_transpose(n, g.rowptr, g.colidx)
    nrp, nci = numpy_transpose(n, g.rowptr, g.colidx)
    assert np.array_equal(trp.numpy(), nrp)
    assert np.array_equal(tci.numpy(), nci)


def test_csr_transpose_rectangular():
    # local view: 10 rows, columns in a 30-id ext space
    rowptr = torch.tensor([0, 2, 4, 6, 8, 10, 12, 14, 16, 18, 20],
                          dtype=torch.int64)
    colidx = torch.arange(20, dtype=torch.int32) % 30
    trp, tci = _C.csr_transpose(30, rowptr, colidx)
    assert trp.numel() == 31
    assert trp[-1].item() == 20
    nrp, nci = numpy_transpose(30, rowptr, colidx)
    assert np.array_equal(trp.numpy(), nrp)
    assert np.array_equal(tci.numpy(), nci)


def test_csr_sort_rows():
    g = synthetic_graph(50, 600, seed=3)
    ci = g.colidx.clone()
    perm = torch.randperm(ci.numel())
    # shuffle within the whole array then re-sort rows
    shuffled = ci.clone()
    _C.csr_sort_rows(g.rowptr, shuffled)
    rp = g.rowptr.numpy()
    s = shuffled.numpy()
    for v in range(50):
        assert np.all(np.diff(s[rp[v]:rp[v + 1]]) >= 0)


def test_divergence_guard():
    from roc_amd.debug import check_metrics, TrainingDiverged
    check_metrics(torch.ones(8))
    with pytest.raises(TrainingDiverged):
        check_metrics(torch.tensor([1.0, float("nan"), 0, 0, 0, 0, 0, 0]))


def test_sync_debug_mode():
    from roc_amd.debug import (enable_sync_debug, check_tensor,
                               sync_debug_enabled, TrainingDiverged)
    enable_sync_debug(True)
    try:
        assert sync_debug_enabled()
        check_tensor(torch.ones(4), "ok-tensor")
        with pytest.raises(TrainingDiverged):
            check_tensor(torch.tensor([1.0, float("inf")]), "bad-tensor")
    finally:
        enable_sync_debug(False)
    assert not sync_debug_enabled()


def test_knobs_doc_covers_all_env_vars():
    """docs/KNOBS.md must document every ROC_* environment variable the
    code actually reads (doc-rot guard)."""
    import re
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    read_pat = re.compile(
        r'(?:environ(?:\.get)?\s*[\[(]\s*"|getenv\s*\(\s*")(ROC_[A-Z0-9_]+)')
    used = set()
    for root, _dirs, files in os.walk(os.path.join(repo, "roc_amd")):
        for fn in files:
            if fn.endswith((".py", ".hip", ".cpp", ".h")):
                with open(os.path.join(root, fn), errors="ignore") as f:
                    used |= set(read_pat.findall(f.read()))
    for fn in ("train.py", "bench.py"):
        with open(os.path.join(repo, fn)) as f:
            used |= set(read_pat.findall(f.read()))
    with open(os.path.join(repo, "docs", "KNOBS.md")) as f:
        documented = set(re.findall(r"ROC_[A-Z0-9_]+", f.read()))
    missing = used - documented
    assert not missing, f"undocumented env knobs: {sorted(missing)}"


def test_profiles_index_covers_all_reports():
    """profiles/INDEX.md must reference every rNN report file (keeps the
    evidence chain navigable)."""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    pdir = os.path.join(repo, "profiles")
    with open(os.path.join(pdir, "INDEX.md")) as f:
        idx = f.read()
    missing = [fn for fn in sorted(os.listdir(pdir))
               if fn.startswith("r") and fn != "INDEX.md"
               and fn not in idx
               and fn.split(".")[0] not in idx]  # stem listing counts
    assert not missing, f"unindexed profile reports: {missing}"
