"""Graph container, dataset loaders and synthetic generators.

Replicates the reference's on-disk formats (ROC `load_task.cu:25-269`,
`gnn.cc:751-872`) with an MI355X-native in-memory layout:

- CSR over *in-edges*: ``rowptr[v]..rowptr[v+1]`` indexes the sources
  ``colidx[e]`` of v's in-neighbors (int64 rowptr, int32 colidx).
- A transpose CSR (CSC of the same matrix) is built once at load time so
  the aggregation backward is exact on asymmetric graphs (the reference
  assumes a symmetric graph and reuses the forward kernel,
  `scattergather_kernel.cu:160-170`; we keep that fast path behind the
  ``symmetric`` flag).

File formats (reference parity):
- ``<name>.add_self_edge.lux``: u32 numNodes, u64 numEdges, then numNodes
  u64 *inclusive-end* row offsets, then numEdges u32 source ids.
- ``<name>.feats.csv`` (one row of floats per vertex) with auto-written
  binary cache ``<name>.feats.bin``.
- ``<name>.label``: one int class id per line.
- ``<name>.mask``: one of ``Train|Val|Test|None`` per line.
"""
from __future__ import annotations

import os
import struct
from dataclasses import dataclass, field
from typing import Optional

import numpy as np
import torch

MASK_NONE = 0
MASK_TRAIN = 1
MASK_VAL = 2
MASK_TEST = 3

_MASK_STR = {"None": MASK_NONE, "Train": MASK_TRAIN, "Val": MASK_VAL, "Test": MASK_TEST}
MASK_NAMES = {v: k for k, v in _MASK_STR.items()}


@dataclass
class CSRGraph:
    """In-edge CSR for one whole graph (or one partition's local view)."""

    num_nodes: int
    num_edges: int
    rowptr: torch.Tensor  # int64 [num_nodes + 1]
    colidx: torch.Tensor  # int32 [num_edges]
    symmetric: bool = False
    # transpose (CSC): built lazily unless symmetric
    t_rowptr: Optional[torch.Tensor] = None
    t_colidx: Optional[torch.Tensor] = None
    # cached per-node in-degree (float32, clamped to >= 1 for rsqrt)
    _indegree: Optional[torch.Tensor] = field(default=None, repr=False)

    def indegree(self) -> torch.Tensor:
        if self._indegree is None:
            deg = (self.rowptr[1:] - self.rowptr[:-1]).to(torch.float32)
            self._indegree = deg.clamp_(min=1.0)
        return self._indegree

    def transpose(self) -> "CSRGraph":
        """Return (building if needed) the transpose adjacency as CSR views."""
        if self.symmetric:
            return self
        if self.t_rowptr is None:
            self.t_rowptr, self.t_colidx = build_transpose(
                self.num_nodes, self.rowptr, self.colidx
            )
        return CSRGraph(
            num_nodes=self.num_nodes,
            num_edges=self.num_edges,
            rowptr=self.t_rowptr,
            colidx=self.t_colidx,
            symmetric=False,
        )

    def to(self, device) -> "CSRGraph":
        g = CSRGraph(
            num_nodes=self.num_nodes,
            num_edges=self.num_edges,
            rowptr=self.rowptr.to(device),
            colidx=self.colidx.to(device),
            symmetric=self.symmetric,
        )
        if self.t_rowptr is not None:
            g.t_rowptr = self.t_rowptr.to(device)
            g.t_colidx = self.t_colidx.to(device)
        if self._indegree is not None:
            g._indegree = self._indegree.to(device)
        return g


def build_transpose(num_nodes: int, rowptr: torch.Tensor, colidx: torch.Tensor):
    """CSC of the in-edge CSR: t_rowptr/t_colidx list, for each source u,
    the destinations v that have u as an in-neighbor.

    Works for rectangular local views too (rowptr rows < colidx id space):
    pass num_nodes = size of the *column* id space.

    Uses the native O(E) counting-sort (roc_amd._C) when built; falls back
    to numpy argsort (identical, deterministic output).
    """
    try:
        from roc_amd import _C
        t_rowptr, t_colidx = _C.csr_transpose(
            num_nodes, rowptr.contiguous(), colidx.contiguous())
        return t_rowptr, t_colidx
    except ImportError:
        pass
    nr = rowptr.numel() - 1
    rp = rowptr.numpy()
    ci = colidx.numpy()
    counts = np.bincount(ci, minlength=num_nodes).astype(np.int64)
    t_rowptr = np.zeros(num_nodes + 1, dtype=np.int64)
    np.cumsum(counts, out=t_rowptr[1:])
    # destination of each edge
    dst = np.repeat(np.arange(nr, dtype=np.int32), np.diff(rp))
    order = np.argsort(ci, kind="stable")
    t_colidx = dst[order]
    return torch.from_numpy(t_rowptr), torch.from_numpy(np.ascontiguousarray(t_colidx))


# ---------------------------------------------------------------------------
# Reference on-disk formats (ROC parity)
# ---------------------------------------------------------------------------

def load_lux(path: str) -> CSRGraph:
    """Load a `.lux` CSR graph (header + inclusive-end offsets + col ids).

    Format per reference `gnn.cc:758-801` / `load_task.cu:201-269`.
    """
    with open(path, "rb") as f:
        num_nodes = struct.unpack("<I", f.read(4))[0]
        num_edges = struct.unpack("<Q", f.read(8))[0]
        raw_rows = np.fromfile(f, dtype=np.uint64, count=num_nodes)
        colidx = np.fromfile(f, dtype=np.uint32, count=num_edges)
    assert raw_rows.shape[0] == num_nodes and colidx.shape[0] == num_edges, (
        f"truncated .lux file {path}"
    )
    # inclusive-end offsets -> standard rowptr
    rowptr = np.zeros(num_nodes + 1, dtype=np.int64)
    rowptr[1:] = raw_rows.astype(np.int64)
    assert np.all(np.diff(rowptr) >= 0), "non-monotone row offsets"
    assert rowptr[-1] == num_edges, "row offsets do not end at numEdges"
    return CSRGraph(
        num_nodes=num_nodes,
        num_edges=num_edges,
        rowptr=torch.from_numpy(rowptr),
        colidx=torch.from_numpy(colidx.astype(np.int32)),
    )


def load_lux_meta(path: str):
    """Header + full row-offset array only (numNodes*8 bytes — the part
    every rank needs for partitioning/degrees; the edge list is read
    per-partition with load_lux_cols, reference `load_task.cu:231-243`)."""
    with open(path, "rb") as f:
        num_nodes = struct.unpack("<I", f.read(4))[0]
        num_edges = struct.unpack("<Q", f.read(8))[0]
        raw_rows = np.fromfile(f, dtype=np.uint64, count=num_nodes)
    rowptr = np.zeros(num_nodes + 1, dtype=np.int64)
    rowptr[1:] = raw_rows.astype(np.int64)
    return num_nodes, num_edges, torch.from_numpy(rowptr)


def load_lux_cols(path: str, num_nodes: int, e0: int, e1: int) -> np.ndarray:
    """Windowed read of source ids for edges [e0, e1)."""
    header = 4 + 8 + 8 * num_nodes
    with open(path, "rb") as f:
        f.seek(header + 4 * e0)
        cols = np.fromfile(f, dtype=np.uint32, count=e1 - e0)
    return cols.astype(np.int64)


def load_features_window(path_prefix: str, num_nodes: int, in_dim: int,
                         lo: int, hi: int) -> torch.Tensor:
    """Rows [lo, hi) of the feature matrix from the binary cache
    (falls back to loading+caching the CSV once)."""
    bin_path = path_prefix + ".feats.bin"
    if not os.path.exists(bin_path):
        load_features(path_prefix, num_nodes, in_dim)  # writes the cache
    with open(bin_path, "rb") as f:
        f.seek(4 * in_dim * lo)
        arr = np.fromfile(f, dtype=np.float32, count=(hi - lo) * in_dim)
    return torch.from_numpy(arr.reshape(hi - lo, in_dim).copy())


def save_lux(path: str, g: CSRGraph) -> None:
    with open(path, "wb") as f:
        f.write(struct.pack("<I", g.num_nodes))
        f.write(struct.pack("<Q", g.num_edges))
        g.rowptr[1:].numpy().astype(np.uint64).tofile(f)
        g.colidx.numpy().astype(np.uint32).tofile(f)


def load_features(path_prefix: str, num_nodes: int, in_dim: int) -> torch.Tensor:
    """CSV features with a binary cache (reference `load_task.cu:39-73`)."""
    bin_path = path_prefix + ".feats.bin"
    csv_path = path_prefix + ".feats.csv"
    if os.path.exists(bin_path):
        arr = np.fromfile(bin_path, dtype=np.float32, count=num_nodes * in_dim)
    else:
        arr = np.loadtxt(csv_path, delimiter=",", dtype=np.float32)
        arr = np.ascontiguousarray(arr, dtype=np.float32)
        arr.tofile(bin_path)
    return torch.from_numpy(arr.reshape(num_nodes, in_dim).copy())


def load_labels(path: str, num_nodes: int) -> torch.Tensor:
    labels = np.loadtxt(path, dtype=np.int64).reshape(-1)
    assert labels.shape[0] == num_nodes
    return torch.from_numpy(labels)


def load_mask(path: str, num_nodes: int) -> torch.Tensor:
    with open(path) as f:
        vals = [_MASK_STR[line.strip()] for line in f if line.strip()]
    assert len(vals) == num_nodes
    return torch.tensor(vals, dtype=torch.int32)


def reorder_graph(g: CSRGraph, perm: torch.Tensor) -> CSRGraph:
    """Relabel nodes by `perm` (new_id = position of old_id in perm):
    rows are permuted and column ids rewritten. Use with a locality
    ordering (degree sort, clustering) before partitioning real graphs —
    contiguous vertex ranges then cut fewer edges. Apply the same perm to
    features/labels/masks: x_new = x_old[perm]."""
    try:
        from roc_amd import _C
        new_rp, new_ci = _C.csr_permute(g.rowptr.contiguous(),
                                        g.colidx.contiguous(),
                                        perm.to(torch.int64).contiguous())
        return CSRGraph(num_nodes=g.num_nodes, num_edges=g.num_edges,
                        rowptr=new_rp, colidx=new_ci)
    except ImportError:
        pass
    perm_np = perm.numpy()
    inv = np.empty_like(perm_np)
    inv[perm_np] = np.arange(g.num_nodes, dtype=perm_np.dtype)
    rp = g.rowptr.numpy()
    deg = np.diff(rp)[perm_np]
    new_rp = np.zeros(g.num_nodes + 1, dtype=np.int64)
    np.cumsum(deg, out=new_rp[1:])
    # vectorized edge permutation: new edge i maps to
    # rp[perm[row(i)]] + (i - new_rp[row(i)])
    old_edge = (np.repeat(rp[perm_np], deg)
                + np.arange(g.num_edges, dtype=np.int64)
                - np.repeat(new_rp[:-1], deg))
    new_ci = inv[g.colidx.numpy()[old_edge]].astype(np.int32)
    out = CSRGraph(num_nodes=g.num_nodes, num_edges=g.num_edges,
                   rowptr=torch.from_numpy(new_rp),
                   colidx=torch.from_numpy(np.ascontiguousarray(new_ci)))
    try:
        from roc_amd import _C
        _C.csr_sort_rows(out.rowptr, out.colidx)
    except ImportError:
        pass
    return out


def degree_order(g: CSRGraph) -> torch.Tensor:
    """Degree-descending relabeling permutation (hubs first — a cheap
    locality ordering for power-law graphs)."""
    deg = (g.rowptr[1:] - g.rowptr[:-1]).numpy()
    return torch.from_numpy(np.argsort(-deg, kind="stable"))


def rcm_order(g: CSRGraph) -> torch.Tensor:
    """Reverse Cuthill-McKee relabeling permutation over the symmetrized
    adjacency (in-edges + out-edges). Clusters each vertex next to its
    neighbors, so (a) SpMM gathers hit an L2-sized source window and
    (b) contiguous vertex-range partitions cut fewer edges (sparser
    halos). Native BFS in ``_C.rcm_order`` (O(E) + per-front sort);
    scipy fallback gives the same class of ordering."""
    t_rowptr, t_colidx = build_transpose(g.num_nodes, g.rowptr, g.colidx)
    try:
        from roc_amd import _C
        return _C.rcm_order(g.rowptr.contiguous(), g.colidx.contiguous(),
                            t_rowptr.contiguous(), t_colidx.contiguous())
    except ImportError:
        from scipy.sparse import csr_matrix
        from scipy.sparse.csgraph import reverse_cuthill_mckee
        a = csr_matrix((np.ones(g.num_edges, dtype=np.int8),
                        g.colidx.numpy(), g.rowptr.numpy()),
                       shape=(g.num_nodes, g.num_nodes))
        return torch.from_numpy(
            reverse_cuthill_mckee(a, symmetric_mode=False).astype(np.int64))


def cluster_order(g: CSRGraph, iters: int = 25) -> torch.Tensor:
    """Label-propagation clustering relabeling: each vertex repeatedly
    adopts its in-neighbors' majority label, then nodes are stably
    sorted by final label — communities become contiguous id ranges.
    The strongest of the three orderings on community graphs (see
    profiles/): gathers land in an L2-sized window and contiguous
    partitions cut few edges. Native OpenMP (``_C.lp_cluster_order``)."""
    from roc_amd import _C
    return _C.lp_cluster_order(g.rowptr.contiguous(), g.colidx.contiguous(),
                               int(iters))


ORDERINGS = {"degree": degree_order, "rcm": rcm_order,
             "cluster": cluster_order}


def apply_ordering(g: CSRGraph, feats: torch.Tensor, labels: torch.Tensor,
                   mask: torch.Tensor, kind: str):
    """Relabel the whole dataset by a named locality ordering ("degree",
    "rcm", "cluster", or "auto"); returns (graph, feats, labels, mask,
    perm — None if "auto" kept the original). Deterministic, so every
    rank computes the identical permutation. Training on the relabeled
    dataset is mathematically the same full-graph problem (permutation
    equivariance).

    "auto": try the LP-cluster ordering and keep it only if it
    meaningfully improves the sampled gather locality (the signal the
    SpMM schedule and the halo plan actually respond to — profiles/r16);
    a genuinely locality-free graph is returned unchanged."""
    if kind == "auto":
        from .parallel.partition import _gather_locality
        before = _gather_locality(g.rowptr.numpy(), g.colidx.numpy())
        perm = cluster_order(g)
        g2 = reorder_graph(g, perm)
        after = _gather_locality(g2.rowptr.numpy(), g2.colidx.numpy())
        if after < 0.5 or after < 2 * before:
            return g, feats, labels, mask, None
        return g2, feats[perm], labels[perm], mask[perm], perm
    perm = ORDERINGS[kind](g)
    return (reorder_graph(g, perm), feats[perm], labels[perm], mask[perm],
            perm)


# ---------------------------------------------------------------------------
# Synthetic graphs (no-network benchmark datasets; BASELINE.json configs)
# ---------------------------------------------------------------------------

def synthetic_graph(
    num_nodes: int,
    num_edges: int,
    seed: int = 1,
    skew: float = 1.0,
    add_self_edges: bool = True,
    locality: float = 0.0,
    num_communities: int = 64,
) -> CSRGraph:
    """Random directed graph with lognormal in-degree skew, self-edges added.

    ``locality`` in [0,1): that fraction of edges stays inside the node's
    own contiguous community block (planted-partition structure — real
    graphs are community-heavy, and contiguous communities make the
    vertex partition's halo SPARSE, exercising the halo/a2a strategy
    instead of allgather).
    Column ids are sorted within each row (better gather locality, and the
    reference's CSR is sorted the same way after construction).
    """
    rng = np.random.default_rng(seed)
    n, e = int(num_nodes), int(num_edges)
    if add_self_edges:
        e_rand = max(e - n, 0)
    else:
        e_rand = e
    # lognormal degree profile normalized to e_rand total
    w = rng.lognormal(mean=0.0, sigma=skew, size=n)
    w = w / w.sum()
    deg = rng.multinomial(e_rand, w).astype(np.int64)
    rowptr = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(deg + (1 if add_self_edges else 0), out=rowptr[1:])
    total = int(rowptr[-1])
    colidx = rng.integers(0, n, size=total, dtype=np.int64)
    if locality > 0.0:
        k = max(int(num_communities), 1)
        block = (n + k - 1) // k
        row_of_edge_l = np.repeat(np.arange(n, dtype=np.int64),
                                  np.diff(rowptr))
        local_sel = rng.random(total) < locality
        base = (row_of_edge_l // block) * block
        width = np.minimum(base + block, n) - base
        colidx[local_sel] = (base[local_sel] +
                             (colidx[local_sel] % width[local_sel]))
    if add_self_edges:
        # overwrite one slot per row with the self edge; then sort rows
        colidx[rowptr[:-1]] = np.arange(n, dtype=np.int64)
    # sort columns within each row via composite key
    row_of_edge = np.repeat(np.arange(n, dtype=np.int64), np.diff(rowptr))
    key = row_of_edge * n + colidx
    key.sort(kind="stable")
    colidx = (key % n).astype(np.int32)
    return CSRGraph(
        num_nodes=n,
        num_edges=total,
        rowptr=torch.from_numpy(rowptr),
        colidx=torch.from_numpy(colidx),
    )


# Named shapes from BASELINE.json (synthetic stand-ins; there is no network
# for real datasets). Node/edge counts follow the reference workloads.
DATASET_SHAPES = {
    # name: (num_nodes, num_edges incl. self, in_dim, num_classes)
    "cora": (2_708, 13_264, 1_433, 7),
    "reddit": (232_965, 114_848_857, 602, 41),
    "amazon": (1_569_960, 132_954_714, 200, 107),
    "ogbn-products": (2_449_029, 126_167_053, 100, 47),
    "papers100M": (111_059_956, 1_726_745_828, 128, 172),
    # scaled-down stand-in for the papers100M offload config (fits CI boxes)
    "papers-synth-small": (4_000_000, 240_000_000, 128, 172),
}


def _agg_norm(g: CSRGraph, z: torch.Tensor) -> torch.Tensor:
    """CPU symmetric-normalized aggregation D^-1/2 A D^-1/2 z (chunked)."""
    rs = g.indegree().rsqrt()
    z = z * rs.unsqueeze(1)
    out = torch.zeros_like(z)
    rp = g.rowptr.numpy()
    dst_all = torch.repeat_interleave(
        torch.arange(g.num_nodes, dtype=torch.int64),
        torch.from_numpy(np.diff(rp)))
    ci = g.colidx.to(torch.int64)
    step = 30_000_000  # chunk the 10^8-edge gather
    for s0 in range(0, g.num_edges, step):
        s1 = min(g.num_edges, s0 + step)
        out.index_add_(0, dst_all[s0:s1], z[ci[s0:s1]])
    return out * rs.unsqueeze(1)


def _learnable_labels(g: CSRGraph, feats: torch.Tensor, c: int,
                      seed: int) -> torch.Tensor:
    """Labels from a random bias-free 2-layer GCN teacher — EXACTLY the
    student's function class, so train AND val accuracy can rise
    (uniform random labels carry no learnable signal)."""
    rng = np.random.default_rng(seed + 7)
    d = feats.shape[1]
    hid = 32
    w1 = torch.from_numpy(
        (rng.standard_normal((d, hid)) / np.sqrt(d)).astype(np.float32))
    w2 = torch.from_numpy(
        (rng.standard_normal((hid, c)) / np.sqrt(hid)).astype(np.float32))
    h = torch.relu(_agg_norm(g, feats @ w1))
    logits = _agg_norm(g, h @ w2)
    return logits.argmax(dim=1)


def synthetic_dataset(name: str, seed: int = 1, scale: float = 1.0,
                      learnable_labels: bool = False, locality: float = 0.0,
                      num_communities: int = 64):
    """Graph + features + labels + masks of the named shape.

    ``scale`` < 1 shrinks nodes/edges proportionally (for tests).
    ``learnable_labels`` draws labels from a random one-hop teacher
    instead of uniformly (same shapes/cost; lets accuracy actually rise).
    ``locality``/``num_communities``: planted community structure (see
    ``synthetic_graph``) — pair with a random relabel + ``--reorder`` to
    study ordering effects, or use directly for sparse-halo runs.
    Returns (graph, features fp32 [N, in_dim], labels int64 [N], mask int32 [N]).
    """
    n, e, d, c = DATASET_SHAPES[name]
    n = max(int(n * scale), 16)
    e = max(int(e * scale), n)
    g = synthetic_graph(n, e, seed=seed, locality=locality,
                        num_communities=num_communities)
    rng = np.random.default_rng(seed + 1)
    feats = torch.from_numpy(rng.standard_normal((n, d), dtype=np.float32))
    if learnable_labels:
        labels = _learnable_labels(g, feats, c, seed)
    else:
        labels = torch.from_numpy(rng.integers(0, c, size=n).astype(np.int64))
    # 70/15/15 split like common full-graph benchmarks
    u = rng.random(n)
    mask = np.full(n, MASK_TRAIN, dtype=np.int32)
    mask[u >= 0.70] = MASK_VAL
    mask[u >= 0.85] = MASK_TEST
    return g, feats, labels, torch.from_numpy(mask), c
