"""Vertex partitioner + halo plan — the explicit scheduler replacing the
reference's Legion mapper / implicit region movement.

The reference replicates the ENTIRE input tensor into every GPU through
pinned host memory per aggregation (`scattergather.cc:68-73`,
`types.cu:22-32`). Here each rank owns a contiguous vertex range
(edge-balanced, like `gnn.cc:804-870`), keeps its activations resident in
HBM, and exchanges only boundary-vertex rows ("halo") over RCCL/xGMI
before each aggregation. The plan below precomputes, once per graph:

- local CSR with columns remapped into [0, n_local + n_halo)
- the transpose CSR (for an exact backward on asymmetric graphs)
- per-destination send index lists (for all_to_all_single)
- degree-scale vectors for fused GCN normalization
- degree-descending row orders (skew-tolerant kernel scheduling)
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List, Optional

import numpy as np
import torch

from ..graph import CSRGraph, build_transpose


def _gather_locality(rowptr: np.ndarray, colidx: np.ndarray,
                     max_sample: int = 1_000_000) -> float:
    """Sampled fraction of edges whose source lies within a 16k-row
    window of the destination (±8192 rows ≈ one XCD's 4 MiB L2 at
    D=256 bf16). High → the graph is community-ordered and the natural
    row schedule keeps consecutive workgroups in the same L2 window."""
    ne = int(colidx.shape[0])
    if ne == 0:
        return 0.0
    if ne > max_sample:
        e = np.linspace(0, ne - 1, max_sample).astype(np.int64)
    else:
        e = np.arange(ne, dtype=np.int64)
    rows = np.searchsorted(rowptr, e, side="right") - 1
    return float((np.abs(rows - colidx[e].astype(np.int64)) < 8192).mean())


@dataclass
class GraphShard:
    """Everything one rank needs to run all graph ops locally."""

    rank: int
    world_size: int
    bounds: List[int]            # global partition bounds, len world_size+1
    lo: int
    hi: int
    n_local: int
    n_halo: int
    n_ext: int                   # n_local + n_halo
    rowptr: torch.Tensor         # int64 [n_local+1]
    colidx: torch.Tensor         # int32 [local_edges], values < n_ext
    t_rowptr: torch.Tensor       # int64 [n_ext+1]
    t_colidx: torch.Tensor       # int32 [local_edges], values < n_local
    deg_local: torch.Tensor      # fp32 global in-degree of local rows (>=1)
    rsqrt_deg_local: torch.Tensor
    inv_deg_local: torch.Tensor
    rsqrt_deg_ext: torch.Tensor  # fp32 [n_ext] (local ++ halo)
    row_order: Optional[torch.Tensor]    # int32 [n_local] degree-desc, or
    t_row_order: Optional[torch.Tensor]  # None = natural schedule (r16)
    halo_ids: torch.Tensor       # int64 [n_halo] global ids (grouped by owner)
    recv_splits: List[int]       # rows received from each rank
    send_idx: torch.Tensor       # int64 local indices to send (concat by dest)
    send_splits: List[int]       # rows sent to each rank
    # --- comm/compute overlap split (world_size > 1 only): local-source
    # edges aggregate while the halo all_to_all is in flight -------------
    loc_rowptr: Optional[torch.Tensor] = None   # [n_local+1], cols < n_local
    loc_colidx: Optional[torch.Tensor] = None
    halo_rowptr: Optional[torch.Tensor] = None  # [n_local+1], cols < n_halo
    halo_colidx: Optional[torch.Tensor] = None  # (halo-local ids)
    t_loc_rowptr: Optional[torch.Tensor] = None   # [n_local+1]
    t_loc_colidx: Optional[torch.Tensor] = None
    t_halo_rowptr: Optional[torch.Tensor] = None  # [n_halo+1]
    t_halo_colidx: Optional[torch.Tensor] = None
    # --- exchange strategy (parallel/aggregate.py) ----------------------
    comm_mode: str = "none"        # none | halo | allgather
    halo_fraction: float = 0.0     # n_halo / remote rows
    ag_max_rows: int = 0           # padded block size for all_gather
    ag_colidx: Optional[torch.Tensor] = None     # cols in gather space
    ag_t_rowptr: Optional[torch.Tensor] = None   # [ws*ag_max_rows+1]
    ag_t_colidx: Optional[torch.Tensor] = None
    ag_t_row_order: Optional[torch.Tensor] = None
    # self/remote split of the gather-space CSR: edges whose source this
    # rank owns aggregate straight from local x while the all_gather is
    # in flight, then remote-source edges accumulate from the gathered
    # buffer (the allgather-mode analogue of the halo loc/halo split)
    ag_self_rowptr: Optional[torch.Tensor] = None  # [n_local+1], cols<n_local
    ag_self_colidx: Optional[torch.Tensor] = None
    ag_rem_rowptr: Optional[torch.Tensor] = None   # [n_local+1], gather space
    ag_rem_colidx: Optional[torch.Tensor] = None
    ag_self_row_order: Optional[torch.Tensor] = None  # degree-desc, int32
    ag_rem_row_order: Optional[torch.Tensor] = None
    ag_t_blk_order: Optional[torch.Tensor] = None  # [ws*mr] block-local
    # --- source-strip-blocked SpMM plans (L2-resident gather windows;
    # build_strip_plan; large local CSRs only) ------------------------
    fwd_strips: Optional[list] = None  # [(rowptr, colidx), ...]
    bwd_strips: Optional[list] = None  # same, for the transpose CSR

    def to(self, device) -> "GraphShard":
        d = {}
        for k, v in self.__dict__.items():
            if k.startswith("_"):  # lazy caches don't survive the move
                continue
            if isinstance(v, torch.Tensor):
                v = v.to(device)
            elif isinstance(v, list) and v and isinstance(v[0], tuple):
                v = [tuple(t.to(device) for t in tup) for tup in v]
            d[k] = v
        return GraphShard(**d)

    @property
    def has_overlap_split(self) -> bool:
        return self.loc_rowptr is not None

    @property
    def num_local_edges(self) -> int:
        """Size of this rank's edge window — edge tensors are
        [num_local_edges, ...] in the local CSR's edge order
        (reference edge-range partition, `gnn.cc:545-589`)."""
        return int(self.colidx.numel())

    def colidx_long(self) -> torch.Tensor:
        """int64 view of colidx (lazily cached — torch index ops need
        int64 and a fresh .long() would allocate E*8 B per call)."""
        c = getattr(self, "_colidx_long", None)
        if c is None:
            c = self.colidx.long()
            object.__setattr__(self, "_colidx_long", c)
        return c

    def row_of_edge(self) -> torch.Tensor:
        """int64 [E_local]: destination row of each local edge (lazily
        cached; does not survive .to())."""
        r = getattr(self, "_row_of_edge", None)
        if r is None:
            deg = (self.rowptr[1:] - self.rowptr[:-1]).long()
            r = torch.repeat_interleave(
                torch.arange(self.n_local, dtype=torch.long,
                             device=self.rowptr.device), deg)
            object.__setattr__(self, "_row_of_edge", r)
        return r

    def t_edge_perm(self) -> torch.Tensor:
        """int64 [E_local]: for transposed edge j, its index in the
        forward edge order (both the native counting-sort transpose and
        the numpy/torch stable argsort produce the same stable order).
        Lets edge values ride the backward SpMM: t_val = val[perm].
        Lazily built and cached; the cache does not survive .to()."""
        p = getattr(self, "_t_eperm", None)
        if p is None:
            p = torch.argsort(self.colidx.long(), stable=True)
            object.__setattr__(self, "_t_eperm", p)
        return p


def build_strip_plan(rowptr: torch.Tensor, colidx: torch.Tensor,
                     num_cols: int, width: int):
    """Split a CSR into ceil(num_cols/width) source-strip CSRs: strip s
    holds each row's edges whose source id lies in [s*width,(s+1)*width).
    Columns are sorted within rows, so every strip segment is a
    contiguous row slice — one composite-key searchsorted per boundary.

    Measured (profiles/r21): at width 16384 (≈8 MB of D=256 bf16 rows,
    two XCD L2s) the multi-pass schedule runs the Reddit D=256 SpMM
    1.39x faster than one pass — sources stay L2-resident per strip.
    Partials accumulate in an fp32 buffer (mixed-output kernel), so
    numerics are exactly one rounding, same as single-pass."""
    rp = rowptr.numpy()
    ci = colidx.numpy()
    nrows = rp.shape[0] - 1
    ci64 = ci.astype(np.int64)
    row = np.repeat(np.arange(nrows, dtype=np.int64), np.diff(rp))
    key = row * (num_cols + 1) + ci64
    rows_q = np.arange(nrows, dtype=np.int64) * (num_cols + 1)
    pos = [rp[:-1].astype(np.int64)]
    for b in range(width, num_cols, width):
        pos.append(np.searchsorted(key, rows_q + b))
    pos.append(rp[1:].astype(np.int64))
    strips = []
    for s in range(len(pos) - 1):
        seg_start, seg_end = pos[s], pos[s + 1]
        cnt = seg_end - seg_start
        srp = np.zeros(nrows + 1, dtype=np.int64)
        np.cumsum(cnt, out=srp[1:])
        if srp[-1]:
            idx = np.repeat(seg_start - srp[:-1], cnt) + np.arange(
                int(srp[-1]), dtype=np.int64)
            sci = np.ascontiguousarray(ci[idx])
        else:
            sci = np.empty(0, dtype=ci.dtype)
        strips.append((torch.from_numpy(srp), torch.from_numpy(sci)))
    return strips


def edge_tensor(shard: GraphShard, dim: Optional[int] = None,
                dtype=torch.float32, init: str = "zeros") -> torch.Tensor:
    """Per-edge tensor for this rank's edge window, aligned with the
    local CSR edge order (the analog of the reference's
    `create_edge_tensor`, `gnn.cc:475-623`: EDGE-typed tensors
    partitioned by the graph's edge ranges; its live driver never
    consumed them — here `ops.functional.scatter_gather_weighted` does).

    dim=None -> [E_local] (scalar per edge); else [E_local, dim].
    init: "zeros" | "ones" | "gcn_norm" (1/sqrt(deg_dst*deg_src) per
    edge — with these weights the weighted aggregation equals the fused
    symmetric-norm GCN path; scalar only).
    """
    e = shard.num_local_edges
    dev = shard.colidx.device
    shape = (e,) if dim is None else (e, dim)
    if init == "zeros":
        return torch.zeros(*shape, dtype=dtype, device=dev)
    if init == "ones":
        return torch.ones(*shape, dtype=dtype, device=dev)
    if init == "gcn_norm":
        assert dim is None, "gcn_norm is a scalar-per-edge init"
        deg = (shard.rowptr[1:] - shard.rowptr[:-1])
        row_of_edge = torch.repeat_interleave(
            torch.arange(shard.n_local, dtype=torch.long, device=dev), deg)
        val = (shard.rsqrt_deg_local[row_of_edge]
               * shard.rsqrt_deg_ext[shard.colidx.long()])
        return val.to(dtype)
    raise ValueError(f"unknown edge_tensor init {init!r}")


def edge_balanced_bounds(rowptr: torch.Tensor, num_parts: int) -> List[int]:
    """Contiguous vertex ranges with ~equal in-edge counts
    (reference `gnn.cc:806-829` greedy pack; here via searchsorted)."""
    rp = rowptr.numpy()
    e = int(rp[-1])
    n = rp.shape[0] - 1
    targets = [(e * k) // num_parts for k in range(num_parts + 1)]
    bounds = np.searchsorted(rp, targets, side="left").tolist()
    bounds[0], bounds[-1] = 0, n
    # ensure monotone non-decreasing (degenerate tiny graphs)
    for i in range(1, len(bounds)):
        bounds[i] = max(bounds[i], bounds[i - 1])
    return bounds


def fit_cost_model(samples: List[tuple]) -> tuple:
    """Least-squares fit of per-rank epoch time ≈ a*edges + b*halo_rows
    over measured (edges, halo, time) samples (accumulated across
    rebalance rounds — the MLSys'20 online-regression partitioner idea).
    Falls back to the pure-edge model (b=0) when the system is
    underdetermined or the comm coefficient comes out negative."""
    A = np.array([[s[0], s[1]] for s in samples], dtype=np.float64)
    y = np.array([s[2] for s in samples], dtype=np.float64)
    a = float((A[:, 0] * y).sum() / max((A[:, 0] ** 2).sum(), 1e-30))
    b = 0.0
    if len(samples) >= 3 and np.linalg.matrix_rank(A) == 2:
        coef, *_ = np.linalg.lstsq(A, y, rcond=None)
        if coef[0] > 0 and coef[1] >= 0:
            a, b = float(coef[0]), float(coef[1])
    return a, b


def rebalance_bounds_comm(rowptr: torch.Tensor, bounds: List[int],
                          per_rank_time: List[float],
                          per_rank_halo: List[int],
                          samples: Optional[List[tuple]] = None) -> List[int]:
    """Comm-aware refinement: fit t_k ≈ a*edges_k + b*halo_k, then give
    each rank an edge budget e'_k = (T - b*halo_k)/a so PREDICTED totals
    equalize (halo_k of the new range approximated by the current one —
    a fixed point the repeated rebalance iterates toward). `samples`
    (mutated in place) carries measurement history across calls."""
    rp = rowptr.numpy().astype(np.float64)
    p = len(bounds) - 1
    edges = [rp[bounds[k + 1]] - rp[bounds[k]] for k in range(p)]
    cur = [(edges[k], float(per_rank_halo[k]), per_rank_time[k])
           for k in range(p)]
    if samples is not None:
        samples.extend(cur)
        cur = samples[-8 * p:]  # bounded history window
    a, b = fit_cost_model(cur)
    T = sum(per_rank_time) / p
    e_total = rp[-1]
    floor = 0.05 * e_total / p
    budget = [max((T - b * per_rank_halo[k]) / max(a, 1e-30), floor)
              for k in range(p)]
    scale = e_total / sum(budget)
    targets = np.cumsum([0.0] + [bk * scale for bk in budget])
    nb = np.searchsorted(rp, targets, side="left").tolist()
    nb[0], nb[-1] = 0, rowptr.numel() - 1
    for i in range(1, len(nb)):
        nb[i] = max(nb[i], nb[i - 1])
    return nb


def rebalance_bounds(rowptr: torch.Tensor, bounds: List[int],
                     per_rank_time: List[float]) -> List[int]:
    """Cost-model refinement (the MLSys'20 Roc idea the reference code lacks):
    re-split so each rank's predicted edge load is proportional to its
    measured throughput (edges_k / time_k)."""
    rp = rowptr.numpy().astype(np.float64)
    p = len(bounds) - 1
    edges = [rp[bounds[k + 1]] - rp[bounds[k]] for k in range(p)]
    thr = [edges[k] / max(per_rank_time[k], 1e-9) for k in range(p)]
    total_thr = sum(thr)
    e_total = rp[-1]
    targets = np.cumsum([0.0] + [e_total * t / total_thr for t in thr])
    nb = np.searchsorted(rp, targets, side="left").tolist()
    nb[0], nb[-1] = 0, rowptr.numel() - 1
    for i in range(1, len(nb)):
        nb[i] = max(nb[i], nb[i - 1])
    return nb


def _send_plan_comm(halo_ids: np.ndarray, recv_splits: List[int], lo: int,
                    world_size: int, group):
    """Exchange halo REQUESTS over the process group: each rank learns
    which of its rows the others need without scanning their edge
    windows (required for windowed dataset loading at scale). The
    request tensors are CPU int64, so this rides the gloo control
    plane (cpu_group) — the training group is typically RCCL, which
    cannot carry CPU tensors."""
    import torch.distributed as dist

    from .comm import cpu_group
    group = cpu_group(group)
    counts_in = torch.tensor(recv_splits, dtype=torch.int64)
    counts_out = torch.empty(world_size, dtype=torch.int64)
    dist.all_to_all_single(counts_out, counts_in, group=group)
    send_splits = [int(c) for c in counts_out]
    req = torch.empty(sum(send_splits), dtype=torch.int64)
    dist.all_to_all_single(req, torch.from_numpy(halo_ids.astype(np.int64)),
                           output_split_sizes=send_splits,
                           input_split_sizes=recv_splits, group=group)
    return (req - lo), send_splits


def build_shard(g: CSRGraph, rank: int, world_size: int,
                bounds: Optional[List[int]] = None,
                use_comm: bool = False, group=None) -> GraphShard:
    """Build this rank's shard from the full graph. The halo SEND plan
    comes either from scanning every rank's edge window (deterministic,
    no process group needed) or — with use_comm=True under an initialized
    torch.distributed group — from an all_to_all of halo requests (each
    rank touches only its own window; scales to windowed file loading)."""
    if bounds is None:
        bounds = edge_balanced_bounds(g.rowptr, world_size)
    lo, hi = bounds[rank], bounds[rank + 1]
    rp = g.rowptr.numpy()
    e0, e1 = int(rp[lo]), int(rp[hi])
    local_cols_global = g.colidx.numpy()[e0:e1].astype(np.int64)
    return build_shard_from_window(
        g.rowptr, local_cols_global, rank, world_size, bounds,
        full_colidx=None if use_comm else g.colidx, group=group)


def build_shard_from_lux(path: str, rank: int, world_size: int,
                         bounds: Optional[List[int]] = None,
                         group=None) -> GraphShard:
    """Windowed shard construction straight from a .lux file: this rank
    reads only its own edge window (reference `load_task.cu:231-243`);
    the send plan rides the process group (world_size > 1)."""
    from ..graph import load_lux_meta, load_lux_cols
    num_nodes, num_edges, rowptr = load_lux_meta(path)
    if bounds is None:
        bounds = edge_balanced_bounds(rowptr, world_size)
    lo, hi = bounds[rank], bounds[rank + 1]
    rp = rowptr.numpy()
    cols = load_lux_cols(path, num_nodes, int(rp[lo]), int(rp[hi]))
    return build_shard_from_window(rowptr, cols, rank, world_size, bounds,
                                   full_colidx=None, group=group)


def build_shard_from_window(rowptr_full: torch.Tensor,
                            local_cols_global: np.ndarray, rank: int,
                            world_size: int, bounds: List[int],
                            full_colidx: Optional[torch.Tensor] = None,
                            group=None) -> GraphShard:
    lo, hi = bounds[rank], bounds[rank + 1]
    n_local = hi - lo
    rp = rowptr_full.numpy()
    e0 = int(rp[lo])
    local_rowptr = (rp[lo:hi + 1] - e0).astype(np.int64)
    local_cols_global = local_cols_global.astype(np.int64, copy=False)

    # halo: remote source vertices, grouped by owning rank then sorted by id
    is_local = (local_cols_global >= lo) & (local_cols_global < hi)
    remote = np.unique(local_cols_global[~is_local])
    owners = np.searchsorted(bounds, remote, side="right") - 1
    order = np.lexsort((remote, owners))
    halo_ids = remote[order]
    halo_owners = owners[order]
    recv_splits = np.bincount(halo_owners, minlength=world_size).tolist()
    n_halo = halo_ids.shape[0]

    # remap columns -> ext index space [0, n_local + n_halo)
    colidx = np.empty(local_cols_global.shape[0], dtype=np.int32)
    colidx[is_local] = (local_cols_global[is_local] - lo).astype(np.int32)
    pos = np.searchsorted(halo_ids, local_cols_global[~is_local])
    colidx[~is_local] = (n_local + pos).astype(np.int32)

    n_ext = n_local + n_halo
    rowptr_t = torch.from_numpy(local_rowptr)
    colidx_t = torch.from_numpy(colidx)
    t_rowptr, t_colidx = build_transpose(n_ext, rowptr_t, colidx_t)

    # exchange strategy (parallel/aggregate.py dispatches on it)
    comm_mode = "none"
    halo_fraction = 0.0
    if world_size > 1:
        halo_fraction = n_halo / max(bounds[-1] - n_local, 1)
        comm_mode = os.environ.get("ROC_COMM_MODE", "auto")
        if comm_mode == "auto":
            comm_mode = "allgather" if halo_fraction > 0.5 else "halo"

    # who needs MY rows (halo mode only; allgather sends whole blocks)
    if world_size == 1 or comm_mode == "allgather":
        send_idx = torch.empty(0, dtype=torch.int64)
        send_splits = [0] * world_size
    elif full_colidx is not None:
        # scan every rank's window (no process group needed; deterministic)
        ci = full_colidx.numpy()
        send_chunks, send_splits = [], []
        for r in range(world_size):
            if r == rank:
                send_chunks.append(np.empty(0, dtype=np.int64))
                send_splits.append(0)
                continue
            rlo, rhi = bounds[r], bounds[r + 1]
            their_cols = ci[rp[rlo]:rp[rhi]]
            theirs_from_me = np.unique(
                their_cols[(their_cols >= lo) & (their_cols < hi)]
            ).astype(np.int64)
            send_chunks.append(theirs_from_me - lo)
            send_splits.append(theirs_from_me.shape[0])
        send_idx = torch.from_numpy(np.concatenate(send_chunks))
    else:
        send_idx, send_splits = _send_plan_comm(
            halo_ids, recv_splits, lo, world_size, group)

    # degrees (GLOBAL in-degree — normalization must match the 1-GPU model)
    deg_all = np.maximum(np.diff(rp), 1).astype(np.float32)
    deg_local = deg_all[lo:hi]
    deg_ext = np.concatenate([deg_local, deg_all[halo_ids]]) if n_halo else deg_local

    # Row scheduling for the SpMM kernels. Two regimes (measured,
    # profiles/r16): on a community-ordered graph the NATURAL order wins
    # ~10% — consecutive workgroups gather from the same L2-scale source
    # window — while on a locality-free ordering the global
    # degree-descending order wins ~7% (hub rows scheduled first, no tail).
    # Auto-pick by a sampled in-window gather fraction; override with
    # ROC_SPMM_SCHEDULE=degree|natural.
    local_deg_edges = np.diff(local_rowptr)
    sched = os.environ.get("ROC_SPMM_SCHEDULE", "auto")
    if sched == "auto":
        sched = ("natural"
                 if _gather_locality(local_rowptr, colidx) >= 0.5
                 else "degree")
    if sched == "natural":
        row_order = None
        t_row_order = None
    else:
        row_order = np.argsort(-local_deg_edges, kind="stable").astype(np.int32)
        t_deg = np.diff(t_rowptr.numpy())
        t_row_order = np.argsort(-t_deg, kind="stable").astype(np.int32)

    shard_kw = {}
    if world_size > 1:
        shard_kw["comm_mode"] = comm_mode
        shard_kw["halo_fraction"] = float(halo_fraction)
    if world_size > 1 and comm_mode == "allgather":
        # gather-space CSR: node (owner r, local i) -> r*max_rows + i
        sizes = [bounds[r + 1] - bounds[r] for r in range(world_size)]
        mr = max(sizes)
        owners = np.searchsorted(bounds, local_cols_global,
                                 side="right") - 1
        ag_cols = (owners.astype(np.int64) * mr +
                   (local_cols_global - np.asarray(bounds)[owners]))
        shard_kw["ag_max_rows"] = int(mr)
        shard_kw["ag_colidx"] = torch.from_numpy(ag_cols.astype(np.int32))
        ag_t = build_transpose(world_size * mr, rowptr_t,
                               shard_kw["ag_colidx"])
        shard_kw["ag_t_rowptr"], shard_kw["ag_t_colidx"] = ag_t
        ag_t_deg = np.diff(ag_t[0].numpy())
        shard_kw["ag_t_row_order"] = torch.from_numpy(
            np.argsort(-ag_t_deg, kind="stable").astype(np.int32))
        # self/remote edge split for comm/compute overlap (fwd only)
        is_self = owners == rank
        row_of_edge = np.repeat(np.arange(n_local, dtype=np.int64),
                                np.diff(local_rowptr))
        for key_pfx, sel, cols in (
                ("ag_self", is_self,
                 (local_cols_global - lo).astype(np.int32)),
                ("ag_rem", ~is_self, ag_cols.astype(np.int32))):
            rows_sel = row_of_edge[sel]
            cnt = np.bincount(rows_sel, minlength=n_local).astype(np.int64)
            part_rp = np.zeros(n_local + 1, dtype=np.int64)
            np.cumsum(cnt, out=part_rp[1:])
            shard_kw[f"{key_pfx}_rowptr"] = torch.from_numpy(part_rp)
            shard_kw[f"{key_pfx}_colidx"] = torch.from_numpy(
                np.ascontiguousarray(cols[sel]))
            shard_kw[f"{key_pfx}_row_order"] = torch.from_numpy(
                np.argsort(-cnt, kind="stable").astype(np.int32))
        # per-owner-block degree-desc row orders for the backward blocks
        ag_t_rp = ag_t[0].numpy()
        blk_ord = np.empty(world_size * mr, dtype=np.int32)
        for r in range(world_size):
            deg_blk = np.diff(ag_t_rp[r * mr:(r + 1) * mr + 1])
            blk_ord[r * mr:(r + 1) * mr] = np.argsort(
                -deg_blk, kind="stable").astype(np.int32)
        shard_kw["ag_t_blk_order"] = torch.from_numpy(blk_ord)
    if world_size > 1 and comm_mode == "halo":
        # split edges by source locality for comm/compute overlap
        is_loc_edge = colidx < n_local
        row_of_edge = np.repeat(np.arange(n_local, dtype=np.int64),
                                np.diff(local_rowptr))
        for key_pfx, sel in (("loc", is_loc_edge), ("halo", ~is_loc_edge)):
            rows_sel = row_of_edge[sel]
            cols_sel = colidx[sel].astype(np.int32)
            if key_pfx == "halo":
                cols_sel = cols_sel - n_local  # halo-local id space
            cnt = np.bincount(rows_sel, minlength=n_local).astype(np.int64)
            part_rp = np.zeros(n_local + 1, dtype=np.int64)
            np.cumsum(cnt, out=part_rp[1:])
            shard_kw[f"{key_pfx}_rowptr"] = torch.from_numpy(part_rp)
            shard_kw[f"{key_pfx}_colidx"] = torch.from_numpy(
                np.ascontiguousarray(cols_sel))
        t_loc = build_transpose(n_local, shard_kw["loc_rowptr"],
                                shard_kw["loc_colidx"])
        t_halo = build_transpose(max(n_halo, 0), shard_kw["halo_rowptr"],
                                 shard_kw["halo_colidx"])
        shard_kw["t_loc_rowptr"], shard_kw["t_loc_colidx"] = t_loc
        shard_kw["t_halo_rowptr"], shard_kw["t_halo_colidx"] = t_halo

    # source-strip-blocked plans for L2-resident gathers (measured
    # 1.39x on the Reddit D=256 SpMM — profiles/r21). Only for CSRs big
    # enough that the strip prep + extra output traffic pays.
    # With column phases on (off by default — see _phase_cols()),
    # passes touch 64-col windows, so strips are 4x wider at the same
    # gather working set.
    phase_on = int(os.environ.get("ROC_SPMM_PHASE_COLS", "0")) > 0
    strip_w = int(os.environ.get("ROC_SPMM_STRIP_WIDTH",
                                 "65536" if phase_on else "16384"))
    strip_min = int(os.environ.get("ROC_SPMM_STRIP_MIN_EDGES",
                                   str(24_000_000)))
    max_k = int(os.environ.get("ROC_SPMM_STRIP_MAX_K", "32"))
    # K passes cost K output re-reads: profitable only while K stays
    # small AND each strip is L2-scale — huge graphs (papers100M:
    # K would be ~6800) skip strips entirely
    n_strips = (n_ext + strip_w - 1) // max(strip_w, 1)
    if (strip_w > 0 and colidx_t.numel() >= strip_min
            and 2 <= n_strips <= max_k):
        shard_kw["fwd_strips"] = build_strip_plan(rowptr_t, colidx_t,
                                                  n_ext, strip_w)
        shard_kw["bwd_strips"] = build_strip_plan(t_rowptr, t_colidx,
                                                  n_local, strip_w)

    return GraphShard(
        rank=rank, world_size=world_size, bounds=list(bounds),
        lo=lo, hi=hi, n_local=n_local, n_halo=n_halo, n_ext=n_ext,
        rowptr=rowptr_t, colidx=colidx_t,
        t_rowptr=t_rowptr, t_colidx=t_colidx, **shard_kw,
        deg_local=torch.from_numpy(deg_local.copy()),
        rsqrt_deg_local=torch.from_numpy(1.0 / np.sqrt(deg_local)),
        inv_deg_local=torch.from_numpy(1.0 / deg_local),
        rsqrt_deg_ext=torch.from_numpy(1.0 / np.sqrt(deg_ext)),
        row_order=(torch.from_numpy(row_order)
                   if row_order is not None else None),
        t_row_order=(torch.from_numpy(t_row_order)
                     if t_row_order is not None else None),
        halo_ids=torch.from_numpy(halo_ids),
        recv_splits=recv_splits,
        send_idx=send_idx,
        send_splits=send_splits,
    )
