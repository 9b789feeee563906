"""One entry point for distributed neighbor aggregation.

Strategy is chosen per-shard at partition time (ROC_COMM_MODE overrides):

- "none"      world_size == 1: plain local SpMM.
- "halo"      sparse boundaries: all_to_allv of exactly the needed rows
              (optionally overlapped with the interior SpMM, ROC_OVERLAP=1).
- "allgather" near-total halo (uniform/dense cuts, halo_fraction > 0.5):
              all_gather the padded local blocks forward and
              reduce_scatter the gradients backward — bandwidth-optimal
              RCCL collectives instead of a2av metadata + gather/scatter
              indexing. This is the design the reference abandoned in
              dead code (`gnn_kernel.cu:65-78`), done properly.

Models call aggregate(x_local, shard, dst_scale, group) and never see
the strategy.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

import os as _os

from .. import streamcheck
from .halo import halo_exchange, halo_aggregate, overlap_enabled, _spmm_part


def ag_overlap_enabled() -> bool:
    """Overlap the allgather-mode collectives with the aggregation SpMMs
    (default ON; ROC_AG_OVERLAP=0 falls back to the sequential
    all_gather -> SpMM forward and SpMM^T -> reduce_scatter backward)."""
    return _os.environ.get("ROC_AG_OVERLAP", "1") == "1"


def _ag_block_bounds(shard):
    """Edge offsets of each owner block in ag_t_colidx ([ws+1] python
    ints, cached on the shard: one D2H sync total, not one per layer)."""
    b = getattr(shard, "_ag_t_bounds", None)
    if b is None:
        mr = shard.ag_max_rows
        idx = torch.arange(0, shard.world_size * mr + 1, mr,
                           device=shard.ag_t_rowptr.device)
        b = [int(v) for v in shard.ag_t_rowptr[idx].cpu()]
        object.__setattr__(shard, "_ag_t_bounds", b)
    return b


def _reduce_scatter(out, inp, group):
    try:
        dist.reduce_scatter_tensor(out, inp, group=group)
    except (RuntimeError, ValueError):  # gloo lacks reduce_scatter
        dist.all_reduce(inp, group=group)
        rank = dist.get_rank(group)
        n = out.shape[0]
        out.copy_(inp[rank * n:(rank + 1) * n])


def _spmm_local(out, x, rowptr, colidx, dst, row_order=None):
    from ..ops import functional as Fn
    if out.is_cuda:
        Fn._hip(out)
        Fn._C.spmm(out, x, rowptr, colidx, dst, None, row_order, False)
    else:
        from ..ops import reference as ref
        part = ref.spmm(x, rowptr, colidx, out.shape[0])
        if dst is not None:
            part = part * dst.unsqueeze(1).to(part.dtype)
        out.copy_(part)


class _GatherAggregate(torch.autograd.Function):
    """forward: all_gather padded blocks -> SpMM over the gathered space.
    backward: SpMM^T -> reduce_scatter of the per-block gradient sums."""

    @staticmethod
    def forward(ctx, x, shard, dst_scale, group):
        ctx.shard = shard
        ctx.group = group
        ctx.save_for_backward(dst_scale)
        D = x.shape[1]
        mr = shard.ag_max_rows
        ws = shard.world_size
        send = x
        if x.shape[0] < mr:  # pad the block to the common size
            send = torch.zeros(mr, D, dtype=x.dtype, device=x.device)
            send[:x.shape[0]] = x
        gathered = torch.empty(ws * mr, D, dtype=x.dtype, device=x.device)
        out = torch.empty(shard.n_local, D, dtype=x.dtype, device=x.device)
        if ag_overlap_enabled() and shard.ag_self_rowptr is not None:
            # self-source edges aggregate from local x while the gather
            # is in flight; remote-source edges then accumulate and the
            # dst scale lands on the final store
            work = dist.all_gather_into_tensor(
                gathered, send.contiguous(), group=group, async_op=True)
            streamcheck.producer(work, "ag-allgather-fwd")
            cuda = out.is_cuda
            _spmm_part(out, x, shard.ag_self_rowptr, shard.ag_self_colidx,
                       None, False,
                       shard.ag_self_row_order if cuda else None)
            streamcheck.consumer(work, "ag-remote-accumulate")
            work.wait()
            _spmm_part(out, gathered, shard.ag_rem_rowptr,
                       shard.ag_rem_colidx, dst_scale, True,
                       shard.ag_rem_row_order if cuda else None)
        else:
            dist.all_gather_into_tensor(gathered, send.contiguous(),
                                        group=group)
            _spmm_local(out, gathered, shard.rowptr, shard.ag_colidx,
                        dst_scale, shard.row_order if out.is_cuda else None)
        return out

    @staticmethod
    def backward(ctx, dy):
        shard, group = ctx.shard, ctx.group
        (dst_scale,) = ctx.saved_tensors
        from .halo import _rowscale
        dy = dy.contiguous()
        if dst_scale is not None:
            dy = _rowscale(dy, dst_scale)
        D = dy.shape[1]
        mr = shard.ag_max_rows
        ws = shard.world_size
        dfull = torch.empty(ws * mr, D, dtype=dy.dtype, device=dy.device)
        if ag_overlap_enabled():
            # per-owner-block SpMM^T + async reduce-to-owner: block r's
            # reduction is in flight while block r+1 computes (same total
            # bytes as reduce_scatter, comm hidden behind compute)
            bnds = _ag_block_bounds(shard)
            works = []
            for r in range(ws):  # same launch order on every rank
                rp_blk = shard.ag_t_rowptr[r * mr:(r + 1) * mr + 1]
                rp_blk = (rp_blk - rp_blk[:1]).contiguous()
                cols = shard.ag_t_colidx[bnds[r]:bnds[r + 1]]
                blk = dfull[r * mr:(r + 1) * mr]
                order = (shard.ag_t_blk_order[r * mr:(r + 1) * mr]
                         if dy.is_cuda and shard.ag_t_blk_order is not None
                         else None)
                _spmm_part(blk, dy, rp_blk, cols, None, False, order)
                dst = (dist.get_global_rank(group, r)
                       if group is not None else r)
                w = dist.reduce(blk, dst=dst, group=group, async_op=True)
                streamcheck.producer(w, "ag-reduce-bwd")
                works.append(w)
            for w in works:
                streamcheck.consumer(w, "ag-bwd-done")
                w.wait()
            own = dfull[shard.rank * mr:shard.rank * mr + shard.n_local]
            return own.contiguous(), None, None, None
        _spmm_local(dfull, dy, shard.ag_t_rowptr, shard.ag_t_colidx, None,
                    shard.ag_t_row_order if dy.is_cuda else None)
        dx_pad = torch.empty(mr, D, dtype=dy.dtype, device=dy.device)
        _reduce_scatter(dx_pad, dfull, group)
        return dx_pad[:shard.n_local].contiguous(), None, None, None


def aggregate(x, shard, dst_scale=None, group=None):
    """Distributed neighbor sum-aggregation of the LOCAL feature block
    (+ optional fused dst-side degree scale). Returns [n_local, D]."""
    from ..ops import functional as Fn
    if shard.world_size == 1:
        return Fn.scatter_gather(x, shard, dst_scale=dst_scale)
    if shard.comm_mode == "allgather":
        return _GatherAggregate.apply(x, shard, dst_scale, group)
    if overlap_enabled() and shard.has_overlap_split:
        return halo_aggregate(x, shard, dst_scale=dst_scale, group=group)
    xe = halo_exchange(x, shard, group)
    return Fn.scatter_gather(xe, shard, dst_scale=dst_scale)
