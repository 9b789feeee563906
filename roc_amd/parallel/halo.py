"""Boundary-vertex halo exchange over RCCL (xGMI) / gloo.

Replaces the reference's full-tensor replication through pinned host
memory (`scattergather.cc:68-73`, `types.cu:22-32`) with an
all_to_all_single of exactly the boundary rows each neighbor needs.
xGMI is point-to-point (7 links/GPU), so the all-to-all uses all links
concurrently instead of a per-link-bound ring.

Autograd: forward gathers halo rows; backward routes halo-row gradients
back to their owners and accumulates into the local gradient.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from .. import streamcheck


def _a2a(out: torch.Tensor, inp: torch.Tensor, out_splits, in_splits, group):
    dist.all_to_all_single(
        out, inp, output_split_sizes=out_splits, input_split_sizes=in_splits,
        group=group,
    )


class _HaloExchange(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, shard, group):
        ctx.shard = shard
        ctx.group = group
        if shard.world_size == 1:
            return x
        assert shard.comm_mode == "halo", (
            "halo_exchange needs a halo-mode shard (this one is "
            f"'{shard.comm_mode}'; use parallel.aggregate.aggregate())")
        # NOTE: the all_to_all is collective — every rank participates even
        # with an empty halo (another rank may still need our rows).
        n_local = shard.n_local
        out = torch.empty(shard.n_ext, x.shape[1], dtype=x.dtype, device=x.device)
        out[:n_local] = x
        send = x[shard.send_idx].contiguous()
        _a2a(out[n_local:], send, shard.recv_splits, shard.send_splits, group)
        return out

    @staticmethod
    def backward(ctx, dy):
        shard, group = ctx.shard, ctx.group
        if shard.world_size == 1:
            return dy, None, None
        n_local = shard.n_local
        dy = dy.contiguous()
        dx = dy[:n_local].contiguous()
        grad_in = torch.empty(
            int(sum(shard.send_splits)), dy.shape[1], dtype=dy.dtype,
            device=dy.device)
        _a2a(grad_in, dy[n_local:].contiguous(),
             shard.send_splits, shard.recv_splits, group)
        dx.index_add_(0, shard.send_idx, grad_in)
        return dx, None, None


def halo_exchange(x, shard, group=None):
    """[n_local, D] -> [n_ext, D] with halo rows appended (owner-grouped)."""
    return _HaloExchange.apply(x, shard, group)


# ---------------------------------------------------------------------------
# Fused halo exchange + aggregation with comm/compute overlap:
# interior (local-source) edges aggregate while the boundary rows are in
# flight on RCCL; the halo-source edges then accumulate into the partial
# sums and the dst-side degree scale lands at the final store.
# Backward mirrors it: halo-row gradients are computed FIRST so the
# reverse all_to_all overlaps the local-source gradient aggregation.
# Enable with ROC_OVERLAP=1 (or overlap=True on the model call sites).
# ---------------------------------------------------------------------------

import os as _os


def overlap_enabled() -> bool:
    """Halo-mode comm/compute overlap (interior SpMM while the a2av is
    in flight). Default ON since round 2: correctness is equality-
    tested vs the sequential path at world sizes 2/3/8 (gloo), the
    collectives run eagerly (no hipGraph interaction — RCCL a2av
    inside a captured graph segfaults on this stack, see
    tests/test_rccl_gpu.py::test_rccl_raw_collectives_in_graph), and
    hiding the exchange behind the interior aggregation is the whole
    point of the split. ROC_OVERLAP=0 reverts."""
    return _os.environ.get("ROC_OVERLAP", "1") == "1"


def _spmm_part(out, x, rowptr, colidx, dst, acc, order=None):
    from ..ops import functional as Fn
    if out.is_cuda:
        Fn._hip(out)
        Fn._C.spmm(out, x, rowptr, colidx, dst, None, order, acc)
    else:
        from ..ops import reference as ref
        part = ref.spmm(x, rowptr, colidx, out.shape[0])
        if acc:
            part = part + out
        if dst is not None:
            part = part * dst.unsqueeze(1).to(part.dtype)
        out.copy_(part)


class _HaloAggregate(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, shard, dst_scale, group):
        ctx.shard = shard
        ctx.group = group
        ctx.save_for_backward(dst_scale)
        D = x.shape[1]
        x = x.contiguous()
        # 1) launch the boundary exchange
        send = x[shard.send_idx].contiguous()
        halo = torch.empty(shard.n_halo, D, dtype=x.dtype, device=x.device)
        work = dist.all_to_all_single(
            halo, send, output_split_sizes=shard.recv_splits,
            input_split_sizes=shard.send_splits, group=group, async_op=True)
        streamcheck.producer(work, "halo-a2a-fwd")
        # 2) interior aggregation overlaps the exchange
        out = torch.empty(shard.n_local, D, dtype=x.dtype, device=x.device)
        _spmm_part(out, x, shard.loc_rowptr, shard.loc_colidx, None, False)
        # 3) boundary contribution + final dst scaling
        streamcheck.consumer(work, "boundary-spmm")
        work.wait()
        _spmm_part(out, halo, shard.halo_rowptr, shard.halo_colidx,
                   dst_scale, True)
        return out

    @staticmethod
    def backward(ctx, dy):
        shard, group = ctx.shard, ctx.group
        (dst_scale,) = ctx.saved_tensors
        from ..ops import functional as Fn
        dy = dy.contiguous()
        if dst_scale is not None:
            dy = _rowscale(dy, dst_scale)
        D = dy.shape[1]
        # 1) halo-row grads first, so the reverse exchange starts early
        dhalo = torch.empty(shard.n_halo, D, dtype=dy.dtype, device=dy.device)
        _spmm_part(dhalo, dy, shard.t_halo_rowptr, shard.t_halo_colidx,
                   None, False)
        grad_in = torch.empty(int(sum(shard.send_splits)), D, dtype=dy.dtype,
                              device=dy.device)
        work = dist.all_to_all_single(
            grad_in, dhalo, output_split_sizes=shard.send_splits,
            input_split_sizes=shard.recv_splits, group=group, async_op=True)
        streamcheck.producer(work, "halo-a2a-bwd")
        # 2) local-source grads overlap the exchange
        dx = torch.empty(shard.n_local, D, dtype=dy.dtype, device=dy.device)
        _spmm_part(dx, dy, shard.t_loc_rowptr, shard.t_loc_colidx,
                   None, False)
        streamcheck.consumer(work, "grad-scatter")
        work.wait()
        dx.index_add_(0, shard.send_idx, grad_in)
        return dx, None, None, None


def _rowscale(x, scale):
    from ..ops import functional as Fn
    if x.is_cuda:
        out = torch.empty_like(x)
        Fn._C.rowscale(out, x, scale)
        return out
    return x * scale.unsqueeze(1).to(x.dtype)


def halo_aggregate(x, shard, dst_scale=None, group=None):
    """Fused halo exchange + sum-aggregation (+ dst-side degree scale)
    with comm/compute overlap. Requires shard.has_overlap_split."""
    return _HaloAggregate.apply(x, shard, dst_scale, group)
