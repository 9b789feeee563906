"""GCN with configurable depth and optional residual connections.

Mirrors the reference driver's model (`gnn.cc:66-92`): per layer
  dropout -> linear -> indegree_norm -> scatter_gather -> indegree_norm
  -> relu (not on last layer) [-> residual projection + add]
The two indegree_norms around the aggregation implement the symmetric
D^-1/2 A D^-1/2 GCN normalization; on GPU they are fused into the SpMM
kernel (`scatter_gather(..., normalize=True)`).
"""
from __future__ import annotations

import os

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.halo import halo_exchange
from ..parallel.aggregate import aggregate


def _adaptive_agg() -> bool:
    """Aggregate-at-narrower-width on widening layers (ROC_ADAPTIVE_AGG,
    default on; 0 disables for A/B)."""
    global _ADAPTIVE
    if _ADAPTIVE is None:
        _ADAPTIVE = os.environ.get("ROC_ADAPTIVE_AGG", "1") != "0"
    return _ADAPTIVE


_ADAPTIVE = None


class GCN(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1,
                 residual: bool = False, fused_norm: bool = True):
        super().__init__()
        self.dims = list(dims)
        self.p = float(dropout)
        self.residual = residual
        self.fused_norm = fused_norm
        self.weights = torch.nn.ParameterList()
        self.res_proj = torch.nn.ParameterList()
        for i in range(len(dims) - 1):
            self.weights.append(torch.nn.Parameter(
                glorot_uniform((dims[i], dims[i + 1]), seed=seed + i)))
            if residual and dims[i] != dims[i + 1]:
                self.res_proj.append(torch.nn.Parameter(
                    glorot_uniform((dims[i], dims[i + 1]), seed=seed + 1000 + i)))
            else:
                self.res_proj.append(torch.nn.Parameter(torch.empty(0)))

    # Activation recompute (capacity tier): when True, each layer's
    # forward is re-run during backward instead of keeping its
    # intermediates resident — O(1 layer) transient HBM instead of
    # O(depth). Dropout stays exact via the per-layer Philox call_id.
    recompute = False

    def _layer(self, i, x, shard, group):
        w = self.weights[i]
        h = F.dropout(x, self.p, self.training, call_id=i)
        if self.fused_norm:
            if w.shape[0] < w.shape[1] and _adaptive_agg():
                # widening layer: aggregate FIRST, at the narrower
                # input width (aggregation is linear: A(XW) == (AX)W),
                # so the gather stream AND any halo exchange move
                # in_dim-wide rows instead of out_dim-wide ones
                h = F.degree_scale(h, shard.rsqrt_deg_local)
                h = aggregate(h, shard, dst_scale=shard.rsqrt_deg_local,
                              group=group)
                h = F.linear(h, w)
            else:
                # source-side D^-1/2 rides the GEMM epilogue (owner
                # rank scales its rows BEFORE the halo exchange);
                # dst-side D^-1/2 rides the SpMM store. No per-edge
                # degree gather.
                h = F.linear(h, w, row_scale=shard.rsqrt_deg_local)
                # strategy (halo a2a / overlap / all_gather) is the
                # shard's choice — see parallel/aggregate.py
                h = aggregate(h, shard, dst_scale=shard.rsqrt_deg_local,
                              group=group)
        else:
            h = F.linear(h, w)
            h = halo_exchange(h, shard, group)
            h = F.indegree_norm(
                F.scatter_gather(F.degree_scale(h, shard.rsqrt_deg_ext), shard),
                shard)
        if i < len(self.weights) - 1:
            h = F.relu(h)
        if self.residual:
            proj = self.res_proj[i]
            if proj.numel() > 0:
                h = F.add(h, F.linear(x, proj))
            elif x.shape == h.shape:
                h = F.add(h, x)
        return h

    def forward(self, x, shard, group=None):
        for i in range(len(self.weights)):
            if self.recompute and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    self._layer, i, x, shard, group,
                    use_reentrant=False, preserve_rng_state=False)
            else:
                x = self._layer(i, x, shard, group)
        return x
