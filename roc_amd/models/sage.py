"""GraphSAGE (mean aggregator, concat variant).

h' = relu( W_self·h_v  +  W_neigh·mean_{u in N(v)} h_u )

The concat-then-linear of the original formulation is algebraically folded
into two GEMMs (no concat materialization), and the aggregation runs at
the NARROWER of (input, output) width — W_neigh is applied before the
exchange+aggregation on shrinking layers and after them on widening
ones (mean_u(h W) == (mean_u h) W) — so the gather stream and the halo
exchange always move the skinnier rows.
"""
from __future__ import annotations

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.aggregate import aggregate


class GraphSAGE(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1):
        super().__init__()
        self.dims = list(dims)
        self.p = float(dropout)
        self.w_self = torch.nn.ParameterList()
        self.w_neigh = torch.nn.ParameterList()
        for i in range(len(dims) - 1):
            self.w_self.append(torch.nn.Parameter(
                glorot_uniform((dims[i], dims[i + 1]), seed=seed + 2 * i)))
            self.w_neigh.append(torch.nn.Parameter(
                glorot_uniform((dims[i], dims[i + 1]), seed=seed + 2 * i + 1)))

    def forward(self, x, shard, group=None):
        for i in range(len(self.w_self)):
            if self.recompute and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    self._layer, i, x, shard, group,
                    use_reentrant=False, preserve_rng_state=False)
            else:
                x = self._layer(i, x, shard, group)
        return x

    recompute = False  # see GCN.recompute

    def _layer(self, i, x, shard, group):
        h = F.dropout(x, self.p, self.training, call_id=i)
        h_self = F.linear(h, self.w_self[i])
        wn = self.w_neigh[i]
        from .gcn import _adaptive_agg
        if wn.shape[0] < wn.shape[1] and _adaptive_agg():
            # widening layer: mean-aggregate first at the narrower
            # input width (mean_u(h W) == (mean_u h) W) — smaller
            # gather stream and halo exchange
            hn = aggregate(h, shard, dst_scale=shard.inv_deg_local,
                           group=group)
            hn = F.linear(hn, wn)
        else:
            hn = F.linear(h, wn)
            hn = aggregate(hn, shard, dst_scale=shard.inv_deg_local,
                           group=group)  # fused mean
        h = F.add(h_self, hn)
        if i < len(self.w_self) - 1:
            h = F.relu(h)
        return h

    def forward_blocks(self, x, blocks):
        """Mini-batch forward over sampled MFG blocks (roc_amd.sampling):
        layer i aggregates over blocks[i]; x holds the input features
        for blocks[0].src_ids rows. Returns [batch, dims[-1]]."""
        from .gcn import _adaptive_agg
        assert len(blocks) == len(self.w_self), (len(blocks), self.dims)
        for i in range(len(self.w_self)):
            blk = blocks[i]
            h = F.dropout(x, self.p, self.training, call_id=i)
            h_self = F.linear(h[:blk.n_dst].contiguous(), self.w_self[i])
            wn = self.w_neigh[i]
            if wn.shape[0] < wn.shape[1] and _adaptive_agg():
                hn = F.scatter_gather(h, blk, dst_scale=blk.inv_deg)
                hn = F.linear(hn, wn)
            else:
                hn = F.linear(h, wn)
                hn = F.scatter_gather(hn, blk, dst_scale=blk.inv_deg)
            x = F.add(h_self, hn)
            if i < len(self.w_self) - 1:
                x = F.relu(x)
        return x
