"""GIN (Graph Isomorphism Network), sum aggregation + 2-layer MLP.

h' = MLP( (1 + eps) * h_v + sum_{u in N(v)} h_u ),  eps learnable.
Exercises the scatter-add aggregation path (BASELINE.json config #4).
"""
from __future__ import annotations

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.aggregate import aggregate


class GIN(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1,
                 mlp_hidden: int = 0):
        super().__init__()
        self.dims = list(dims)
        self.p = float(dropout)
        self.eps = torch.nn.ParameterList()
        self.w1 = torch.nn.ParameterList()
        self.w2 = torch.nn.ParameterList()
        for i in range(len(dims) - 1):
            hid = mlp_hidden or dims[i + 1]
            self.eps.append(torch.nn.Parameter(torch.zeros(1)))
            self.w1.append(torch.nn.Parameter(
                glorot_uniform((dims[i], hid), seed=seed + 2 * i)))
            self.w2.append(torch.nn.Parameter(
                glorot_uniform((hid, dims[i + 1]), seed=seed + 2 * i + 1)))

    recompute = False  # see GCN.recompute

    def _layer(self, i, x, shard, group):
        h = F.dropout(x, self.p, self.training, call_id=i)
        agg = aggregate(h, shard, group=group)
        h = agg + (1.0 + self.eps[i]).to(h.dtype) * h  # scalar-eps glue
        h = F.linear(h, self.w1[i], activation="relu")
        h = F.linear(h, self.w2[i])
        if i < len(self.w1) - 1:
            h = F.relu(h)
        return h

    def forward(self, x, shard, group=None):
        for i in range(len(self.w1)):
            if self.recompute and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    self._layer, i, x, shard, group,
                    use_reentrant=False, preserve_rng_state=False)
            else:
                x = self._layer(i, x, shard, group)
        return x
