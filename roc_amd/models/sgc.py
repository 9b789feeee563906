"""SGC (Simplified Graph Convolution, Wu et al. 2019).

logits = (D^-1/2 A D^-1/2)^K X W — the K-hop propagation has no
nonlinearity, so it is PRECOMPUTED once (distributed aggregate under
no_grad; works in halo and allgather modes) and every epoch is a single
GEMM over the cached propagated features. The cheapest family for
serving and for quick label refreshes on a fixed graph; not in the
reference (its model zoo is the GCN of `gnn.cc:66-92`), included for
framework breadth.
"""
from __future__ import annotations

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.aggregate import aggregate


class SGC(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1,
                 k: int = 2):
        super().__init__()
        assert len(dims) >= 2, "SGC needs [in_dim, ..., num_classes]"
        self.dims = list(dims)
        self.k = int(k)
        self.p = float(dropout)
        # single linear by construction: dims[0] -> dims[-1]
        self.weight = torch.nn.Parameter(
            glorot_uniform((dims[0], dims[-1]), seed=seed))
        self._cache = None  # (key, propagated features)

    recompute = False  # propagation is cached; nothing to recompute

    def _propagated(self, x, shard, group):
        key = (x.data_ptr(), x.shape, x.dtype, id(shard), self.k)
        if self._cache is not None and self._cache[0] == key:
            return self._cache[1]
        with torch.no_grad():
            s = x
            for _ in range(self.k):
                s = F.degree_scale(s, shard.rsqrt_deg_local)
                s = aggregate(s, shard, dst_scale=shard.rsqrt_deg_local,
                              group=group)
        self._cache = (key, s)
        return s

    def forward(self, x, shard, group=None):
        s = self._propagated(x, shard, group)
        h = F.dropout(s, self.p, self.training, call_id=0)
        return F.linear(h, self.weight)
