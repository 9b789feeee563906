"""APPNP (Approximate Personalized Propagation of Neural Predictions,
Klicpera et al. 2019).

h = MLP(x);  z_0 = h;  z_{t+1} = (1-alpha) * D^-1/2 A D^-1/2 z_t + alpha*h

The K propagation steps are DIFFERENTIABLE (gradients flow through all
K distributed aggregations — the transpose-CSR backward makes this
exact on asymmetric graphs). Decouples depth-of-propagation from
depth-of-transform; not in the reference, included for breadth.
"""
from __future__ import annotations

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.aggregate import aggregate


class APPNP(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1,
                 k: int = 10, alpha: float = 0.1):
        super().__init__()
        self.dims = list(dims)
        self.k = int(k)
        self.alpha = float(alpha)
        self.p = float(dropout)
        self.weights = torch.nn.ParameterList(
            torch.nn.Parameter(glorot_uniform((dims[i], dims[i + 1]),
                                              seed=seed + i))
            for i in range(len(dims) - 1))

    recompute = False

    def forward(self, x, shard, group=None):
        h = x
        for i, w in enumerate(self.weights):
            h = F.dropout(h, self.p, self.training, call_id=i)
            h = F.linear(h, w,
                         activation="relu" if i < len(self.weights) - 1
                         else None)
        z = h
        a = self.alpha
        for t in range(self.k):
            z = F.degree_scale(z, shard.rsqrt_deg_local)
            z = aggregate(z, shard, dst_scale=shard.rsqrt_deg_local,
                          group=group)
            z = (1.0 - a) * z + a * h
        return z
