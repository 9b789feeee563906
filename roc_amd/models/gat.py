"""Graph Attention Network (GAT, Velickovic et al. 2018).

Per layer (H heads, concatenated except the last layer):
  h = W·x ; score(u→v) = LeakyReLU(a_src·h_u + a_dst·h_v)
  alpha = softmax over v's in-edges (fused edge_softmax kernel)
  out[v] = Σ_u alpha(u→v) · h_u          (spmm_edge kernel)

This exercises the full edge-tensor surface end-to-end (the reference
declared edge tensors, `gnn.cc:475-623`, but had no consumer op and no
attention model). Distribution: node scores ride the usual halo
exchange (scores are computed from the EXCHANGED rows, so attention
over remote neighbors is exact); attention coefficients themselves are
partition-local edge tensors — no extra communication.

At world_size > 1 the shard must be in "halo" comm mode (attention
needs per-edge source rows, which the allgather path never
materializes per-rank); build_shard picks halo automatically for
sparse cuts or force ROC_COMM_MODE=halo.
"""
from __future__ import annotations

import torch

from ..ops import functional as F
from ..ops.reference import glorot_uniform
from ..parallel.halo import halo_exchange


class GAT(torch.nn.Module):
    def __init__(self, dims, dropout: float = 0.5, seed: int = 1,
                 heads: int = 4, negative_slope: float = 0.2):
        super().__init__()
        self.dims = list(dims)
        self.p = float(dropout)
        self.slope = float(negative_slope)
        self.weights = torch.nn.ParameterList()
        self.a_src = torch.nn.ParameterList()
        self.a_dst = torch.nn.ParameterList()
        self.heads = []
        for i in range(len(dims) - 1):
            h = heads if i < len(dims) - 2 else 1  # last layer: 1 head
            assert dims[i + 1] % h == 0, \
                f"layer {i}: out dim {dims[i + 1]} not divisible by {h} heads"
            dh = dims[i + 1] // h
            self.heads.append(h)
            self.weights.append(torch.nn.Parameter(
                glorot_uniform((dims[i], dims[i + 1]), seed=seed + 3 * i)))
            self.a_src.append(torch.nn.Parameter(
                glorot_uniform((h, dh), seed=seed + 3 * i + 1)))
            self.a_dst.append(torch.nn.Parameter(
                glorot_uniform((h, dh), seed=seed + 3 * i + 2)))

    recompute = False  # see GCN.recompute

    def forward(self, x, shard, group=None):
        if shard.world_size > 1:
            assert shard.comm_mode == "halo", (
                "GAT needs a halo-mode shard at world_size>1 "
                f"(got '{shard.comm_mode}'; set ROC_COMM_MODE=halo)")
        for i in range(len(self.weights)):
            if self.recompute and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    self._layer, i, x, shard, group,
                    use_reentrant=False, preserve_rng_state=False)
            else:
                x = self._layer(i, x, shard, group)
        return x

    def _layer(self, i, x, shard, group):
        nh, dh = self.heads[i], self.dims[i + 1] // self.heads[i]
        h = F.dropout(x, self.p, self.training, call_id=i)
        h = F.linear(h, self.weights[i])            # [n_local, nh*dh]
        h_ext = halo_exchange(h, shard, group)      # [n_ext, nh*dh]
        # ALL heads' score halves in one skinny GEMM each (a per-head
        # fp32 gemv hit rocBLAS's slow gemvn path — 2.2 ms/call, 12% of
        # the epoch, r2c16): block-diagonal [nh*dh, nh] projection,
        # then transpose so each head's column is a contiguous row.
        Wsrc = torch.block_diag(*(self.a_src[i][k].unsqueeze(1)
                                  for k in range(nh))).to(h_ext.dtype)
        Wdst = torch.block_diag(*(self.a_dst[i][k].unsqueeze(1)
                                  for k in range(nh))).to(h_ext.dtype)
        s_src_all = (h_ext @ Wsrc).float().t().contiguous()   # [nh, n_ext]
        s_dst_all = (h_ext[:shard.n_local] @ Wdst).float().t().contiguous()
        outs = []
        for k in range(nh):
            hk = h_ext[:, k * dh:(k + 1) * dh]
            # fused: lrelu(s_src[col]+s_dst[row]) -> segment softmax,
            # one kernel each direction (no E-length intermediates)
            alpha = F.attention_softmax(s_src_all[k], s_dst_all[k],
                                        shard, self.slope)
            outs.append(F.scatter_gather_weighted(hk.contiguous(), alpha,
                                                  shard))
        out = outs[0] if nh == 1 else torch.cat(outs, dim=1)
        if i < len(self.weights) - 1:
            out = F.relu(out)
        return out
