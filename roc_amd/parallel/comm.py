"""Small process-group helpers shared by the parallel layer.

RCCL ("nccl" backend on ROCm) only moves GPU tensors; plan/metadata
exchanges (halo send plans, global train counts, divergence flags) are
tiny CPU int tensors. `cpu_group()` hands back a cached gloo side group
for those when the training group is RCCL, so control-plane collectives
never touch the GPU data plane. (The reference had no control plane at
all — Legion moved metadata implicitly through region trees.)
"""
from __future__ import annotations

import torch
import torch.distributed as dist

_CPU_GROUP = None


def cpu_group(group=None):
    """A process group that supports CPU tensors.

    Returns `group` unchanged when it already does (gloo), otherwise a
    lazily-created, cached gloo group over ALL ranks. Every rank must
    reach the first call together (dist.new_group is collective) — true
    for all call sites (shard build, trainer init, recovery check).
    Only valid for world-spanning groups; subgroup callers must pass a
    gloo subgroup themselves.
    """
    global _CPU_GROUP
    if not dist.is_initialized():
        return group
    backend = dist.get_backend(group) if group is not None \
        else dist.get_backend()
    if "gloo" in str(backend):
        return group
    if _CPU_GROUP is None:
        _CPU_GROUP = dist.new_group(backend="gloo")
    return _CPU_GROUP


def allreduce_scalar_int(value: int, group=None, op="sum") -> int:
    """All-reduce one python int over the CPU control plane."""
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return value
    t = torch.tensor([value], dtype=torch.int64)
    red = dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM
    dist.all_reduce(t, op=red, group=cpu_group(group))
    return int(t.item())
