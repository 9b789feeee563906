"""Checkpoint / resume.

The reference has NO checkpointing; this layout is a new
design: model weights + Adam (m, v, t) + epoch counter + RNG state +
partition bounds, one file per job (weights are replicated across ranks,
so rank 0 writes and every rank can restore).
"""
from __future__ import annotations

import os
from typing import Optional

import torch

from ..ops import functional as F


def save_checkpoint(path: str, trainer, extra: Optional[dict] = None) -> None:
    if trainer.shard.rank != 0:
        return
    state = {
        "format_version": 1,
        "model": {k: v.cpu() for k, v in trainer.model.state_dict().items()},
        "optim": {
            "t": trainer.optimizer.t,
            "m": [t.cpu() for t in trainer.optimizer.m],
            "v": [t.cpu() for t in trainer.optimizer.v],
        },
        "epoch": trainer.epoch,
        "dropout_state": {k: v for k, v in F._DROPOUT_STATE.items()
                          if k != "counter"},
        "bounds": trainer.shard.bounds,
        "world_size": trainer.shard.world_size,
        "torch_rng": torch.get_rng_state(),
        "extra": extra or {},
    }
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)


def load_checkpoint(path: str, trainer) -> dict:
    state = torch.load(path, map_location="cpu", weights_only=False)
    trainer.model.load_state_dict(
        {k: v for k, v in state["model"].items()})
    trainer.model.to(trainer.device)
    opt = trainer.optimizer
    opt.t = state["optim"]["t"]
    for dst, src in zip(opt.m, state["optim"]["m"]):
        dst.copy_(src.to(dst.device))
    for dst, src in zip(opt.v, state["optim"]["v"]):
        dst.copy_(src.to(dst.device))
    trainer.epoch = state["epoch"]
    if getattr(trainer, "_step_dev", None) is not None:
        trainer._step_dev.fill_(opt.t)  # resync device schedule counter
    counter = F._DROPOUT_STATE.get("counter")
    F._DROPOUT_STATE.update(state["dropout_state"])
    F._DROPOUT_STATE["counter"] = counter  # device tensor is not persisted
    # the saved seed is RANK 0's (= the base seed); re-derive this
    # rank's decorrelated seed exactly as Trainer.__init__ does, so a
    # resumed multi-rank job keeps distinct per-rank dropout streams
    # (rank 0 resume stays bit-exact: base + 0*7919 == base)
    F._DROPOUT_STATE["seed"] = (state["dropout_state"]["seed"]
                                + trainer.shard.rank * 7919)
    torch.set_rng_state(state["torch_rng"])
    F.bump_weight_version()  # invalidate cached weight casts (in-place load)
    return state.get("extra", {})
