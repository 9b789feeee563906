"""Debug / health subsystems.

The reference prevented races by construction (Legion region coherence)
and had no failure handling beyond assert/exit. Here:

- sync-debug mode: synchronize + NaN/Inf-check the output of every
  roc_amd op right after it runs (catches async kernel faults at the
  faulting op, not 10 launches later) + torch autograd anomaly mode.
- loss health guard: Trainer.train_epoch raises TrainingDiverged when
  the fused-loss metrics go non-finite, so a driver can restore the
  last checkpoint and resume (utils.checkpoint).
"""
from __future__ import annotations

import torch

_SYNC_DEBUG = False


class TrainingDiverged(RuntimeError):
    pass


def enable_sync_debug(enable: bool = True) -> None:
    """Also set HIP_LAUNCH_BLOCKING=1 in the environment BEFORE process
    start for precise fault attribution."""
    global _SYNC_DEBUG
    _SYNC_DEBUG = enable
    torch.autograd.set_detect_anomaly(enable)


def sync_debug_enabled() -> bool:
    return _SYNC_DEBUG


def check_tensor(t: torch.Tensor, where: str) -> None:
    """Called by ops when sync-debug is on."""
    if not _SYNC_DEBUG:
        return
    if t.is_cuda:
        torch.cuda.synchronize(t.device)
    bad = (~torch.isfinite(t.float())).sum().item()
    if bad:
        raise TrainingDiverged(
            f"{where}: {bad}/{t.numel()} non-finite values")


def check_metrics(metrics: torch.Tensor) -> None:
    """Cheap divergence guard on the 8-float metrics vector."""
    m = metrics.detach()
    if not torch.isfinite(m).all():
        raise TrainingDiverged(
            f"non-finite training metrics: {m.cpu().tolist()}; "
            "restore the last checkpoint (utils.load_checkpoint) and "
            "lower the learning rate")
