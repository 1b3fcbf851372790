"""Checkpoint / resume.

Reference: VGG/dl_trainer.py:624-634 (per-epoch {iter, epoch, state_dict}
save by rank 0; the actual torch.save is commented out there),
BERT/bert/main_bert.py:207-219 (per-stage checkpoint.%d.pth.tar.epoch.%d),
and SLURM-interrupt save/restore (main_bert.py:73-153).

Improvement over the reference (SURVEY.md section 5 flags the gap): the
compressor's error-feedback residual state and thresholds ARE checkpointed —
both optimizers' state_dict() include reducer TensorState — so resume
continues the sparse-allreduce stream exactly.
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch


def checkpoint_path(directory: str, tag: str, epoch: Optional[int] = None,
                    stage: int = 0) -> str:
    name = f"checkpoint.{stage}.pth.tar"
    if epoch is not None:
        name += f".epoch.{epoch}"
    return os.path.join(directory, tag, name)


def save_checkpoint(
    path: str,
    model: torch.nn.Module,
    optimizer,
    iteration: int = 0,
    epoch: int = 0,
    rank: int = 0,
    extra: Optional[dict] = None,
) -> Optional[str]:
    """Rank 0 writes {iter, epoch, model, optimizer(+reducer residuals)}."""
    if rank != 0:
        return None
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    state = {
        "iter": iteration,
        "epoch": epoch,
        "state_dict": model.state_dict(),
        "optimizer": optimizer.state_dict() if optimizer is not None else None,
        "extra": extra or {},
    }
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)
    return path


def load_checkpoint(
    path: str,
    model: torch.nn.Module,
    optimizer=None,
    map_location="cpu",
) -> Tuple[int, int, dict]:
    """Returns (iteration, epoch, extra)."""
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state["state_dict"])
    if optimizer is not None:
        if state.get("optimizer") is not None:
            optimizer.load_state_dict(state["optimizer"])
        elif hasattr(optimizer, "sync_master_from_model"):
            # model-only restore: re-derive fp32 masters from loaded weights
            optimizer.sync_master_from_model()
    return int(state.get("iter", 0)), int(state.get("epoch", 0)), state.get("extra", {})
