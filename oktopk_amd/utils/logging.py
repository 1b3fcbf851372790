"""Logging + metrics observability.

Reference: hostname-scoped logger with file+stream handlers
(VGG/settings.py:27-38) and tensorboardX scalar/histogram writing from rank 0
(VGG/dl_trainer.py:611-613).  tensorboard is not available offline, so
MetricWriter emits JSON-lines (one file per run) — the same scalars, greppable
and plottable without any dependency.
"""
from __future__ import annotations

import json
import logging
import os
import socket
import time
from typing import Optional


def get_logger(name: str = "oktopk_amd", logfile: Optional[str] = None) -> logging.Logger:
    logger = logging.getLogger(name)
    if logger.handlers:
        return logger
    logger.setLevel(logging.INFO)
    host = socket.gethostname()
    fmt = logging.Formatter(
        f"%(asctime)s [{host}] %(levelname)s %(name)s: %(message)s"
    )
    sh = logging.StreamHandler()
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    if logfile:
        os.makedirs(os.path.dirname(logfile) or ".", exist_ok=True)
        fh = logging.FileHandler(logfile)
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    return logger


class MetricWriter:
    """Rank-0 scalar writer (tensorboardX replacement): JSON lines of
    {wall, step, tag, value}."""

    def __init__(self, path: Optional[str], rank: int = 0):
        self.path = path if rank == 0 and path else None
        self._f = None
        if self.path:
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            self._f = open(self.path, "a", buffering=1)

    def add_scalar(self, tag: str, value: float, step: int) -> None:
        if self._f:
            self._f.write(
                json.dumps(
                    {"wall": time.time(), "step": step, "tag": tag, "value": float(value)}
                )
                + "\n"
            )

    def add_histogram(self, tag: str, values, step: int, bins: int = 16) -> None:
        """Weight-histogram parity with tensorboardX (reference
        dl_trainer.py:572-577): stores quantiles + moments as JSON."""
        if not self._f:
            return
        import torch

        v = values.detach().float().reshape(-1) if hasattr(values, "detach") else None
        if v is None or v.numel() == 0:
            return
        qs = torch.quantile(
            v, torch.linspace(0, 1, bins + 1, device=v.device)
        ).cpu().tolist()
        self._f.write(
            json.dumps(
                {
                    "wall": time.time(),
                    "step": step,
                    "tag": tag,
                    "hist_quantiles": [round(float(x), 6) for x in qs],
                    "mean": float(v.mean()),
                    "std": float(v.std()),
                }
            )
            + "\n"
        )

    def add_dict(self, scalars: dict, step: int) -> None:
        for k, v in scalars.items():
            if isinstance(v, (int, float)):
                self.add_scalar(k, v, step)

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None
