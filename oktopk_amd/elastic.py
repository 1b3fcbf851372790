"""Elastic shrink support.

Reference: MPI.ERRORS_RETURN + err_callback(new_nworkers, new_rank)
(VGG/allreducer.py:220,237, VGG/main_trainer.py:42-44) feeding
DLTrainer.update_nworker which rebuilds the data sampler for the smaller
world (VGG/dl_trainer.py:472-493); plus BERT's SLURM-preemption
save/requeue handlers (BERT/bert/main_bert.py:73-153).

torch.distributed has no in-place communicator repair, so the shrink is
cooperative: surviving ranks form a new (sub)group, swap it into the engine
with `shrink_comm`, and the per-P engine state (region boundaries) is reset
and re-derived on the next repartition interval.  SIGUSR1/SIGTERM
checkpoint-and-requeue is in `install_preemption_handler`.
"""
from __future__ import annotations

import signal
from typing import Callable, Iterable, List, Optional

import torch.distributed as dist

from .comm import Comm


def shrink_comm(surviving_ranks: List[int], backend: Optional[str] = None) -> Optional[Comm]:
    """All CURRENT ranks must call this collectively (new_group semantics).
    Returns the new Comm for survivors, None for ranks not in the new world."""
    group = dist.new_group(ranks=sorted(surviving_ranks), backend=backend)
    me = dist.get_rank()
    if me in surviving_ranks:
        return Comm(group)
    return None


def apply_shrink(reducer, trainer, new_comm: Comm) -> None:
    """Swap the communicator into a running engine + trainer
    (the err_callback body; reference err_callback -> update_nworker)."""
    reducer.set_comm(new_comm)
    if trainer is not None:
        trainer.update_nworker(new_comm)


def install_preemption_handler(save_fn: Callable[[], None],
                               signals: Iterable[int] = (signal.SIGUSR1, signal.SIGTERM)):
    """Checkpoint on SLURM preemption signals (reference
    BERT/bert/main_bert.py:73-153 saves interrupted state and requeues)."""
    def handler(signum, frame):
        save_fn()
        raise SystemExit(128 + signum)

    for s in signals:
        signal.signal(s, handler)
    return handler
