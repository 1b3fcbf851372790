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

import datetime
import json
import signal
import threading
import time
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


class RankFailure(RuntimeError):
    """A peer rank is believed dead; the caller should run
    ElasticAgent.rebuild() and retry the step from its last snapshot."""

    def __init__(self, dead: List[int]):
        super().__init__(f"dead ranks: {dead}")
        self.dead = dead


class ElasticAgent:
    """Failure detection + re-rendezvous — the torch.distributed analogue of
    the reference's MPI.ERRORS_RETURN + err_callback
    (/root/reference/VGG/allreducer.py:220,237, VGG/main_trainer.py:42-44).

    MPI with ERRORS_RETURN surfaces a dead peer as an error return from the
    collective and ULFM-style repair rebuilds the communicator; torch has
    neither, so the equivalent here is:

    * a side TCPStore (independent of any process group) carrying per-rank
      heartbeat counters from a daemon thread;
    * failure evidence = a collective timeout (set a short process-group
      timeout) OR a heartbeat stalled for > grace_s, confirmed by sampling
      the counters twice grace_s apart;
    * a race-free alarm: the first detector publishes the dead set with
      compare_set; every survivor — including ones already blocked in a
      doomed collective until its timeout fires — converges on the same
      alarm and calls rebuild();
    * rebuild() = destroy_process_group + init_process_group on a
      generation-prefixed view of the same store with the survivor ranks
      renumbered — the err_callback(new_nworkers, new_rank) moment; hand
      the returned Comm to apply_shrink().

    The store server lives in original rank 0 (host it externally for
    rank-0 fault tolerance — the reference has the same single point of
    failure in mpirun).  A falsely-suspected live rank finds itself in the
    dead set at its next check and must exit (fencing).

    The engine's EF residuals mutate before the failed collective, so the
    caller retries the step from a snapshot (params + optimizer/engine
    state_dict — the per-step cost is one device-side copy) or restores the
    last checkpoint; tests/test_elastic_checkpoint.py does the former.
    """

    def __init__(self, host: str, port: int, rank: int, world: int,
                 heartbeat_s: float = 0.25, grace_s: float = 2.0,
                 store: Optional["dist.Store"] = None):
        self.rank = rank
        self.ranks = list(range(world))  # original rank ids, current world
        self.generation = 0
        self.heartbeat_s = heartbeat_s
        self.grace_s = grace_s
        self.store = store if store is not None else dist.TCPStore(
            host, port, world, rank == 0, wait_for_workers=False)
        self._beat = 0
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._heartbeat_loop,
                                        daemon=True, name="oktopk-heartbeat")
        self._thread.start()

    # -- heartbeat ------------------------------------------------------
    def _heartbeat_loop(self) -> None:
        while not self._stop.is_set():
            self._beat += 1
            try:
                self.store.set(f"hb/{self.rank}", str(self._beat))
            except Exception:  # noqa: BLE001 — store server died with rank 0
                return
            self._stop.wait(self.heartbeat_s)

    def _read_beats(self) -> dict:
        out = {}
        for r in self.ranks:
            key = f"hb/{r}"
            out[r] = int(self.store.get(key)) if self.store.check([key]) else -1
        return out

    # -- detection ------------------------------------------------------
    def find_dead(self) -> List[int]:
        """Sample heartbeats twice grace_s apart; a rank whose counter did
        not advance is dead.  Call this after a collective timeout, or
        periodically from the training loop."""
        a = self._read_beats()
        time.sleep(self.grace_s)
        b = self._read_beats()
        return [r for r in self.ranks if r != self.rank and b[r] <= a[r]]

    def _alarm_key(self) -> str:
        return f"alarm/gen{self.generation}"

    def raise_alarm(self, dead: List[int]) -> List[int]:
        """Publish the dead set for this generation (first writer wins);
        returns the agreed set."""
        winner = self.store.compare_set(
            self._alarm_key(), "", json.dumps(sorted(dead)))
        return json.loads(winner)

    def check_alarm(self) -> Optional[List[int]]:
        """Non-blocking: the agreed dead set if any rank raised the alarm
        for the current generation."""
        if self.store.check([self._alarm_key()]):
            return json.loads(self.store.get(self._alarm_key()))
        return None

    # -- repair ---------------------------------------------------------
    def rebuild(self, dead: List[int], backend: Optional[str] = None,
                timeout_s: float = 60.0) -> Comm:
        """Collective among survivors: tear down the default process group
        and re-rendezvous generation g+1 on the shared store.  Returns the
        new Comm; the caller passes it to apply_shrink()."""
        if self.rank in dead:
            raise SystemExit("fenced: this rank was declared dead")
        backend = backend or (dist.get_backend() if dist.is_initialized() else "gloo")
        if dist.is_initialized():
            dist.destroy_process_group()
        survivors = [r for r in self.ranks if r not in dead]
        self.generation += 1
        new_rank = survivors.index(self.rank)
        prefix = dist.PrefixStore(f"gen{self.generation}", self.store)
        dist.init_process_group(
            backend, store=prefix, rank=new_rank, world_size=len(survivors),
            timeout=datetime.timedelta(seconds=timeout_s))
        self.ranks = survivors
        return Comm(dist.group.WORLD)

    def stop(self) -> None:
        self._stop.set()
        self._thread.join(timeout=5)


def snapshot_trainer(trainer) -> dict:
    """In-memory recovery point (model + optimizer incl. EF residuals +
    iteration) — what a failed step is retried from."""
    import copy

    return {
        "model": copy.deepcopy(trainer.model.state_dict()),
        "opt": copy.deepcopy(trainer.opt.state_dict()),
        "iteration": trainer.iteration,
    }


def restore_trainer(trainer, snap: dict) -> None:
    trainer.model.load_state_dict(snap["model"])
    trainer.opt.load_state_dict(snap["opt"])
    trainer.iteration = snap["iteration"]


class ElasticRunner:
    """Production step loop with rank-failure recovery (the shape proven in
    tests/test_elastic_checkpoint.py): every step runs against the newest
    snapshot; a collective error or heartbeat stall triggers alarm ->
    generation re-rendezvous -> restore -> apply_shrink -> retry.  Restore
    MUST precede apply_shrink so the snapshot's old-world region boundaries
    are reset for the new world size.  `snapshot_interval` > 1 amortises
    the snapshot cost for large models at the price of losing up to N-1
    steps of progress on failure (checkpoint semantics)."""

    def __init__(self, trainer, agent: ElasticAgent,
                 snapshot_interval: int = 1):
        self.trainer = trainer
        self.agent = agent
        self.snapshot_interval = max(1, snapshot_interval)
        self._snap = None
        self._since_snap = 0

    def step(self) -> float:
        tr = self.trainer
        dead = self.agent.check_alarm()
        loss = None
        if dead is None:
            if self._snap is None or self._since_snap >= self.snapshot_interval:
                self._snap = snapshot_trainer(tr)
                self._since_snap = 0
            try:
                loss = tr.step()
                self._since_snap += 1
            except RuntimeError:
                dead = self.agent.raise_alarm(self.agent.find_dead())
        if dead is not None:
            new_comm = self.agent.rebuild(dead)
            restore_trainer(tr, self._snap)
            apply_shrink(tr.opt.reducer, tr, new_comm)
            loss = tr.step()
            self._since_snap = 1
        return loss


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
