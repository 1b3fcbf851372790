"""Training CLI — `python -m oktopk_amd.train`.

Driver parity with the reference launchers:
  VGG/main_trainer.py:143-180 (argparse + robust_ssgd epoch loop, per-epoch
  checkpoint, images/sec logging, PROFILING_NORM EPS dumps) and
  BERT/bert/main_bert.py:641-765 (flag surface; SLURM env handled by
  torchrun env vars here).

Launch (mirrors the reference's srun/mpirun scripts — see launch/):
  torchrun --nnodes=1 --nproc-per-node 8 -m oktopk_amd.train \
      --dnn bert_base --compressor oktopk --density 0.001 ...
"""
from __future__ import annotations

import argparse
import os
import time

from .comm import init_from_env
from .config import EngineConfig
from .trainer import Trainer
from .utils import MetricWriter, get_logger, save_checkpoint, load_checkpoint
from .elastic import ElasticAgent, ElasticRunner, install_preemption_handler


def build_argparser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="oktopk_amd distributed trainer")
    # model/data (reference --dnn/--dataset/--max-epochs/--batch-size/--lr)
    p.add_argument("--dnn", "--model", dest="dnn", type=str, default="vgg16")
    p.add_argument("--dataset", type=str, default="synthetic",
                   help="synthetic | cifar10-shape | an4-shape | bert-shape")
    p.add_argument("--batch-size", type=int, default=16)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--lr", type=float, default=None)
    p.add_argument("--max-epochs", type=int, default=1)
    p.add_argument("--iters-per-epoch", type=int, default=50)
    p.add_argument("--nsteps-update", type=int, default=1,
                   help="gradient accumulation steps (reference --nsteps-update)")
    # sparsification (reference --compressor/--density/--sigma-scale)
    p.add_argument("--compressor", type=str, default="oktopk")
    p.add_argument("--density", type=float, default=0.02)
    p.add_argument("--sigma-scale", type=float, default=2.5,
                   help="accepted for reference-script compatibility; the "
                        "gaussian modes compute the exact normal-ppf "
                        "threshold (reference utils.py:136) so the manual "
                        "sigma-scale approximation is not needed")
    p.add_argument("--balanced-allgather", action="store_true",
                   help="oktopk round-2 load-balanced redistribution")
    p.add_argument("--pipeline-chunks", type=int, default=1,
                   help="chunked engine pipeline (docs/overlap_design.md)")
    p.add_argument("--dense-warmup", type=int, default=None,
                   help="dense allreduce iterations before sparsifying")
    # optimizer
    p.add_argument("--optimizer", type=str, default="auto", choices=["auto", "sgd", "adam"])
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    # checkpoints (reference --pretrain / per-epoch saves)
    p.add_argument("--checkpoint-dir", type=str, default="")
    p.add_argument("--pretrain", type=str, default="",
                   help="checkpoint path to resume from")
    # observability (reference settings.PROFILING / PROFILING_NORM, tensorboard)
    p.add_argument("--profiling", action="store_true")
    p.add_argument("--profiling-norm", action="store_true",
                   help="run the dense oracle alongside and log EPS")
    p.add_argument("--logdir", type=str, default="logs")
    # elastic failure recovery (reference MPI.ERRORS_RETURN + err_callback)
    p.add_argument("--elastic", action="store_true",
                   help="rank-failure detection + shrink-and-continue "
                        "(heartbeats over a side TCPStore; survivors "
                        "re-rendezvous and retry from an in-memory snapshot)")
    p.add_argument("--elastic-port", type=int, default=0,
                   help="side TCPStore port (default MASTER_PORT+17)")
    p.add_argument("--elastic-snapshot-interval", type=int, default=1)
    return p


def main(argv=None) -> int:
    args = build_argparser().parse_args(argv)
    comm = init_from_env()
    logger = get_logger(
        "oktopk_amd.train",
        os.path.join(args.logdir, f"rank{comm.rank}.log") if args.logdir else None,
    )
    writer = MetricWriter(
        os.path.join(args.logdir, "metrics.jsonl") if args.logdir else None,
        rank=comm.rank,
    )

    preset = ("bert" if args.dnn.startswith("bert")
              else ("lstm" if args.dnn.startswith("lstm") else "vgg"))
    overrides = dict(compressor=args.compressor, density=args.density,
                     profiling=args.profiling, profiling_norm=args.profiling_norm,
                     balanced_allgather=args.balanced_allgather,
                     pipeline_chunks=args.pipeline_chunks)
    if args.dense_warmup is not None:
        overrides["dense_warmup_iters"] = args.dense_warmup
    cfg = EngineConfig.preset(preset, **overrides)

    trainer = Trainer(
        model_name=args.dnn,
        batch_size=args.batch_size,
        seq_len=args.seq_len,
        comm=comm,
        cfg=cfg,
        optimizer=args.optimizer,
        lr=args.lr,
        dtype=args.dtype,
        nsteps_update=args.nsteps_update,
    )
    logger.info(
        "rank %d/%d model=%s compressor=%s density=%g params=%d",
        comm.rank, comm.size, args.dnn, args.compressor, args.density,
        sum(p.numel() for p in trainer.model.parameters()),
    )

    start_epoch = 0
    if args.pretrain:
        it, start_epoch, _ = load_checkpoint(args.pretrain, trainer.model, trainer.opt)
        trainer.iteration = it
        logger.info("resumed from %s at iter %d epoch %d", args.pretrain, it, start_epoch)

    if args.checkpoint_dir:
        install_preemption_handler(
            lambda: save_checkpoint(
                os.path.join(args.checkpoint_dir, "interrupted.pth"),
                trainer.model, trainer.opt, trainer.iteration, start_epoch,
                rank=comm.rank,
            )
        )

    runner = None
    if args.elastic and comm.size > 1:
        port = args.elastic_port or (
            int(os.environ.get("MASTER_PORT", "29500")) + 17)
        agent = ElasticAgent(os.environ.get("MASTER_ADDR", "127.0.0.1"),
                             port, comm.rank, comm.size)
        runner = ElasticRunner(trainer, agent,
                               args.elastic_snapshot_interval)
        logger.info("elastic recovery armed (store port %d)", port)

    samples_per_iter = args.batch_size * comm.size
    for epoch in range(start_epoch, args.max_epochs):
        trainer.set_epoch(epoch)
        t0 = time.time()
        for it in range(args.iters_per_epoch):
            loss = runner.step() if runner is not None else trainer.step()
            # comm may have shrunk mid-epoch: rank/size via the trainer
            comm = trainer.comm
            if it % 10 == 0 and comm.rank == 0:
                elapsed = time.time() - t0
                ips = samples_per_iter * (it + 1) / max(elapsed, 1e-9)
                logger.info("epoch %d iter %d loss %.4f  %.1f samples/s",
                            epoch, it, loss, ips)
                writer.add_scalar("train/loss", loss, trainer.iteration)
                writer.add_scalar("train/samples_per_s", ips, trainer.iteration)
        # EPS oracle dump (reference PROFILING_NORM norm files,
        # VGG/main_trainer.py:107-138)
        red = getattr(trainer.opt, "reducer", None)
        if args.profiling_norm and red is not None and comm.rank == 0:
            for step_i, eps in red.eps_log:
                writer.add_scalar("oracle/eps", eps, step_i)
            for step_i, rk in red.randk_log:
                writer.add_scalar("oracle/randk", rk, step_i)
            for step_i, ub in red.upbound_log:
                writer.add_scalar("oracle/upbound", ub, step_i)
            logger.info("EPS oracle: %d samples, last %.4f",
                        len(red.eps_log), red.eps_log[-1][1] if red.eps_log else -1)
        if args.profiling and red is not None and comm.rank == 0:
            for name, phases in red.timers.items():
                logger.info("timing[%s]: %s", name,
                            {k: round(v, 4) for k, v in phases.items()})
        if args.profiling and comm.rank == 0:
            mem = trainer.memory_stats()
            logger.info("gpu-mem: %s", mem)
            writer.add_dict({f"mem/{k}": v for k, v in mem.items()},
                            trainer.iteration)
        if args.checkpoint_dir:
            path = os.path.join(args.checkpoint_dir, f"checkpoint.epoch.{epoch}.pth")
            save_checkpoint(path, trainer.model, trainer.opt, trainer.iteration,
                            epoch, rank=comm.rank)
            if comm.rank == 0:
                logger.info("saved %s", path)
        comm.barrier()
    writer.close()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
