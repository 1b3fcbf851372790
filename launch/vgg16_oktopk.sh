#!/bin/bash
# VGG-16 / CIFAR-shape Ok-Topk training on one MI355X node
# (parity with the reference SLURM launcher /root/reference/VGG/vgg16_oktopk.sh;
#  srun+mpi4py becomes torchrun+RCCL, 1 rank per GPU over xGMI).
set -e
source "$(dirname "$0")/exp_configs/vgg16.conf"
NGPUS=${NGPUS:-8}
density=${density:-0.02}
torchrun --nnodes=1 --nproc-per-node "$NGPUS" --master-addr 127.0.0.1 \
  -m oktopk_amd.train \
  --dnn "$dnn" --batch-size "$batch_size" --lr "$lr" \
  --compressor oktopk --density "$density" \
  --max-epochs "$max_epochs" --nsteps-update "$nstepsupdate"
